// PcieMover: serialized PCIe hops, one dedicated stream per direction.
//
// Measured on MI355X (tools/copy_probe.hip, profiles/): a single-stream
// hipMemcpyAsync D2H/H2D runs on SDMA at ~55-57 GB/s (the PCIe Gen5 wire
// limit), but CONCURRENT copies from multiple threads/streams collapse to
// ~22 GB/s aggregate blit-kernel copies that also occupy CUs. One stream
// per direction is therefore optimal: it saturates the wire, keeps D2H/H2D
// full-duplex on separate SDMA engines, and leaves every CU to the serving
// engine. Workers submit and block on a per-request future; ordering
// against producer kernels rides an event recorded on the worker stream.
#pragma once

#include <condition_variable>
#include <deque>
#include <future>
#include <mutex>
#include <thread>

#include "common.h"

namespace kvo {

class PcieMover {
 public:
  explicit PcieMover(bool gpu_mode, int device) : gpu_mode_(gpu_mode) {
    if (!gpu_mode_) return;
    for (int dir = 0; dir < 2; ++dir) {
      lanes_[dir].thread = std::thread([this, dir, device] {
        KVO_HIP_CHECK(hipSetDevice(device));
        hipStream_t stream;
        KVO_HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
        lane_loop(lanes_[dir], stream,
                  dir == 0 ? hipMemcpyDeviceToHost : hipMemcpyHostToDevice);
        (void)hipStreamDestroy(stream);
      });
    }
  }

  ~PcieMover() {
    if (!gpu_mode_) return;
    for (auto& lane : lanes_) {
      {
        std::lock_guard<std::mutex> g(lane.mu);
        lane.stopping = true;
      }
      lane.cv.notify_all();
      lane.thread.join();
    }
  }

  // Blocking: device -> pinned host. pre_event (optional) is waited on the
  // mover stream before the copy (producer-kernel ordering without a host
  // sync on the worker).
  void d2h(void* dst, const void* src, size_t n, hipEvent_t pre_event) {
    submit(lanes_[0], dst, src, n, pre_event);
  }

  // Blocking: pinned host -> device.
  void h2d(void* dst, const void* src, size_t n) {
    submit(lanes_[1], dst, src, n, nullptr);
  }

 private:
  struct Req {
    void* dst;
    const void* src;
    size_t n;
    hipEvent_t pre;
    std::promise<void> done;
    std::exception_ptr error;
  };
  struct Lane {
    std::mutex mu;
    std::condition_variable cv;
    std::deque<Req*> q;
    bool stopping = false;
    std::thread thread;
  };

  void submit(Lane& lane, void* dst, const void* src, size_t n, hipEvent_t pre) {
    Req req;
    req.dst = dst;
    req.src = src;
    req.n = n;
    req.pre = pre;
    auto fut = req.done.get_future();
    {
      std::lock_guard<std::mutex> g(lane.mu);
      lane.q.push_back(&req);
    }
    lane.cv.notify_one();
    fut.get();
    if (req.error) std::rethrow_exception(req.error);
  }

  void lane_loop(Lane& lane, hipStream_t stream, hipMemcpyKind kind) {
    for (;;) {
      Req* req = nullptr;
      {
        std::unique_lock<std::mutex> g(lane.mu);
        lane.cv.wait(g, [&] { return lane.stopping || !lane.q.empty(); });
        if (lane.q.empty()) return;
        req = lane.q.front();
        lane.q.pop_front();
      }
      try {
        if (req->pre) KVO_HIP_CHECK(hipStreamWaitEvent(stream, req->pre, 0));
        KVO_HIP_CHECK(hipMemcpyAsync(req->dst, req->src, req->n, kind, stream));
        KVO_HIP_CHECK(hipStreamSynchronize(stream));
      } catch (...) {
        req->error = std::current_exception();
      }
      req->done.set_value();
    }
  }

  bool gpu_mode_;
  Lane lanes_[2];  // 0 = D2H, 1 = H2D
};

}  // namespace kvo
