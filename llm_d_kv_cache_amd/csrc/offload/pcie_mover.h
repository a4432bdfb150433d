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

#include <atomic>
#include <chrono>
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

  // Async submit: the returned future resolves (or rethrows) when the copy
  // has completed on the wire. Same-direction requests execute FIFO on one
  // stream, so chunked callers can overlap file I/O of chunk i with the
  // copy of chunk i+1.
  std::future<void> d2h_async(void* dst, const void* src, size_t n,
                              hipEvent_t pre_event = nullptr) {
    return submit(lanes_[0], dst, src, n, pre_event);
  }
  std::future<void> h2d_async(void* dst, const void* src, size_t n) {
    return submit(lanes_[1], dst, src, n, nullptr);
  }

  // Blocking conveniences.
  void d2h(void* dst, const void* src, size_t n, hipEvent_t pre_event) {
    d2h_async(dst, src, n, pre_event).get();
  }
  void h2d(void* dst, const void* src, size_t n) {
    h2d_async(dst, src, n).get();
  }

  // Lane utilization (busy = wall spent inside waitEvent+memcpy+sync, i.e.
  // everything but sitting on an empty queue): the direct measure of
  // whether the wire is fed. Index 0 = D2H, 1 = H2D.
  struct LaneStats {
    double busy_s = 0;
    uint64_t bytes = 0;
    uint64_t requests = 0;
  };
  LaneStats lane_stats(int dir) const {
    LaneStats s;
    s.busy_s = lanes_[dir].busy_ns.load(std::memory_order_relaxed) * 1e-9;
    s.bytes = lanes_[dir].bytes.load(std::memory_order_relaxed);
    s.requests = lanes_[dir].requests.load(std::memory_order_relaxed);
    return s;
  }

 private:
  struct Req {
    void* dst;
    const void* src;
    size_t n;
    hipEvent_t pre;
    std::promise<void> done;
  };
  struct Lane {
    std::mutex mu;
    std::condition_variable cv;
    std::deque<std::unique_ptr<Req>> q;
    bool stopping = false;
    std::thread thread;
    std::atomic<uint64_t> busy_ns{0}, bytes{0}, requests{0};
  };

  std::future<void> submit(Lane& lane, void* dst, const void* src, size_t n,
                           hipEvent_t pre) {
    auto req = std::make_unique<Req>();
    req->dst = dst;
    req->src = src;
    req->n = n;
    req->pre = pre;
    auto fut = req->done.get_future();
    {
      std::lock_guard<std::mutex> g(lane.mu);
      lane.q.push_back(std::move(req));
    }
    lane.cv.notify_one();
    return fut;
  }

  void lane_loop(Lane& lane, hipStream_t stream, hipMemcpyKind kind) {
    for (;;) {
      std::unique_ptr<Req> req;
      {
        std::unique_lock<std::mutex> g(lane.mu);
        lane.cv.wait(g, [&] { return lane.stopping || !lane.q.empty(); });
        if (lane.q.empty()) return;
        req = std::move(lane.q.front());
        lane.q.pop_front();
      }
      auto b0 = std::chrono::steady_clock::now();
      try {
        if (req->pre) KVO_HIP_CHECK(hipStreamWaitEvent(stream, req->pre, 0));
        KVO_HIP_CHECK(hipMemcpyAsync(req->dst, req->src, req->n, kind, stream));
        KVO_HIP_CHECK(hipStreamSynchronize(stream));
        req->done.set_value();
      } catch (...) {
        req->done.set_exception(std::current_exception());
      }
      lane.busy_ns.fetch_add(
          std::chrono::duration_cast<std::chrono::nanoseconds>(
              std::chrono::steady_clock::now() - b0)
              .count(),
          std::memory_order_relaxed);
      lane.bytes.fetch_add(req->n, std::memory_order_relaxed);
      lane.requests.fetch_add(1, std::memory_order_relaxed);
    }
  }

  bool gpu_mode_;
  Lane lanes_[2];  // 0 = D2H, 1 = H2D
};

}  // namespace kvo
