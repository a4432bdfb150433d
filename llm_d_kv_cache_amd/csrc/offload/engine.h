// StorageOffloadEngine: the KV-block offload job orchestrator.
//
// Capability parity with the reference StorageOffloadEngine
// (csrc/storage/storage_offload.cpp): async store/load of GPU KV blocks to
// content-addressed files, KV-ready event fencing against the serving
// engine's stream, per-file tasks on the NUMA-pinned I/O pool (loads HIGH
// priority), skip-if-exists dedupe with atime touch, EMA-driven dynamic
// write-queue limit with write dropping, job cancellation (preemption), and
// get_finished() polling.
//
// MI355X-first redesign of the copy path: the default "staged" mode runs
// the CDNA4 gather kernel HBM->HBM into a contiguous device bounce (brief,
// bandwidth-bound CU usage) and crosses PCIe with hipMemcpyAsync on the
// SDMA engines — zero CU occupancy while the wire is busy, so serving
// compute is undisturbed. "zero_copy" mode instead has the kernel write
// device-mapped pinned host memory directly (the reference's only mode).
// Constructed against host tensors the same control logic runs on memcpy
// ("host" mode, CPU-only CI) — an explicit mode, never a silent fallback.
#pragma once

#include <algorithm>
#include <array>
#include <cmath>
#include <chrono>
#include <functional>
#include <thread>
#include <list>
#include <map>
#include <unordered_map>

#include "common.h"
#include "file_io.h"
#include "host_cache.h"
#include "pcie_mover.h"
#include "thread_pool.h"

// Launchers take host block ids (by-value kernarg, <=128) OR a device
// pointer (ids_dev != nullptr) for bigger transfers — no kernel-side
// blocks-per-file ceiling.
extern "C" hipError_t kvc_launch_gather(const void* const*, const uint64_t*, int,
                                        uint64_t, const int32_t*, int,
                                        const int32_t*, uint8_t*, hipStream_t);
extern "C" hipError_t kvc_launch_scatter(const void* const*, const uint64_t*, int,
                                         uint64_t, const int32_t*, int,
                                         const int32_t*, const uint8_t*,
                                         hipStream_t);
extern "C" hipError_t kvc_launch_gather_fp8(const void* const*, const uint64_t*,
                                            int, uint64_t, const int32_t*, int,
                                            const int32_t*, uint8_t*,
                                            hipStream_t);
extern "C" hipError_t kvc_launch_gather_fp8_split(
    const void* const*, const uint64_t*, int, uint64_t, const int32_t*, int,
    const int32_t*, uint8_t*, float*, hipStream_t);
extern "C" hipError_t kvc_launch_scatter_fp8(const void* const*, const uint64_t*,
                                             int, uint64_t, const int32_t*, int,
                                             const int32_t*, const uint8_t*,
                                             hipStream_t);

namespace kvo {

constexpr int kMaxBlocksPerFileHost = 128;  // mirrors kernels.hip kMaxBlocksPerFile
constexpr int kMaxBlocksPerFileDev = 4096;  // via per-worker device id buffer

// Intra-transfer pipeline granularity: the PCIe hop and the file I/O of one
// transfer overlap at this chunk size (copy chunk i+1 rides the SDMA stream
// while chunk i is written/was read). Transfers at or below 64 MiB run
// single-shot: with 16 workers the pipeline across tasks already overlaps
// file I/O with the wire, and measured intra-task chunking at 32 MiB files
// only added memory-traffic contention (read-side accumulated time 3x).
constexpr size_t kPipelineChunkBytes = 16ull << 20;

inline size_t pipeline_chunk(size_t bytes) {
  return bytes <= 4 * kPipelineChunkBytes ? bytes : kPipelineChunkBytes;
}

enum class CopyPath { kStaged, kZeroCopy, kHostMemcpy };

// Payload serialization: raw bytes, or fp8 e4m3fn quantization of bf16
// pages (halves wire + storage bytes; one f32 scale per (block, layer)
// tile appended to each tile record).
enum class Serialize { kRaw, kFp8E4M3 };

// Store completion semantics: kThrough = the file rename happened (crash
// durable); kBack = the slab is resident in the pinned-DRAM tier and the
// file flush runs asynchronously (store latency ~= gather + D2H; loses
// only not-yet-flushed cache entries on crash — acceptable for a cache).
// kBack needs host_cache_bytes > 0; transfers that cannot get a slot fall
// back to write-through.
enum class WritePolicy { kThrough, kBack };

struct GroupDesc {
  std::vector<void*> layer_ptrs;        // per-layer block-0 base address
  std::vector<uint64_t> layer_strides;  // bytes between consecutive blocks
  uint64_t block_bytes = 0;             // bytes per (block, layer)
  int64_t num_blocks = 0;               // device pages per layer (0 = unchecked)
};

struct EngineConfig {
  int io_threads = 16;
  int gpu_blocks_per_file = 16;
  double read_preferring_ratio = 0.75;
  double max_write_queued_seconds = 30.0;
  bool gpu_mode = false;
  int device = 0;
  CopyPath copy_path = CopyPath::kStaged;
  Serialize serialize = Serialize::kRaw;
  // Pinned host-DRAM cache tier (0 = disabled): write-through on store,
  // cache-hit loads skip the filesystem (host_cache.h).
  size_t host_cache_bytes = 0;
  WritePolicy write_policy = WritePolicy::kThrough;
  // O_DIRECT file I/O (page-cache bypass for local NVMe; auto-fallback on
  // unsupported filesystems/unaligned transfers) — the GDS-substitute
  // mode (no cuFile exists on ROCm; see docs/architecture.md).
  bool direct_io = false;
};

struct FileTransfer {
  int group = 0;
  std::string path;
  std::vector<int32_t> block_ids;
  // Block-slot offset of this span within the file's nominal layout. Stores
  // emit slot_offset == 0 (head-partial files are simply short); loads may
  // skip an already-cached head.
  int slot_offset = 0;
};

struct FinishedJob {
  int64_t id = 0;
  bool success = false;
  bool dropped = false;
};

struct EngineStats {
  uint64_t stores_submitted = 0;
  uint64_t loads_submitted = 0;
  uint64_t files_written = 0;
  uint64_t files_deduped = 0;
  uint64_t files_read = 0;
  uint64_t writes_dropped = 0;
  uint64_t tasks_cancelled = 0;
  uint64_t host_cache_hits = 0;
  uint64_t host_cache_stores = 0;
  uint64_t writeback_flushes = 0;
  uint64_t errors = 0;
  double avg_write_seconds = 0;
  uint64_t bytes_stored = 0;
  uint64_t bytes_loaded = 0;
  // accumulated per-phase wall time across all workers (ms)
  double t_gather_ms = 0, t_d2h_ms = 0, t_write_ms = 0;
  double t_read_ms = 0, t_h2d_ms = 0, t_scatter_ms = 0;
  // PCIe lane utilization (mover-side truth: busy wall per direction)
  double d2h_lane_busy_ms = 0, h2d_lane_busy_ms = 0;
  uint64_t d2h_lane_bytes = 0, h2d_lane_bytes = 0;
};

class StorageOffloadEngine {
 public:
  StorageOffloadEngine(EngineConfig cfg, std::vector<GroupDesc> groups)
      : cfg_(cfg), groups_(std::move(groups)) {
    if (cfg_.gpu_blocks_per_file > kMaxBlocksPerFileDev)
      throw std::invalid_argument(
          "gpu_blocks_per_file exceeds engine limit (4096)");
    if (cfg_.gpu_mode) {
      // Transient init failures have been observed right after another
      // process released the device: retry briefly before giving up.
      int count = 0;
      hipError_t err = hipSuccess;
      for (int attempt = 0; attempt < 10; ++attempt) {
        err = hipGetDeviceCount(&count);
        if (err == hipSuccess && count > cfg_.device) break;
        std::this_thread::sleep_for(std::chrono::milliseconds(200));
      }
      if (err != hipSuccess || count <= cfg_.device)
        throw HipError(std::string("gpu_mode requested but device ") +
                       std::to_string(cfg_.device) + " unavailable: " +
                       hipGetErrorString(err) + " (count=" +
                       std::to_string(count) + ")");
      KVO_HIP_CHECK(hipSetDevice(cfg_.device));
    } else {
      cfg_.copy_path = CopyPath::kHostMemcpy;
    }

    size_t max_file_bytes = 0;
    for (auto& g : groups_) {
      if (g.block_bytes % 16 != 0)
        throw std::invalid_argument("block_bytes must be a multiple of 16");
      if (g.layer_ptrs.size() != g.layer_strides.size())
        throw std::invalid_argument("layer ptr/stride length mismatch");
      max_file_bytes =
          std::max(max_file_bytes, static_cast<size_t>(cfg_.gpu_blocks_per_file) *
                                       g.layer_ptrs.size() * g.block_bytes);
      if (cfg_.gpu_mode) {
        void* dp = nullptr;
        KVO_HIP_CHECK(hipMalloc(&dp, g.layer_ptrs.size() * sizeof(void*)));
        KVO_HIP_CHECK(hipMemcpy(dp, g.layer_ptrs.data(),
                                g.layer_ptrs.size() * sizeof(void*),
                                hipMemcpyHostToDevice));
        dev_layer_ptrs_.push_back(static_cast<void**>(dp));
        void* ds = nullptr;
        KVO_HIP_CHECK(hipMalloc(&ds, g.layer_strides.size() * sizeof(uint64_t)));
        KVO_HIP_CHECK(hipMemcpy(ds, g.layer_strides.data(),
                                g.layer_strides.size() * sizeof(uint64_t),
                                hipMemcpyHostToDevice));
        dev_layer_strides_.push_back(static_cast<uint64_t*>(ds));
      }
    }

    size_t device_staging = cfg_.copy_path == CopyPath::kStaged ? max_file_bytes : 0;
    bool mapped = cfg_.copy_path == CopyPath::kZeroCopy;
    pool_ = std::make_unique<IoThreadPool>(
        cfg_.io_threads, cfg_.gpu_mode, cfg_.device, max_file_bytes, device_staging,
        cfg_.read_preferring_ratio, mapped);
    if (cfg_.gpu_mode && cfg_.copy_path == CopyPath::kStaged)
      mover_ = std::make_unique<PcieMover>(true, cfg_.device);
    if (cfg_.host_cache_bytes > 0 && cfg_.copy_path != CopyPath::kZeroCopy &&
        max_file_bytes > 0)
      cache_ = std::make_unique<HostPinnedCache>(cfg_.host_cache_bytes,
                                                 max_file_bytes, cfg_.gpu_mode);
  }

  ~StorageOffloadEngine() {
    pool_.reset();  // drain workers before freeing device arrays
    for (auto p : dev_layer_ptrs_) (void)hipFree(p);
    for (auto p : dev_layer_strides_) (void)hipFree(p);
  }

  const EngineConfig& config() const { return cfg_; }

  // Peer/DRAM-tier bridge: copy a resident host-cache entry for `path` into
  // caller memory so a peer GPU can be served over xGMI even after the
  // blocks left HBM. Returns payload bytes on hit, -1 on miss, -2 when cap
  // is too small. The copy (vs handing out the slot) decouples the pinned
  // slab's lifetime from the caller's async send.
  int64_t host_cache_read(const std::string& path, uint8_t* dst, size_t cap) {
    if (!cache_) return -1;
    HostPinnedCache::Slot* s = cache_->lookup(path);
    if (s == nullptr) return -1;
    const size_t n = s->bytes_used;
    if (n > cap) {
      cache_->release(s);
      return -2;
    }
    std::memcpy(dst, s->buf->host(), n);
    cache_->release(s);
    return static_cast<int64_t>(n);
  }

  // ---- store ----------------------------------------------------------------

  int64_t async_store(std::vector<FileTransfer> files, uintptr_t caller_stream) {
    // Validate EVERYTHING before any task is enqueued: a throw mid-list
    // after partial enqueue would leave job->remaining never reaching 0
    // (wait_job deadlock, pending_writes_ leak).
    for (const auto& ft : files) validate(ft);
    auto job = new_job(files.size(), /*is_store=*/true);
    if (files.empty()) {
      complete_empty(job);
      return job->id;
    }

    // Dynamic write-queue limit (EMA of write duration): when the backlog
    // exceeds threads * max_queued_seconds / avg_write_seconds, the whole
    // store is dropped — offload is a cache, dropping is always safe.
    {
      double avg = avg_write_seconds();
      if (avg > 0) {
        size_t limit = static_cast<size_t>(
            pool_->size() * cfg_.max_write_queued_seconds / avg);
        if (pending_writes_.load(std::memory_order_relaxed) > limit) {
          stats_inc([&](EngineStats& s) { s.writes_dropped += files.size(); });
          std::lock_guard<std::mutex> g(jobs_mu_);
          job->remaining = 0;
          finished_.push_back({job->id, true, /*dropped=*/true});
          jobs_.erase(job->id);
          done_cv_.notify_all();
          return job->id;
        }
      }
      pending_writes_.fetch_add(files.size(), std::memory_order_relaxed);
    }

    if (cfg_.gpu_mode) {
      KVO_HIP_CHECK(hipEventCreateWithFlags(&job->kv_ready, hipEventDisableTiming));
      KVO_HIP_CHECK(hipEventRecord(job->kv_ready,
                                   reinterpret_cast<hipStream_t>(caller_stream)));
    }
    stats_inc([&](EngineStats& s) { s.stores_submitted += files.size(); });

    for (auto& ft : files) {
      pool_->enqueue(Priority::kNormal, [this, job, ft](WorkerCtx& ctx) {
        bool ok = true;
        double t0 = now_s();
        bool wrote = false;
        bool deferred = false;
        try {
          if (!job->cancelled.load(std::memory_order_acquire)) {
            if (file_exists(ft.path)) {
              touch_atime(ft.path);
              stats_inc([](EngineStats& s) { s.files_deduped++; });
            } else {
              deferred = store_one(ctx, job, ft);
              wrote = !deferred;
            }
          } else {
            stats_inc([](EngineStats& s) { s.tasks_cancelled++; });
          }
        } catch (const std::exception& e) {
          KVO_LOG_ERROR("store %s failed: %s", ft.path.c_str(), e.what());
          stats_inc([](EngineStats& s) { s.errors++; });
          ok = false;
        }
        if (deferred) return;  // write continuation owns EMA/pending/done
        if (wrote) update_write_ema(now_s() - t0);
        pending_writes_.fetch_sub(1, std::memory_order_relaxed);
        task_done(job, ok);
      });
    }
    return job->id;
  }

  // ---- load -----------------------------------------------------------------

  int64_t async_load(std::vector<FileTransfer> files) {
    for (const auto& ft : files) validate(ft);  // before ANY enqueue (see store)
    auto job = new_job(files.size(), /*is_store=*/false);
    if (files.empty()) {
      complete_empty(job);
      return job->id;
    }
    stats_inc([&](EngineStats& s) { s.loads_submitted += files.size(); });
    for (auto& ft : files) {
      pool_->enqueue(Priority::kHigh, [this, job, ft](WorkerCtx& ctx) {
        bool ok = true;
        try {
          if (!job->cancelled.load(std::memory_order_acquire)) {
            load_one(ctx, ft);
          } else {
            stats_inc([](EngineStats& s) { s.tasks_cancelled++; });
          }
        } catch (const std::exception& e) {
          KVO_LOG_ERROR("load %s failed: %s", ft.path.c_str(), e.what());
          stats_inc([](EngineStats& s) { s.errors++; });
          ok = false;
        }
        task_done(job, ok);
      });
    }
    return job->id;
  }

  // ---- completion -----------------------------------------------------------

  std::vector<FinishedJob> get_finished() {
    std::lock_guard<std::mutex> g(jobs_mu_);
    std::vector<FinishedJob> out;
    out.swap(finished_);
    return out;
  }

  // Non-blocking cancel: queued tasks of the job bail when dequeued.
  // Returns false when the job already finished. Preemption sets flags for
  // every affected job FIRST, then waits — a combined cancel+wait per job
  // always loses the race against the worker's next dequeue.
  bool cancel_job(int64_t id) {
    std::lock_guard<std::mutex> g(jobs_mu_);
    auto it = jobs_.find(id);
    if (it == jobs_.end()) return false;
    it->second->cancelled.store(true, std::memory_order_release);
    return true;
  }

  // Cancel a job's queued tasks (preemption) and wait for in-flight ones.
  // Returns true when every task that DID run succeeded.
  bool wait_job(int64_t id) {
    std::shared_ptr<Job> job;
    {
      std::lock_guard<std::mutex> g(jobs_mu_);
      auto it = jobs_.find(id);
      if (it == jobs_.end()) return true;  // already finished
      job = it->second;
    }
    job->cancelled.store(true, std::memory_order_release);
    std::unique_lock<std::mutex> g(jobs_mu_);
    done_cv_.wait(g, [&] { return job->remaining.load() == 0; });
    return job->success.load();
  }

  EngineStats stats() {
    EngineStats out;
    for (auto& sh : stats_shards_) {
      std::lock_guard<std::mutex> g(sh.mu);
      const EngineStats& s = sh.s;
      out.stores_submitted += s.stores_submitted;
      out.loads_submitted += s.loads_submitted;
      out.files_written += s.files_written;
      out.files_deduped += s.files_deduped;
      out.files_read += s.files_read;
      out.writes_dropped += s.writes_dropped;
      out.tasks_cancelled += s.tasks_cancelled;
      out.host_cache_hits += s.host_cache_hits;
      out.host_cache_stores += s.host_cache_stores;
      out.writeback_flushes += s.writeback_flushes;
      out.errors += s.errors;
      out.bytes_stored += s.bytes_stored;
      out.bytes_loaded += s.bytes_loaded;
      out.t_gather_ms += s.t_gather_ms;
      out.t_d2h_ms += s.t_d2h_ms;
      out.t_write_ms += s.t_write_ms;
      out.t_read_ms += s.t_read_ms;
      out.t_h2d_ms += s.t_h2d_ms;
      out.t_scatter_ms += s.t_scatter_ms;
    }
    out.avg_write_seconds = avg_write_seconds();
    if (mover_) {
      auto d = mover_->lane_stats(0), h = mover_->lane_stats(1);
      out.d2h_lane_busy_ms = d.busy_s * 1e3;
      out.h2d_lane_busy_ms = h.busy_s * 1e3;
      out.d2h_lane_bytes = d.bytes;
      out.h2d_lane_bytes = h.bytes;
    }
    return out;
  }

  size_t pending_writes() {
    return pending_writes_.load(std::memory_order_relaxed);
  }

 private:
  struct Job {
    int64_t id;
    std::atomic<int> remaining{0};
    std::atomic<bool> cancelled{false};
    std::atomic<bool> success{true};
    hipEvent_t kv_ready = nullptr;
  };

  void validate(const FileTransfer& ft) {
    if (ft.group < 0 || ft.group >= static_cast<int>(groups_.size()))
      throw std::invalid_argument("bad group index");
    if (ft.block_ids.empty() ||
        ft.block_ids.size() > static_cast<size_t>(cfg_.gpu_blocks_per_file))
      throw std::invalid_argument("bad block count for file transfer");
    int64_t nb = groups_[ft.group].num_blocks;
    if (nb > 0) {
      for (int32_t id : ft.block_ids)
        if (id < 0 || id >= nb)
          throw std::invalid_argument(
              "block id " + std::to_string(id) + " out of range [0, " +
              std::to_string(nb) + ")");
    }
  }

  void complete_empty(const std::shared_ptr<Job>& job) {
    std::lock_guard<std::mutex> g(jobs_mu_);
    finished_.push_back({job->id, true, false});
    jobs_.erase(job->id);
    done_cv_.notify_all();
  }

  std::shared_ptr<Job> new_job(size_t n_tasks, bool) {
    auto job = std::make_shared<Job>();
    std::lock_guard<std::mutex> g(jobs_mu_);
    job->id = next_job_id_++;
    job->remaining = static_cast<int>(n_tasks);
    jobs_[job->id] = job;
    return job;
  }

  void task_done(const std::shared_ptr<Job>& job, bool ok) {
    if (!ok) job->success.store(false, std::memory_order_relaxed);
    if (job->remaining.fetch_sub(1, std::memory_order_acq_rel) == 1) {
      if (job->kv_ready) {
        (void)hipEventDestroy(job->kv_ready);
        job->kv_ready = nullptr;
      }
      std::lock_guard<std::mutex> g(jobs_mu_);
      finished_.push_back({job->id, job->success.load(), false});
      jobs_.erase(job->id);
      done_cv_.notify_all();
    }
  }

  // Per-thread-sharded stats: phase-timing updates run ~6x per task from
  // every worker, so they must not contend on the job-completion mutex
  // (or each other). Each worker thread maps to one shard; stats()
  // aggregates.
  static constexpr size_t kStatsShards = 32;
  struct alignas(64) StatsShard {
    std::mutex mu;
    EngineStats s;
  };

  template <typename F>
  void stats_inc(F f) {
    thread_local size_t idx =
        std::hash<std::thread::id>{}(std::this_thread::get_id()) % kStatsShards;
    auto& sh = stats_shards_[idx];
    std::lock_guard<std::mutex> g(sh.mu);
    f(sh.s);
  }

  double avg_write_seconds() const {
    uint64_t bits = write_ema_bits_.load(std::memory_order_relaxed);
    double v;
    std::memcpy(&v, &bits, 8);
    return v;
  }

  void update_write_ema(double dt) {
    // EMA alpha=0.05 of write duration drives the queue limit; lock-free
    // CAS keeps it off the completion mutex.
    uint64_t old_bits = write_ema_bits_.load(std::memory_order_relaxed);
    for (;;) {
      double old_v;
      std::memcpy(&old_v, &old_bits, 8);
      double new_v = old_v == 0 ? dt : 0.95 * old_v + 0.05 * dt;
      uint64_t new_bits;
      std::memcpy(&new_bits, &new_v, 8);
      if (write_ema_bits_.compare_exchange_weak(old_bits, new_bits,
                                                std::memory_order_relaxed))
        return;
    }
  }

  size_t tile_record_bytes(const GroupDesc& g) const {
    return cfg_.serialize == Serialize::kFp8E4M3 ? g.block_bytes / 2 + 4
                                                 : g.block_bytes;
  }

  size_t file_bytes(const GroupDesc& g, size_t n_blocks) const {
    return n_blocks * g.layer_ptrs.size() * tile_record_bytes(g);
  }

  // Two-stage store A/B switch (KVC_TWO_STAGE=1 opts in). Measured SLOWER
  // than the serial per-worker D2H->write flow on every box tried (same-
  // box: 52.0 vs 55.4 GB/s at 32 workers, 42.9 vs 49.1 at 16): splitting
  // the write off starves the D2H queue worse than the write blocking the
  // worker does, because submission — not lane service — is the scarce
  // resource. Kept for future re-measurement; default off.
  static bool two_stage_stores() {
    static const bool on = [] {
      const char* e = std::getenv("KVC_TWO_STAGE");
      return e != nullptr && e[0] == '1';
    }();
    return on;
  }

  // Block ids above the kernarg limit ride the worker's device buffer.
  const int32_t* stage_ids(WorkerCtx& ctx, const FileTransfer& ft) {
    if (ft.block_ids.size() <= static_cast<size_t>(kMaxBlocksPerFileHost))
      return nullptr;
    KVO_HIP_CHECK(hipMemcpyAsync(ctx.dev_ids, ft.block_ids.data(),
                                 ft.block_ids.size() * sizeof(int32_t),
                                 hipMemcpyHostToDevice, ctx.stream));
    return ctx.dev_ids;
  }

  // Returns true when completion was DEFERRED to a write-continuation task
  // (two-stage store): the caller must then skip EMA/pending/task_done.
  bool store_one(WorkerCtx& ctx, const std::shared_ptr<Job>& job,
                 const FileTransfer& ft) {
    const GroupDesc& g = groups_[ft.group];
    const int nb = static_cast<int>(ft.block_ids.size());
    const int nl = static_cast<int>(g.layer_ptrs.size());
    const size_t bytes = file_bytes(g, nb);

    HostPinnedCache::Slot* slot =
        cache_ ? cache_->acquire(ft.path, bytes) : nullptr;
    uint8_t* host_buf = slot ? slot->buf->host() : ctx.host_staging->host();
    struct SlotGuard {
      HostPinnedCache* c;
      HostPinnedCache::Slot* s;
      bool ok = false;
      ~SlotGuard() {
        if (!c || !s) return;
        if (ok) {
          c->publish(s);
          c->release(s);
        } else {
          c->abandon(s);
        }
      }
    } guard{cache_.get(), slot};

    if (cfg_.copy_path == CopyPath::kHostMemcpy) {
      gather_host(g, ft.block_ids, host_buf);
      if (cfg_.write_policy == WritePolicy::kBack && slot != nullptr) {
        guard.ok = true;
        cache_->addref(slot);
        auto* cache = cache_.get();
        std::string path = ft.path;
        pool_->enqueue(Priority::kCont,
                       [this, cache, slot, path, bytes](WorkerCtx&) {
                         try {
                           write_file_atomic(path, slot->buf->host(), bytes);
                           stats_inc([](EngineStats& s) {
                             s.files_written++;
                             s.writeback_flushes++;
                           });
                         } catch (const std::exception& e) {
                           KVO_LOG_ERROR("writeback flush %s failed: %s",
                                         path.c_str(), e.what());
                           stats_inc([](EngineStats& s) { s.errors++; });
                         }
                         cache->release(slot);
                       });
        stats_inc([&](EngineStats& s) {
          s.bytes_stored += bytes;
          s.host_cache_stores++;
        });
        return false;
      }
    } else {
      // KV-ready fence: the gather must observe the serving engine's
      // completed KV writes for these blocks.
      double t0 = now_s();
      KVO_HIP_CHECK(hipStreamWaitEvent(ctx.stream, job->kv_ready, 0));
      uint8_t* kernel_dst = cfg_.copy_path == CopyPath::kStaged
                                ? ctx.device_staging->ptr()
                                : ctx.host_staging->device();
      const int32_t* ids_dev = stage_ids(ctx, ft);
      hipError_t err;
      if (cfg_.serialize == Serialize::kFp8E4M3) {
        if (cfg_.copy_path == CopyPath::kStaged) {
          // split amax+quantize (chip-filling); the partial-max scratch
          // (tiles x slices floats, slices <= 2048/tiles + 1) lives in the
          // tail of the device bounce — packed fp8 is half of raw, so the
          // raw-sized bounce leaves bytes/1 >= tiles*slices*4 of room
          // (worst case 2048+tiles floats ~ 16 KiB)
          float* scratch = reinterpret_cast<float*>(
              ctx.device_staging->ptr() + bytes);
          err = kvc_launch_gather_fp8_split(
              const_cast<const void* const*>(dev_layer_ptrs_[ft.group]),
              dev_layer_strides_[ft.group], nl, g.block_bytes,
              ft.block_ids.data(), nb, ids_dev, kernel_dst, scratch,
              ctx.stream);
        } else {
          // zero-copy writes pinned host directly: atomics over PCIe are
          // pathological, keep the fused single-workgroup variant
          err = kvc_launch_gather_fp8(
              const_cast<const void* const*>(dev_layer_ptrs_[ft.group]),
              dev_layer_strides_[ft.group], nl, g.block_bytes,
              ft.block_ids.data(), nb, ids_dev, kernel_dst, ctx.stream);
        }
      } else {
        err = kvc_launch_gather(
            const_cast<const void* const*>(dev_layer_ptrs_[ft.group]),
            dev_layer_strides_[ft.group], nl, g.block_bytes,
            ft.block_ids.data(), nb, ids_dev, kernel_dst, ctx.stream);
      }
      if (err != hipSuccess) throw HipError(hipGetErrorString(err));
      double t1 = now_s();
      stats_inc([&](EngineStats& s) { s.t_gather_ms += (t1 - t0) * 1e3; });
      if (cfg_.copy_path == CopyPath::kStaged &&
          cfg_.write_policy == WritePolicy::kBack && slot != nullptr) {
        // Write-back: the store completes once the slab is resident in
        // the DRAM tier; the file flush runs as a background task.
        hipEvent_t gather_done;
        KVO_HIP_CHECK(hipEventCreateWithFlags(&gather_done, hipEventDisableTiming));
        KVO_HIP_CHECK(hipEventRecord(gather_done, ctx.stream));
        double c0 = now_s();
        try {
          mover_->d2h(host_buf, ctx.device_staging->ptr(), bytes, gather_done);
        } catch (...) {
          (void)hipEventDestroy(gather_done);
          throw;
        }
        (void)hipEventDestroy(gather_done);
        double c1 = now_s();
        guard.ok = true;  // publish + release the store's reference
        cache_->addref(slot);
        auto* cache = cache_.get();
        std::string path = ft.path;
        pool_->enqueue(Priority::kCont,
                       [this, cache, slot, path, bytes](WorkerCtx&) {
                         double w0 = now_s();
                         try {
                           AtomicFileWriter w(path, cfg_.direct_io);
                           w.write_at(0, slot->buf->host(), bytes);
                           w.commit();
                           stats_inc([&](EngineStats& s) {
                             s.files_written++;
                             s.writeback_flushes++;
                             s.t_write_ms += (now_s() - w0) * 1e3;
                           });
                         } catch (const std::exception& e) {
                           KVO_LOG_ERROR("writeback flush %s failed: %s",
                                         path.c_str(), e.what());
                           stats_inc([](EngineStats& s) { s.errors++; });
                         }
                         cache->release(slot);
                       });
        stats_inc([&](EngineStats& s) {
          s.t_d2h_ms += (c1 - c0) * 1e3;
          s.bytes_stored += bytes;
          s.host_cache_stores++;
        });
        return false;
      }
      if (cfg_.copy_path == CopyPath::kStaged && slot != nullptr &&
          two_stage_stores()) {
        // Two-stage write-through store: this worker completes once the
        // D2H lands in the DRAM slot; the file write runs as its OWN task
        // (completion still waits for the rename — durability unchanged).
        // Per-worker serialization of [D2H -> write] left the D2H lane
        // idle whenever every worker sat in write(): the duplex wire
        // measures ~104 GB/s total (profiles/r02) while the bench ran
        // ~54 — the lane must stay fed from the next task's D2H. The
        // continuation runs in the kCont class: read-preferring workers
        // pick writes up after their loads drain while write-preferring
        // workers keep pumping store D2Hs (thread_pool.h preferences).
        hipEvent_t gather_done;
        KVO_HIP_CHECK(hipEventCreateWithFlags(&gather_done, hipEventDisableTiming));
        KVO_HIP_CHECK(hipEventRecord(gather_done, ctx.stream));
        double c0 = now_s();
        try {
          mover_->d2h(host_buf, ctx.device_staging->ptr(), bytes, gather_done);
        } catch (...) {
          (void)hipEventDestroy(gather_done);
          throw;
        }
        (void)hipEventDestroy(gather_done);
        double c1 = now_s();
        guard.ok = true;  // publish the slot; store's reference released
        cache_->addref(slot);
        auto* cache = cache_.get();
        std::string path = ft.path;
        pool_->enqueue(Priority::kCont,
                       [this, job, cache, slot, path, bytes](WorkerCtx&) {
          bool wok = true;
          if (!job->cancelled.load(std::memory_order_acquire)) {
            double w0 = now_s();
            try {
              AtomicFileWriter w(path, cfg_.direct_io);
              w.write_at(0, slot->buf->host(), bytes);
              w.commit();
              double dt = now_s() - w0;
              update_write_ema(dt);
              stats_inc([&](EngineStats& s) {
                s.files_written++;
                s.t_write_ms += dt * 1e3;
              });
            } catch (const std::exception& e) {
              KVO_LOG_ERROR("store write %s failed: %s", path.c_str(),
                            e.what());
              stats_inc([](EngineStats& s) { s.errors++; });
              wok = false;
            }
          } else {
            stats_inc([](EngineStats& s) { s.tasks_cancelled++; });
          }
          cache->release(slot);
          pending_writes_.fetch_sub(1, std::memory_order_relaxed);
          task_done(job, wok);
        });
        stats_inc([&](EngineStats& s) {
          s.t_d2h_ms += (c1 - c0) * 1e3;
          s.bytes_stored += bytes;
          s.host_cache_stores++;
        });
        return true;
      }
      if (cfg_.copy_path == CopyPath::kStaged) {
        // Chunked pipeline: D2H of chunk i+1 rides the SDMA mover while
        // chunk i is written to the file (zero CU occupancy on the wire;
        // one stream per direction saturates it — see pcie_mover.h).
        hipEvent_t gather_done;
        KVO_HIP_CHECK(hipEventCreateWithFlags(&gather_done, hipEventDisableTiming));
        KVO_HIP_CHECK(hipEventRecord(gather_done, ctx.stream));
        double t_copy = 0, t_io = 0;
        try {
          const size_t chunk = pipeline_chunk(bytes);
          AtomicFileWriter writer(ft.path, cfg_.direct_io);
          std::vector<std::future<void>> futs;
          for (size_t off = 0; off < bytes; off += chunk) {
            size_t n = std::min(chunk, bytes - off);
            futs.push_back(mover_->d2h_async(
                host_buf + off, ctx.device_staging->ptr() + off,
                n, off == 0 ? gather_done : nullptr));
          }
          size_t i = 0;
          for (size_t off = 0; off < bytes; off += chunk, ++i) {
            size_t n = std::min(chunk, bytes - off);
            double w0 = now_s();
            futs[i].get();
            double w1 = now_s();
            writer.write_at(off, host_buf + off, n);
            double w2 = now_s();
            t_copy += w1 - w0;
            t_io += w2 - w1;
          }
          writer.commit();
        } catch (...) {
          (void)hipEventDestroy(gather_done);
          throw;
        }
        (void)hipEventDestroy(gather_done);
        guard.ok = true;
        stats_inc([&](EngineStats& s) {
          s.t_d2h_ms += t_copy * 1e3;
          s.t_write_ms += t_io * 1e3;
          s.files_written++;
          s.bytes_stored += bytes;
          if (slot) s.host_cache_stores++;
        });
        return false;
      }
      KVO_HIP_CHECK(hipStreamSynchronize(ctx.stream));
      double t2 = now_s();
      stats_inc([&](EngineStats& s) { s.t_d2h_ms += (t2 - t1) * 1e3; });
    }
    double tw = now_s();
    write_file_atomic(ft.path, host_buf, bytes);
    double tw2 = now_s();
    guard.ok = true;
    stats_inc([&](EngineStats& s) {
      s.files_written++;
      s.bytes_stored += bytes;
      s.t_write_ms += (tw2 - tw) * 1e3;
      if (slot) s.host_cache_stores++;
    });
    return false;
  }

  void load_one(WorkerCtx& ctx, const FileTransfer& ft) {
    const GroupDesc& g = groups_[ft.group];
    const int nb = static_cast<int>(ft.block_ids.size());
    const int nl = static_cast<int>(g.layer_ptrs.size());
    const size_t bytes = file_bytes(g, nb);
    const uint64_t offset =
        static_cast<uint64_t>(ft.slot_offset) * nl * tile_record_bytes(g);

    // (file coverage is validated on the miss paths — a pinned-DRAM cache
    // hit must serve even when the file was already evicted from disk)
    auto check_file_span = [&]() {
      int64_t fsz = file_size(ft.path);
      if (fsz < 0 || static_cast<uint64_t>(fsz) < offset + bytes)
        throw FileIoError("file " + ft.path + " does not cover requested span");
    };

    if (cfg_.copy_path == CopyPath::kStaged) {
      // Pinned-DRAM tier: a cache hit skips the filesystem read entirely —
      // the H2D streams straight from the resident slab.
      HostPinnedCache::Slot* hit = cache_ ? cache_->lookup(ft.path) : nullptr;
      if (hit && hit->bytes_used < offset + bytes) {
        cache_->release(hit);
        hit = nullptr;
      }
      HostPinnedCache::Slot* fill = nullptr;
      uint8_t* host_buf = ctx.host_staging->host();
      if (hit) {
        host_buf = hit->buf->host() + offset;
      } else if (cache_ && offset == 0) {
        // populate the tier on a miss (full-span loads only)
        fill = cache_->acquire(ft.path, bytes);
        if (fill) host_buf = fill->buf->host();
      }
      struct LoadGuard {
        HostPinnedCache* c;
        HostPinnedCache::Slot* hit;
        HostPinnedCache::Slot* fill;
        bool ok = false;
        ~LoadGuard() {
          if (!c) return;
          if (hit) c->release(hit);
          if (fill) {
            if (ok) {
              c->publish(fill);
              c->release(fill);
            } else {
              c->abandon(fill);
            }
          }
        }
      } lguard{cache_.get(), hit, fill};

      const size_t chunk = pipeline_chunk(bytes);
      double t_read = 0, t_h2d = 0;
      std::vector<std::future<void>> futs;
      if (hit) {
        futs.push_back(mover_->h2d_async(ctx.device_staging->ptr(), host_buf,
                                         bytes));
      } else {
        check_file_span();
        FileReader reader(ft.path, cfg_.direct_io);
        double r0 = now_s();
        for (size_t off = 0; off < bytes; off += chunk) {
          size_t n = std::min(chunk, bytes - off);
          reader.read_at(offset + off, host_buf + off, n);
          double r1 = now_s();
          t_read += r1 - r0;
          futs.push_back(mover_->h2d_async(ctx.device_staging->ptr() + off,
                                           host_buf + off, n));
          r0 = now_s();
        }
      }
      double h0 = now_s();
      for (auto& f : futs) f.get();
      t_h2d = now_s() - h0;
      touch_atime(ft.path);
      const int32_t* ids_dev = stage_ids(ctx, ft);
      hipError_t err =
          cfg_.serialize == Serialize::kFp8E4M3
              ? kvc_launch_scatter_fp8(
                    const_cast<const void* const*>(dev_layer_ptrs_[ft.group]),
                    dev_layer_strides_[ft.group], nl, g.block_bytes,
                    ft.block_ids.data(), nb, ids_dev,
                    ctx.device_staging->ptr(), ctx.stream)
              : kvc_launch_scatter(
                    const_cast<const void* const*>(dev_layer_ptrs_[ft.group]),
                    dev_layer_strides_[ft.group], nl, g.block_bytes,
                    ft.block_ids.data(), nb, ids_dev,
                    ctx.device_staging->ptr(), ctx.stream);
      if (err != hipSuccess) throw HipError(hipGetErrorString(err));
      double s0 = now_s();
      KVO_HIP_CHECK(hipStreamSynchronize(ctx.stream));
      double s1 = now_s();
      lguard.ok = true;
      stats_inc([&](EngineStats& s) {
        s.t_read_ms += t_read * 1e3;
        s.t_h2d_ms += t_h2d * 1e3;
        s.t_scatter_ms += (s1 - s0) * 1e3;
        s.files_read++;
        s.bytes_loaded += bytes;
        if (hit) s.host_cache_hits++;
      });
      return;
    }

    if (cfg_.copy_path == CopyPath::kHostMemcpy && cache_) {
      HostPinnedCache::Slot* hit = cache_->lookup(ft.path);
      if (hit && hit->bytes_used >= offset + bytes) {
        scatter_host(g, ft.block_ids, hit->buf->host() + offset);
        touch_atime(ft.path);
        cache_->release(hit);
        stats_inc([&](EngineStats& s) {
          s.files_read++;
          s.bytes_loaded += bytes;
          s.host_cache_hits++;
        });
        return;
      }
      if (hit) cache_->release(hit);
    }
    check_file_span();
    double t0 = now_s();
    read_file_range(ft.path, offset, ctx.host_staging->host(), bytes);
    touch_atime(ft.path);
    double t1 = now_s();
    stats_inc([&](EngineStats& s) { s.t_read_ms += (t1 - t0) * 1e3; });

    if (cfg_.copy_path == CopyPath::kHostMemcpy) {
      scatter_host(g, ft.block_ids, ctx.host_staging->host());
    } else {
      const uint8_t* kernel_src = ctx.host_staging->device();
      double t2 = now_s();
      stats_inc([&](EngineStats& s) { s.t_h2d_ms += (t2 - t1) * 1e3; });
      const int32_t* ids_dev = stage_ids(ctx, ft);
      hipError_t err =
          cfg_.serialize == Serialize::kFp8E4M3
              ? kvc_launch_scatter_fp8(
                    const_cast<const void* const*>(dev_layer_ptrs_[ft.group]),
                    dev_layer_strides_[ft.group], nl, g.block_bytes,
                    ft.block_ids.data(), nb, ids_dev, kernel_src, ctx.stream)
              : kvc_launch_scatter(
                    const_cast<const void* const*>(dev_layer_ptrs_[ft.group]),
                    dev_layer_strides_[ft.group], nl, g.block_bytes,
                    ft.block_ids.data(), nb, ids_dev, kernel_src, ctx.stream);
      if (err != hipSuccess) throw HipError(hipGetErrorString(err));
      KVO_HIP_CHECK(hipStreamSynchronize(ctx.stream));
      double t3 = now_s();
      stats_inc([&](EngineStats& s) { s.t_scatter_ms += (t3 - t2) * 1e3; });
    }
    stats_inc([&](EngineStats& s) {
      s.files_read++;
      s.bytes_loaded += bytes;
    });
  }

  void gather_host(const GroupDesc& g, const std::vector<int32_t>& ids,
                   uint8_t* dst) const {
    const size_t nl = g.layer_ptrs.size();
    const size_t rec = tile_record_bytes(g);
    for (size_t bi = 0; bi < ids.size(); ++bi) {
      for (size_t l = 0; l < nl; ++l) {
        const uint8_t* src = static_cast<const uint8_t*>(g.layer_ptrs[l]) +
                             static_cast<uint64_t>(ids[bi]) * g.layer_strides[l];
        uint8_t* out = dst + (bi * nl + l) * rec;
        if (cfg_.serialize == Serialize::kFp8E4M3)
          fp8_quantize_tile(src, g.block_bytes, out);
        else
          std::memcpy(out, src, g.block_bytes);
      }
    }
  }

  void scatter_host(const GroupDesc& g, const std::vector<int32_t>& ids,
                    const uint8_t* src) const {
    const size_t nl = g.layer_ptrs.size();
    const size_t rec = tile_record_bytes(g);
    for (size_t bi = 0; bi < ids.size(); ++bi) {
      for (size_t l = 0; l < nl; ++l) {
        uint8_t* dst = static_cast<uint8_t*>(g.layer_ptrs[l]) +
                       static_cast<uint64_t>(ids[bi]) * g.layer_strides[l];
        const uint8_t* in = src + (bi * nl + l) * rec;
        if (cfg_.serialize == Serialize::kFp8E4M3)
          fp8_dequantize_tile(in, g.block_bytes, dst);
        else
          std::memcpy(dst, in, g.block_bytes);
      }
    }
  }

  // ---- software fp8 e4m3fn (host-mode twin of the CDNA4 kernels) -----------

  static float bf16_to_f32(uint16_t u) {
    uint32_t w = static_cast<uint32_t>(u) << 16;
    float f;
    std::memcpy(&f, &w, 4);
    return f;
  }
  static uint16_t f32_to_bf16(float f) {
    uint32_t w;
    std::memcpy(&w, &f, 4);
    uint32_t rounding = 0x7fff + ((w >> 16) & 1);
    return static_cast<uint16_t>((w + rounding) >> 16);
  }
  static uint8_t f32_to_fp8(float x) {
    // OCP e4m3fn: 1s 4e 3m, bias 7, max 448, no inf, single NaN.
    if (x != x) return 0x7f;
    uint8_t sign = x < 0 ? 0x80 : 0;
    float a = std::abs(x);
    if (a >= 448.0f) return sign | 0x7e;  // saturate to max normal
    if (a < 0.0009765625f) {              // subnormal range (< 2^-10 = min subnormal/2... )
      // subnormals: value = m * 2^-9, m in [0,7]
      int m = static_cast<int>(a * 512.0f + 0.5f);
      if (m > 7) m = 7;
      return sign | static_cast<uint8_t>(m);
    }
    int e;
    float frac = std::frexp(a, &e);  // a = frac * 2^e, frac in [0.5, 1)
    // normalized: a = 1.mmm * 2^(e-1); exponent field = (e-1)+7
    int exp_field = e - 1 + 7;
    float mant = frac * 2.0f - 1.0f;  // [0,1)
    int m = static_cast<int>(mant * 8.0f + 0.5f);
    if (m == 8) {
      m = 0;
      exp_field += 1;
    }
    if (exp_field >= 16) return sign | 0x7e;
    if (exp_field <= 0) {
      // underflow into subnormal
      int ms = static_cast<int>(a * 512.0f + 0.5f);
      if (ms > 7) ms = 7;
      return sign | static_cast<uint8_t>(ms);
    }
    return sign | static_cast<uint8_t>((exp_field << 3) | m);
  }
  static float fp8_to_f32(uint8_t b) {
    uint8_t sign = b & 0x80;
    int exp_field = (b >> 3) & 0xf;
    int m = b & 0x7;
    if (exp_field == 0xf && m == 0x7) return NAN;
    float v;
    if (exp_field == 0)
      v = m * 0.001953125f;  // m * 2^-9
    else
      v = std::ldexp(1.0f + m / 8.0f, exp_field - 7);
    return sign ? -v : v;
  }

  public:
  // Shared software fp8 path (BlockCopier's host mode reuses it).
  static void host_fp8_copy(const GroupDesc& g, const std::vector<int32_t>& ids,
                            uint8_t* packed, bool quantize) {
    const size_t nl = g.layer_ptrs.size();
    const size_t rec = g.block_bytes / 2 + 4;
    for (size_t bi = 0; bi < ids.size(); ++bi) {
      for (size_t l = 0; l < nl; ++l) {
        uint8_t* page = static_cast<uint8_t*>(g.layer_ptrs[l]) +
                        static_cast<uint64_t>(ids[bi]) * g.layer_strides[l];
        uint8_t* slab = packed + (bi * nl + l) * rec;
        if (quantize)
          fp8_quantize_tile(page, g.block_bytes, slab);
        else
          fp8_dequantize_tile(slab, g.block_bytes, page);
      }
    }
  }

  private:
  static void fp8_quantize_tile(const uint8_t* src, size_t block_bytes,
                         uint8_t* out) {
    const size_t n = block_bytes / 2;
    const uint16_t* in = reinterpret_cast<const uint16_t*>(src);
    float amax = 0.0f;
    for (size_t i = 0; i < n; ++i)
      amax = std::max(amax, std::abs(bf16_to_f32(in[i])));
    if (amax <= 0.0f) amax = 1.0f;
    float scale = amax / 448.0f;
    float inv = 448.0f / amax;
    for (size_t i = 0; i < n; ++i)
      out[i] = f32_to_fp8(bf16_to_f32(in[i]) * inv);
    std::memcpy(out + n, &scale, 4);
  }

  static void fp8_dequantize_tile(const uint8_t* in, size_t block_bytes,
                           uint8_t* dst) {
    const size_t n = block_bytes / 2;
    float scale;
    std::memcpy(&scale, in + n, 4);
    uint16_t* out = reinterpret_cast<uint16_t*>(dst);
    for (size_t i = 0; i < n; ++i)
      out[i] = f32_to_bf16(fp8_to_f32(in[i]) * scale);
  }

  EngineConfig cfg_;
  std::vector<GroupDesc> groups_;
  std::vector<void**> dev_layer_ptrs_;
  std::vector<uint64_t*> dev_layer_strides_;
  std::unique_ptr<IoThreadPool> pool_;
  std::unique_ptr<PcieMover> mover_;
  std::unique_ptr<HostPinnedCache> cache_;

  std::mutex jobs_mu_;  // guards jobs_/finished_/next_job_id_ only
  std::condition_variable done_cv_;
  std::unordered_map<int64_t, std::shared_ptr<Job>> jobs_;
  std::vector<FinishedJob> finished_;
  int64_t next_job_id_ = 1;
  std::atomic<size_t> pending_writes_{0};
  std::array<StatsShard, kStatsShards> stats_shards_;
  std::atomic<uint64_t> write_ema_bits_{0};  // double bit-pattern (0.0 = unset)
};

}  // namespace kvo
