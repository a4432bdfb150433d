// KVEvents ingestion: HMA group catalog + sharded worker pool.
//
// Capability parity with the reference pkg/kvevents.Pool (pool.go) and
// pkg/kvcache/kvblock/hma.go, re-designed as native threads: N workers each
// own an ordered MPSC queue; messages shard by FNV-1a(pod) so one pod's
// events are processed in order; decoding and index mutation never touch
// the Python interpreter (no GIL on the ingest path).
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <mutex>
#include <optional>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include <algorithm>

#include "../common/fnv.h"
#include "../core/index.h"
#include "../core/token_processor.h"
#include "events.h"

namespace kvc {

struct GroupMetadata {
  std::string kind;
  int block_size = 0;
  std::optional<int32_t> sliding_window;
};

// Thread-safe per-pod GroupID -> GroupMetadata catalog learned from
// BlockStored HMA fields.
class GroupCatalog {
 public:
  void learn(const std::string& pod, int32_t group, GroupMetadata md) {
    std::lock_guard<std::mutex> g(mu_);
    catalog_[pod][group] = std::move(md);
    version_.fetch_add(1, std::memory_order_relaxed);
  }

  // Monotonic change counter + pod count: lets scorers cache window hints
  // and skip the per-pod queries entirely while no HMA fields were seen.
  uint64_t version() const {
    return version_.load(std::memory_order_relaxed);
  }
  size_t pod_count() const {
    std::lock_guard<std::mutex> g(mu_);
    return catalog_.size();
  }

  // A cleared (restarted) pod may come back with a different model
  // config: forget its learned structure until new events re-teach it.
  void forget(const std::string& pod) {
    std::lock_guard<std::mutex> g(mu_);
    if (catalog_.erase(pod) > 0)
      version_.fetch_add(1, std::memory_order_relaxed);
  }
  std::optional<GroupMetadata> get(const std::string& pod, int32_t group) const {
    std::lock_guard<std::mutex> g(mu_);
    auto it = catalog_.find(pod);
    if (it == catalog_.end()) return std::nullopt;
    auto jt = it->second.find(group);
    if (jt == it->second.end()) return std::nullopt;
    return jt->second;
  }

  // Sliding-window hint: the pod's window size in TOKENS if every learned
  // KV-cache group on the pod uses sliding-window attention, else 0.
  int32_t sliding_window_tokens(const std::string& pod) const {
    std::lock_guard<std::mutex> g(mu_);
    auto it = catalog_.find(pod);
    if (it == catalog_.end() || it->second.empty()) return 0;
    int32_t w = 0;
    for (const auto& [gid, md] : it->second) {
      if (!md.sliding_window.has_value() || *md.sliding_window <= 0) return 0;
      w = std::max(w, *md.sliding_window);
    }
    return w;
  }

  // Full group structure for hybrid-aware scoring: group id -> sliding
  // window in TOKENS (0 = full attention). Empty map = unknown pod.
  std::unordered_map<int32_t, int32_t> group_windows(
      const std::string& pod) const {
    std::lock_guard<std::mutex> g(mu_);
    std::unordered_map<int32_t, int32_t> out;
    auto it = catalog_.find(pod);
    if (it == catalog_.end()) return out;
    for (const auto& [gid, md] : it->second)
      out[gid] = md.sliding_window.value_or(0) > 0 ? *md.sliding_window : 0;
    return out;
  }

  // Whole-catalog snapshot (pod -> group -> window tokens): unfiltered
  // scoring needs hints for every cataloged pod.
  std::unordered_map<std::string, std::unordered_map<int32_t, int32_t>>
  snapshot() const {
    std::lock_guard<std::mutex> g(mu_);
    std::unordered_map<std::string, std::unordered_map<int32_t, int32_t>> out;
    for (const auto& [pod, groups] : catalog_) {
      auto& m = out[pod];
      for (const auto& [gid, md] : groups)
        m[gid] = md.sliding_window.value_or(0) > 0 ? *md.sliding_window : 0;
    }
    return out;
  }

 private:
  mutable std::mutex mu_;
  std::atomic<uint64_t> version_{0};
  std::unordered_map<std::string, std::unordered_map<int32_t, GroupMetadata>> catalog_;
};

struct RawMessage {
  std::string topic;
  uint64_t seq = 0;
  std::string payload;
};

struct PoolStats {
  uint64_t enqueued = 0;
  uint64_t processed = 0;
  uint64_t parse_failures = 0;
  uint64_t handler_failures = 0;
  uint64_t dropped_parent_misses = 0;
  uint64_t dropped_backpressure = 0;
};

class EventPool {
 public:
  // dp_rank_routing: when a batch carries DataParallelRank, treat each DP
  // rank as its own scheduling target ("<pod>-dp<r>") so the scorer can
  // route to the rank that actually holds the blocks. The reference
  // decodes the field but leaves routing as future work
  // (docs/architecture.md:292, vllm_adapter.go:91-96).
  // max_queue_depth bounds each shard's backlog (0 = unbounded): when a
  // flood outruns the workers, the OLDEST queued message of that shard is
  // dropped (the index converges from later full-prefix events; memory
  // stays bounded — the workqueue-backpressure role of the reference's
  // rate-limited queues, pool.go:37-86).
  EventPool(std::shared_ptr<TokenProcessor> tp, std::shared_ptr<IndexBackend> index,
            size_t concurrency = 4, bool dp_rank_routing = false,
            size_t max_queue_depth = 0)
      : tp_(std::move(tp)), index_(std::move(index)),
        queues_(std::max<size_t>(1, concurrency)),
        dp_rank_routing_(dp_rank_routing), max_queue_depth_(max_queue_depth) {}

  ~EventPool() { shutdown(); }

  GroupCatalog& group_catalog() { return catalog_; }

  void start() {
    std::lock_guard<std::mutex> g(lifecycle_mu_);
    if (running_) return;
    running_ = true;
    for (size_t i = 0; i < queues_.size(); ++i)
      workers_.emplace_back([this, i] { worker_loop(i); });
  }

  void shutdown() {
    {
      std::lock_guard<std::mutex> g(lifecycle_mu_);
      if (!running_) return;
      running_ = false;
    }
    for (auto& q : queues_) q.cv.notify_all();
    for (auto& w : workers_) w.join();
    workers_.clear();
  }

  // Enqueue a raw transport message; same pod -> same worker -> ordered.
  void add_task(RawMessage msg) {
    std::string pod, model;
    parse_topic(msg.topic, pod, model);
    size_t shard = fnv32a(pod) % queues_.size();
    auto& q = queues_[shard];
    {
      std::lock_guard<std::mutex> g(q.mu);
      if (max_queue_depth_ > 0 && q.items.size() >= max_queue_depth_) {
        q.items.pop_front();
        dropped_backpressure_.fetch_add(1, std::memory_order_relaxed);
      }
      q.items.push_back(std::move(msg));
    }
    enqueued_.fetch_add(1, std::memory_order_relaxed);
    q.cv.notify_one();
  }

  // Synchronous processing (offline/batch path and tests).
  void process(const RawMessage& msg) { process_message(msg); }

  // Block until every queued message has been processed (test helper).
  void drain() {
    for (auto& q : queues_) {
      std::unique_lock<std::mutex> g(q.mu);
      q.idle_cv.wait(g, [&q] { return q.items.empty() && !q.busy; });
    }
  }

  PoolStats stats() const {
    PoolStats s;
    s.enqueued = enqueued_.load(std::memory_order_relaxed);
    s.processed = processed_.load(std::memory_order_relaxed);
    s.parse_failures = parse_failures_.load(std::memory_order_relaxed);
    s.handler_failures = handler_failures_.load(std::memory_order_relaxed);
    s.dropped_parent_misses = dropped_parent_misses_.load(std::memory_order_relaxed);
    s.dropped_backpressure = dropped_backpressure_.load(std::memory_order_relaxed);
    return s;
  }

 private:
  struct Queue {
    std::mutex mu;
    std::condition_variable cv;
    std::condition_variable idle_cv;
    std::deque<RawMessage> items;
    bool busy = false;

    Queue() = default;
    Queue(const Queue&) {}
  };

  void worker_loop(size_t i) {
    auto& q = queues_[i];
    for (;;) {
      RawMessage msg;
      {
        std::unique_lock<std::mutex> g(q.mu);
        q.busy = false;
        q.idle_cv.notify_all();
        q.cv.wait(g, [&] { return !q.items.empty() || !running_; });
        if (q.items.empty()) return;  // shutting down and drained
        msg = std::move(q.items.front());
        q.items.pop_front();
        q.busy = true;
      }
      process_message(msg);
    }
  }

  void process_message(const RawMessage& msg) {
    std::string pod, model;
    parse_topic(msg.topic, pod, model);
    EventBatch batch;
    try {
      batch = parse_batch(reinterpret_cast<const uint8_t*>(msg.payload.data()),
                          msg.payload.size());
    } catch (const MsgpackError&) {
      parse_failures_.fetch_add(1, std::memory_order_relaxed);
      return;
    }
    if (dp_rank_routing_ && batch.dp_rank.has_value())
      pod += "-dp" + std::to_string(*batch.dp_rank);
    for (const auto& ev : batch.events) {
      // Handlers can throw past the parser: a shared backend (RespError on
      // a transient Redis/Valkey outage, malformed numeric fields) must not
      // escape the worker thread — that would std::terminate the whole
      // indexer process on a routine network blip. Count and drop the event;
      // the index converges from later events / the other replicas.
      try {
        switch (ev.type) {
          case EventType::kBlockStored:
            handle_stored(pod, model, ev.stored);
            break;
          case EventType::kBlockRemoved:
            handle_removed(pod, ev.removed);
            break;
          case EventType::kAllBlocksCleared: {
            // Intern (not find): with a shared backend (Redis/Valkey) this
            // replica may be asked to clear a pod whose stores another
            // process wrote.
            index_->clear(index_->strings().intern(pod));
            catalog_.forget(pod);
            break;
          }
        }
      } catch (const std::exception&) {
        handler_failures_.fetch_add(1, std::memory_order_relaxed);
      }
    }
    processed_.fetch_add(1, std::memory_order_relaxed);
  }

  static std::string lower(std::string s) {
    for (auto& c : s) c = static_cast<char>(std::tolower(static_cast<unsigned char>(c)));
    return s;
  }

  PodEntry make_entry(const std::string& pod, const std::string& medium,
                      const std::optional<int32_t>& group) {
    PodEntry e;
    e.pod = index_->strings().intern(pod);
    e.tier = index_->strings().intern(medium.empty() ? std::string("gpu") : lower(medium));
    if (group.has_value()) {
      e.flags |= 2;
      e.group = *group;
    }
    return e;
  }

  // Convert per-engine-block extras to per-canonical-block granularity
  // (1:many replicate, many:1 merge) so the hash chain sees one entry per
  // canonical chunk.
  static std::vector<BlockExtra> realign_extras(
      const std::vector<BlockExtra>& engine, size_t canonical) {
    const size_t e = engine.size();
    if (canonical == 0) return {};
    if (e == 0 || e == canonical) return engine;
    std::vector<BlockExtra> out(canonical);
    if (e < canonical) {
      for (size_t i = 0; i < canonical; ++i) out[i] = engine[i * e / canonical];
    } else {
      for (size_t i = 0; i < e; ++i) {
        if (!engine[i].has_value()) continue;
        size_t ci = i * canonical / e;
        if (!out[ci].has_value()) out[ci].emplace();
        out[ci]->insert(out[ci]->end(), engine[i]->begin(), engine[i]->end());
      }
    }
    return out;
  }

  void handle_stored(const std::string& pod, const std::string& model,
                     const BlockStoredEvent& ev) {
    std::string eff_model = model;
    if (ev.lora_name.has_value() && !ev.lora_name->empty()) eff_model = *ev.lora_name;

    if (ev.group_idx.has_value()) {
      GroupMetadata md;
      md.kind = ev.spec_kind;
      md.block_size = ev.block_size;
      md.sliding_window = ev.sliding_window;
      catalog_.learn(pod, *ev.group_idx, std::move(md));
    }
    std::vector<PodEntry> entries{make_entry(pod, ev.medium, ev.group_idx)};

    uint64_t parent_request_key = 0;
    if (ev.parent_hash != 0) {
      if (!index_->get_request_key(ev.parent_hash, &parent_request_key)) {
        // Message loss tolerance: an unknown parent chain is dropped
        // gracefully (the index converges from later full-prefix events).
        dropped_parent_misses_.fetch_add(1, std::memory_order_relaxed);
        return;
      }
    }

    std::vector<BlockExtra> extras;
    if (ev.has_extra_keys) {
      size_t canonical = ev.tokens.size() / static_cast<size_t>(tp_->block_size());
      extras = realign_extras(ev.extra_keys, canonical);
    }

    auto request_keys = tp_->tokens_to_block_keys(
        parent_request_key, ev.tokens.data(), ev.tokens.size(), eff_model,
        extras.empty() ? nullptr : &extras);

    if (request_keys.empty()) {
      // Token-less events (e.g. CPU-offload tier updates) resolve existing
      // request keys through the engine bridge.
      if (!ev.tokens.empty() || ev.block_hashes.empty()) return;
      std::vector<uint64_t> resolved;
      std::unordered_set<uint64_t> seen;
      for (uint64_t ek : ev.block_hashes) {
        uint64_t rk;
        if (index_->get_request_key(ek, &rk) && seen.insert(rk).second)
          resolved.push_back(rk);
      }
      if (!resolved.empty()) index_->add({}, resolved, entries);
      return;
    }

    index_->add(ev.block_hashes, request_keys, entries);
  }

  void handle_removed(const std::string& pod, const BlockRemovedEvent& ev) {
    std::vector<PodEntry> entries{make_entry(pod, ev.medium, ev.group_idx)};
    for (uint64_t h : ev.block_hashes)
      index_->evict(h, KeyType::kEngine, entries);
  }

  std::shared_ptr<TokenProcessor> tp_;
  std::shared_ptr<IndexBackend> index_;
  GroupCatalog catalog_;
  std::vector<Queue> queues_;
  std::vector<std::thread> workers_;
  std::mutex lifecycle_mu_;
  bool dp_rank_routing_ = false;
  size_t max_queue_depth_ = 0;
  std::atomic<bool> running_{false};
  std::atomic<uint64_t> enqueued_{0}, processed_{0}, parse_failures_{0},
      handler_failures_{0}, dropped_parent_misses_{0}, dropped_backpressure_{0};
};

}  // namespace kvc
