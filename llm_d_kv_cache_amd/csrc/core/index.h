// In-memory KV-block index: requestKey -> pod entries, plus the
// engineKey -> requestKey bridge used to stitch event-side hash chains.
//
// Capability parity with the reference Index interface and InMemoryIndex
// (pkg/kvcache/kvblock/index.go:120-193, in_memory.go), re-designed for
// multi-core ingest: the store is sharded by key hash with one mutex, one
// open hash map and one intrusive LRU list per shard, and pod entries are
// 12-byte interned records (string_table.h). There is no global lock on the
// Lookup/Add path; the reference's TOCTOU window between the emptiness check
// and map removal is closed structurally because a shard's mutex covers the
// whole check-and-act.
#pragma once

#include <algorithm>
#include <atomic>
#include <cstdio>
#include <cstring>
#include <cstdint>
#include <list>
#include <mutex>
#include <stdexcept>
#include <unordered_map>
#include <unordered_set>
#include <vector>

#include "string_table.h"

namespace kvc {

struct PodEntry {
  uint32_t pod = 0;
  uint32_t tier = 0;
  uint8_t flags = 0;  // bit0: speculative, bit1: has_group
  int32_t group = 0;

  bool speculative() const { return flags & 1; }
  bool has_group() const { return flags & 2; }

  // Eviction match: BlockRemoved events often omit group (and never carry
  // the speculative bit), so eviction matches on pod + tier, constraining
  // group only when the probe names one. Strict operator== stays for add()
  // dedupe.
  static bool evict_match(const PodEntry& stored, const PodEntry& probe) {
    if (stored.pod != probe.pod || stored.tier != probe.tier) return false;
    if (probe.has_group())
      return stored.has_group() && stored.group == probe.group;
    return true;
  }
  bool operator==(const PodEntry& o) const {
    return pod == o.pod && tier == o.tier && flags == o.flags && group == o.group;
  }
};

enum class KeyType { kEngine = 0, kRequest = 1 };

struct IndexStats {
  uint64_t admissions = 0;
  uint64_t evictions = 0;
  uint64_t rejections = 0;  // admission-control rejects (byte-budget mode)
  uint64_t lookups = 0;
  uint64_t hits = 0;
  uint64_t keys = 0;
};

struct InMemoryIndexConfig {
  size_t size = 100000000;   // max request keys (reference default 1e8)
  size_t pods_per_key = 10;  // max pod entries per key
  size_t shards = 64;        // power of two
  // Byte budget (0 = unbounded): when set, the index evicts LRU keys until
  // the approximate resident size fits (cost-aware mode, capability parity
  // with the reference CostAwareMemoryIndex / ristretto backend —
  // implemented as strict sharded LRU-by-bytes instead of probabilistic
  // admission).
  size_t max_bytes = 0;
};

// Pluggable index backend interface (capability parity with the reference
// Index interface, pkg/kvcache/kvblock/index.go:120-155). Implementations:
// InMemoryIndex (below, with optional byte budget) and RedisIndex
// (redis_index.h, network-backed for Valkey/Redis).
class IndexBackend {
 public:
  virtual ~IndexBackend() = default;
  virtual std::vector<std::pair<uint64_t, std::vector<PodEntry>>> lookup(
      const std::vector<uint64_t>& request_keys,
      const std::unordered_set<uint32_t>& pod_filter) = 0;
  virtual void add(const std::vector<uint64_t>& engine_keys,
                   const std::vector<uint64_t>& request_keys,
                   const std::vector<PodEntry>& entries) = 0;
  virtual void evict(uint64_t key, KeyType type,
                     const std::vector<PodEntry>& entries) = 0;
  virtual bool get_request_key(uint64_t engine_key, uint64_t* out) = 0;
  virtual void clear(uint32_t pod_id) = 0;
  virtual IndexStats stats() const = 0;
  StringTable& strings() { return strings_; }

 protected:
  StringTable strings_;
};

class InMemoryIndex : public IndexBackend {
 public:
  explicit InMemoryIndex(const InMemoryIndexConfig& cfg = {}) : cfg_(cfg) {
    if (cfg_.shards == 0 || (cfg_.shards & (cfg_.shards - 1)) != 0)
      throw std::invalid_argument("shards must be a power of two");
    shards_ = std::vector<Shard>(cfg_.shards);
    eng_shards_ = std::vector<EngShard>(cfg_.shards);
    shard_cap_ = std::max<size_t>(1, cfg_.size / cfg_.shards);
  }

  // Lookup keys in order. Filter by pod ids when filter is non-empty.
  // A key that is present but has no (filtered) pods ends the scan early
  // (prefix chain is broken); an absent key is simply skipped.
  std::vector<std::pair<uint64_t, std::vector<PodEntry>>> lookup(
      const std::vector<uint64_t>& request_keys,
      const std::unordered_set<uint32_t>& pod_filter) override {
    if (request_keys.empty())
      throw std::invalid_argument("no request keys provided for lookup");
    lookups_.fetch_add(1, std::memory_order_relaxed);

    std::vector<std::pair<uint64_t, std::vector<PodEntry>>> out;
    out.reserve(request_keys.size());
    bool any_hit = false;
    for (uint64_t key : request_keys) {
      Shard& sh = shard(key);
      std::lock_guard<std::mutex> g(sh.mu);
      auto it = sh.map.find(key);
      if (it == sh.map.end()) continue;
      if (it->second.pods.empty()) break;  // chain break: present but empty
      sh.touch(it);
      if (cfg_.max_bytes > 0) sh.freq.bump(key);  // lookups keep keys warm
      std::vector<PodEntry> pods;
      if (pod_filter.empty()) {
        pods = it->second.pods;
      } else {
        for (const auto& e : it->second.pods)
          if (pod_filter.count(e.pod)) pods.push_back(e);
      }
      if (!pods.empty()) {
        any_hit = true;
        out.emplace_back(key, std::move(pods));
      }
    }
    if (any_hit) hits_.fetch_add(1, std::memory_order_relaxed);
    return out;
  }

  // Add entries under every request key; when engine_keys is non-empty also
  // record the engine->request bridge, inferring 1:1 / many:1 / 1:many from
  // the length ratio (both derive from one token count, so they divide).
  void add(const std::vector<uint64_t>& engine_keys,
           const std::vector<uint64_t>& request_keys,
           const std::vector<PodEntry>& entries) override {
    if (request_keys.empty() || entries.empty())
      throw std::invalid_argument("no keys or entries provided for add");

    if (!engine_keys.empty()) {
      const size_t ne = engine_keys.size(), nr = request_keys.size();
      const size_t n = std::max(ne, nr);
      uint64_t cur_ek = 0;
      std::vector<uint64_t> rks;
      bool have = false;
      for (size_t i = 0; i < n; ++i) {
        uint64_t ek = engine_keys[i * ne / n];
        uint64_t rk = request_keys[i * nr / n];
        if (!have || ek != cur_ek) {
          if (have) put_engine_mapping(cur_ek, rks);
          cur_ek = ek;
          rks.clear();
          have = true;
        }
        rks.push_back(rk);
      }
      if (have) put_engine_mapping(cur_ek, rks);
    }

    for (uint64_t key : request_keys) {
      Shard& sh = shard(key);
      std::lock_guard<std::mutex> g(sh.mu);
      auto it = sh.map.find(key);
      if (it == sh.map.end()) {
        if (cfg_.max_bytes > 0) {
          sh.freq.bump(key);
          // Admission control: a new key that would force eviction must be
          // at least as warm as the LRU victim, or it is rejected — a scan
          // flood of one-shot keys cannot wash out the looked-up set.
          size_t shard_budget = cfg_.max_bytes / cfg_.shards;
          if (sh.bytes + kKeyOverheadBytes + entries.size() * sizeof(PodEntry) >
                  shard_budget &&
              !sh.lru.empty() &&
              sh.freq.estimate(key) < sh.freq.estimate(sh.lru.back())) {
            rejections_.fetch_add(1, std::memory_order_relaxed);
            continue;
          }
        }
        it = sh.map.emplace(key, KeyEntry{}).first;
        sh.lru.push_front(key);
        it->second.lru_it = sh.lru.begin();
        sh.bytes += kKeyOverheadBytes;
        if (sh.map.size() > shard_cap_) evict_lru_locked(sh);
      } else {
        if (cfg_.max_bytes > 0) sh.freq.bump(key);
        sh.touch(it);
      }
      auto& pods = it->second.pods;
      size_t before = pods.size();
      for (const auto& e : entries) {
        // Per-key pod LRU: move-to-front on re-add, bounded capacity.
        for (size_t i = 0; i < pods.size(); ++i) {
          if (pods[i] == e) {
            pods.erase(pods.begin() + i);
            break;
          }
        }
        pods.insert(pods.begin(), e);
        if (pods.size() > cfg_.pods_per_key) pods.pop_back();
      }
      sh.bytes += (pods.size() - before) * sizeof(PodEntry);
      admissions_.fetch_add(entries.size(), std::memory_order_relaxed);
      if (cfg_.max_bytes > 0) {
        // byte-budget mode: evict LRU keys until this shard fits its slice
        size_t shard_budget = cfg_.max_bytes / cfg_.shards;
        while (sh.bytes > shard_budget && sh.map.size() > 1)
          evict_lru_locked(sh);
      }
    }
  }

  // Evict the given pod entries from the key. Engine keys resolve through
  // the bridge (possibly to several request keys); request keys apply
  // directly. When every resolved request key is gone/empty the bridge
  // mapping itself is dropped.
  void evict(uint64_t key, KeyType type,
             const std::vector<PodEntry>& entries) override {
    if (entries.empty())
      throw std::invalid_argument("no entries provided for evict");
    if (type == KeyType::kRequest) {
      evict_from_request_key(key, entries);
      return;
    }
    std::vector<uint64_t> rks;
    {
      EngShard& es = eng_shard(key);
      std::lock_guard<std::mutex> g(es.mu);
      auto it = es.map.find(key);
      if (it == es.map.end()) return;
      rks = it->second.rks;
    }
    for (uint64_t rk : rks) evict_from_request_key(rk, entries);
    bool all_empty = true;
    for (uint64_t rk : rks) {
      Shard& sh = shard(rk);
      std::lock_guard<std::mutex> g(sh.mu);
      auto it = sh.map.find(rk);
      if (it != sh.map.end() && !it->second.pods.empty()) {
        all_empty = false;
        break;
      }
    }
    if (all_empty) {
      EngShard& es = eng_shard(key);
      std::lock_guard<std::mutex> g(es.mu);
      auto it = es.map.find(key);
      if (it != es.map.end()) {
        es.lru.erase(it->second.lru_it);
        es.map.erase(it);
      }
    }
  }

  // Last request key of the engine key's chain segment (the one whose chunk
  // ends where the engine block ends) — what parent-hash resolution needs.
  // Returns false when the mapping is missing (e.g. already evicted).
  bool get_request_key(uint64_t engine_key, uint64_t* out) override {
    EngShard& es = eng_shard(engine_key);
    std::lock_guard<std::mutex> g(es.mu);
    auto it = es.map.find(engine_key);
    if (it == es.map.end() || it->second.rks.empty()) return false;
    es.touch(it);
    *out = it->second.rks.back();
    return true;
  }

  // Drop every entry of the pod, across all tiers. O(N) scan, off the hot
  // path (backs AllBlocksCleared). The engine bridge is intentionally left
  // alone: stale mappings resolve to emptied keys that break the prefix
  // chain correctly and the LRU self-heals.
  void clear(uint32_t pod_id) override {
    for (auto& sh : shards_) {
      std::lock_guard<std::mutex> g(sh.mu);
      for (auto it = sh.map.begin(); it != sh.map.end();) {
        auto& pods = it->second.pods;
        size_t before = pods.size();
        pods.erase(std::remove_if(pods.begin(), pods.end(),
                                  [pod_id](const PodEntry& e) { return e.pod == pod_id; }),
                   pods.end());
        evictions_.fetch_add(before - pods.size(), std::memory_order_relaxed);
        sh.bytes -= (before - pods.size()) * sizeof(PodEntry);
        if (pods.empty()) {
          sh.lru.erase(it->second.lru_it);
          sh.bytes -= kKeyOverheadBytes;
          it = sh.map.erase(it);
        } else {
          ++it;
        }
      }
    }
  }

  IndexStats stats() const override {
    IndexStats s;
    s.admissions = admissions_.load(std::memory_order_relaxed);
    s.evictions = evictions_.load(std::memory_order_relaxed);
    s.rejections = rejections_.load(std::memory_order_relaxed);
    s.lookups = lookups_.load(std::memory_order_relaxed);
    s.hits = hits_.load(std::memory_order_relaxed);
    for (const auto& sh : shards_) {
      std::lock_guard<std::mutex> g(const_cast<std::mutex&>(sh.mu));
      s.keys += sh.map.size();
    }
    return s;
  }

  // ---- snapshot / restore ---------------------------------------------------
  // Warm-restart aid the reference lacks (its index is ephemeral-only and
  // re-converges from the event stream — which still works here; a
  // snapshot just skips the cold-start window). Point-in-time per shard:
  // concurrent writers during save() land in either the file or the live
  // index, both fine for a cache. load() MERGES into the current index.

  void save(const std::string& path) {
    FILE* f = std::fopen((path + ".tmp").c_str(), "wb");
    if (f == nullptr) throw std::runtime_error("snapshot open failed: " + path);
    auto w64 = [&](uint64_t v) { std::fwrite(&v, 8, 1, f); };
    auto w32 = [&](uint32_t v) { std::fwrite(&v, 4, 1, f); };
    std::fwrite("KVIXSNP1", 8, 1, f);
    // string table (ids are dense 0..n-1)
    std::vector<std::string> strs;
    const uint32_t n_ids = static_cast<uint32_t>(strings_.size());
    strs.reserve(n_ids);
    for (uint32_t i = 0; i < n_ids; ++i) strs.push_back(strings_.get(i));
    w32(static_cast<uint32_t>(strs.size()));
    for (const auto& v : strs) {
      w32(static_cast<uint32_t>(v.size()));
      std::fwrite(v.data(), 1, v.size(), f);
    }
    uint64_t n_keys = 0;
    long n_keys_pos = std::ftell(f);
    w64(0);  // patched below
    for (auto& sh : shards_) {
      std::lock_guard<std::mutex> g(sh.mu);
      // oldest -> newest so a load that push_fronts restores LRU order
      for (auto it = sh.lru.rbegin(); it != sh.lru.rend(); ++it) {
        auto ke = sh.map.find(*it);
        if (ke == sh.map.end()) continue;
        w64(*it);
        w32(static_cast<uint32_t>(ke->second.pods.size()));
        for (const auto& e : ke->second.pods) {
          w32(e.pod);
          w32(e.tier);
          uint8_t fl = e.flags;
          std::fwrite(&fl, 1, 1, f);
          int32_t gr = e.group;
          std::fwrite(&gr, 4, 1, f);
        }
        ++n_keys;
      }
    }
    uint64_t n_eng = 0;
    long n_eng_pos = std::ftell(f);
    w64(0);
    for (auto& es : eng_shards_) {
      std::lock_guard<std::mutex> g(es.mu);
      for (auto it = es.lru.rbegin(); it != es.lru.rend(); ++it) {
        auto ee = es.map.find(*it);
        if (ee == es.map.end()) continue;
        w64(*it);
        w32(static_cast<uint32_t>(ee->second.rks.size()));
        for (uint64_t rk : ee->second.rks) w64(rk);
        ++n_eng;
      }
    }
    std::fseek(f, n_keys_pos, SEEK_SET);
    w64(n_keys);
    std::fseek(f, n_eng_pos, SEEK_SET);
    w64(n_eng);
    std::fclose(f);
    if (std::rename((path + ".tmp").c_str(), path.c_str()) != 0)
      throw std::runtime_error("snapshot rename failed: " + path);
  }

  void load(const std::string& path) {
    FILE* f = std::fopen(path.c_str(), "rb");
    if (f == nullptr) throw std::runtime_error("snapshot open failed: " + path);
    struct Closer { FILE* f; ~Closer() { std::fclose(f); } } closer{f};
    auto fail = [&]() -> std::runtime_error {
      return std::runtime_error("corrupt snapshot: " + path);
    };
    auto r64 = [&]() {
      uint64_t v;
      if (std::fread(&v, 8, 1, f) != 1) throw fail();
      return v;
    };
    auto r32 = [&]() {
      uint32_t v;
      if (std::fread(&v, 4, 1, f) != 1) throw fail();
      return v;
    };
    char magic[8];
    if (std::fread(magic, 8, 1, f) != 1 || std::memcmp(magic, "KVIXSNP1", 8))
      throw fail();
    uint32_t n_strs = r32();
    if (n_strs > (1u << 24)) throw fail();
    std::vector<uint32_t> remap(n_strs);
    for (uint32_t i = 0; i < n_strs; ++i) {
      uint32_t len = r32();
      if (len > (1u << 20)) throw fail();
      std::string v(len, '\0');
      if (len && std::fread(v.data(), 1, len, f) != len) throw fail();
      remap[i] = strings_.intern(v);
    }
    uint64_t n_keys = r64();
    for (uint64_t i = 0; i < n_keys; ++i) {
      uint64_t key = r64();
      uint32_t n = r32();
      if (n > 4096) throw fail();
      std::vector<PodEntry> entries(n);
      for (uint32_t j = 0; j < n; ++j) {
        uint32_t pod = r32(), tier = r32();
        if (pod >= n_strs || tier >= n_strs) throw fail();
        entries[j].pod = remap[pod];
        entries[j].tier = remap[tier];
        uint8_t fl;
        int32_t gr;
        if (std::fread(&fl, 1, 1, f) != 1 || std::fread(&gr, 4, 1, f) != 1)
          throw fail();
        entries[j].flags = fl;
        entries[j].group = gr;
      }
      // add() dedupes + maintains LRU/caps/byte budget
      std::reverse(entries.begin(), entries.end());  // preserve front order
      add({}, {key}, entries);
    }
    uint64_t n_eng = r64();
    for (uint64_t i = 0; i < n_eng; ++i) {
      uint64_t ek = r64();
      uint32_t n = r32();
      if (n > (1u << 20)) throw fail();
      std::vector<uint64_t> rks(n);
      for (uint32_t j = 0; j < n; ++j) rks[j] = r64();
      put_engine_mapping(ek, rks);
    }
  }

 private:

  struct KeyEntry {
    std::vector<PodEntry> pods;  // front = most recently added
    std::list<uint64_t>::iterator lru_it;
  };
  // TinyLFU-style frequency sketch (byte-budget mode): two 8-bit counters
  // per key (min estimate), halved every 8x-table-size bumps so history
  // ages out. Plays ristretto's admission role (reference
  // cost_aware_memory.go:41-52): a one-shot scan flood cannot evict keys
  // that lookups keep hot.
  struct FreqSketch {
    std::vector<uint8_t> c;
    uint64_t ops = 0;
    static constexpr size_t kSlots = 8192;  // power of two

    void ensure() {
      if (c.empty()) c.assign(kSlots, 0);
    }
    static uint64_t h2(uint64_t k) { return k * 0x9E3779B97F4A7C15ull; }
    void bump(uint64_t k) {
      ensure();
      const size_t i1 = k & (kSlots - 1), i2 = h2(k) & (kSlots - 1);
      if (c[i1] < 255) c[i1]++;
      if (i2 != i1 && c[i2] < 255) c[i2]++;  // colliding slots count once
      if (++ops >= kSlots * 8) {
        for (auto& x : c) x >>= 1;
        ops = 0;
      }
    }
    uint32_t estimate(uint64_t k) const {
      if (c.empty()) return 0;
      return std::min(c[k & (kSlots - 1)], c[h2(k) & (kSlots - 1)]);
    }
  };

  struct Shard {
    std::mutex mu;
    std::unordered_map<uint64_t, KeyEntry> map;
    std::list<uint64_t> lru;  // front = most recent
    size_t bytes = 0;  // approximate resident cost (byte-budget mode)
    FreqSketch freq;   // admission sketch (byte-budget mode only)

    Shard() = default;
    Shard(const Shard&) {}
    void touch(std::unordered_map<uint64_t, KeyEntry>::iterator it) {
      lru.splice(lru.begin(), lru, it->second.lru_it);
    }
  };
  struct EngEntry {
    std::vector<uint64_t> rks;
    std::list<uint64_t>::iterator lru_it;
  };
  struct EngShard {
    std::mutex mu;
    std::unordered_map<uint64_t, EngEntry> map;
    std::list<uint64_t> lru;

    EngShard() = default;
    EngShard(const EngShard&) {}
    void touch(std::unordered_map<uint64_t, EngEntry>::iterator it) {
      lru.splice(lru.begin(), lru, it->second.lru_it);
    }
  };

  static uint64_t mix(uint64_t k) {
    // splitmix64 finalizer: decorrelates shard choice from raw block hashes.
    k += 0x9e3779b97f4a7c15ull;
    k = (k ^ (k >> 30)) * 0xbf58476d1ce4e5b9ull;
    k = (k ^ (k >> 27)) * 0x94d049bb133111ebull;
    return k ^ (k >> 31);
  }
  Shard& shard(uint64_t key) { return shards_[mix(key) & (cfg_.shards - 1)]; }
  EngShard& eng_shard(uint64_t key) {
    return eng_shards_[mix(key) & (cfg_.shards - 1)];
  }

  void put_engine_mapping(uint64_t ek, const std::vector<uint64_t>& rks) {
    EngShard& es = eng_shard(ek);
    std::lock_guard<std::mutex> g(es.mu);
    auto it = es.map.find(ek);
    if (it == es.map.end()) {
      it = es.map.emplace(ek, EngEntry{}).first;
      es.lru.push_front(ek);
      it->second.lru_it = es.lru.begin();
      if (es.map.size() > shard_cap_) {
        uint64_t victim = es.lru.back();
        es.lru.pop_back();
        es.map.erase(victim);
      }
    } else {
      es.touch(it);
    }
    it->second.rks = rks;
  }

  void evict_from_request_key(uint64_t key, const std::vector<PodEntry>& entries) {
    Shard& sh = shard(key);
    std::lock_guard<std::mutex> g(sh.mu);
    auto it = sh.map.find(key);
    if (it == sh.map.end()) return;
    auto& pods = it->second.pods;
    for (const auto& e : entries) {
      for (size_t i = 0; i < pods.size(); ++i) {
        if (PodEntry::evict_match(pods[i], e)) {
          pods.erase(pods.begin() + i);
          sh.bytes -= sizeof(PodEntry);
          evictions_.fetch_add(1, std::memory_order_relaxed);
          break;
        }
      }
    }
    if (pods.empty()) {
      sh.lru.erase(it->second.lru_it);
      sh.bytes -= kKeyOverheadBytes;
      sh.map.erase(it);
    }
  }

  static constexpr size_t kKeyOverheadBytes = 96;  // map node + LRU node

  void evict_lru_locked(Shard& sh) {
    if (sh.lru.empty()) return;
    uint64_t victim = sh.lru.back();
    sh.lru.pop_back();
    auto it = sh.map.find(victim);
    if (it != sh.map.end()) {
      evictions_.fetch_add(it->second.pods.size(), std::memory_order_relaxed);
      sh.bytes -= kKeyOverheadBytes + it->second.pods.size() * sizeof(PodEntry);
      sh.map.erase(it);
    }
  }

  InMemoryIndexConfig cfg_;
  size_t shard_cap_;
  std::vector<Shard> shards_;
  std::vector<EngShard> eng_shards_;
  std::atomic<uint64_t> admissions_{0}, evictions_{0}, rejections_{0},
      lookups_{0}, hits_{0};
};

}  // namespace kvc
