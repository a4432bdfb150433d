// Pinned host-DRAM cache tier: the middle of the HBM -> pinned DRAM ->
// filesystem hierarchy.
//
// Stores are write-through: the packed slab lands in a pinned cache slot,
// the file is written from that slot, and the slot stays resident keyed by
// the (content-addressed) file path. A load that hits skips the filesystem
// read — pure SDMA H2D at the wire limit instead of a page-cache memcpy
// plus the DMA. Content addressing makes staleness impossible: a path's
// bytes never change, so an entry is valid until evicted (LRU over fixed
// uniform slots; referenced slots are never evicted).
#pragma once

#include <list>
#include <memory>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "common.h"

namespace kvo {

class HostPinnedCache {
 public:
  struct Slot {
    std::unique_ptr<HostStaging> buf;
    std::string key;
    size_t bytes_used = 0;
    int refs = 0;
    bool valid = false;
    std::list<Slot*>::iterator lru_it;
  };

  HostPinnedCache(size_t cache_bytes, size_t slot_bytes, bool gpu_mode) {
    size_t n = slot_bytes ? cache_bytes / slot_bytes : 0;
    slots_.reserve(n);
    for (size_t i = 0; i < n; ++i) {
      auto s = std::make_unique<Slot>();
      s->buf = std::make_unique<HostStaging>(slot_bytes, gpu_mode);
      lru_.push_front(s.get());
      s->lru_it = lru_.begin();
      slots_.push_back(std::move(s));
    }
    KVO_LOG_INFO("host cache: %zu slots x %zu MiB", n, slot_bytes >> 20);
  }

  bool enabled() const { return !slots_.empty(); }

  // Hit: returns a referenced, valid slot for the key (caller must
  // release()); miss: nullptr.
  Slot* lookup(const std::string& key) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = map_.find(key);
    if (it == map_.end() || !it->second->valid) return nullptr;
    Slot* s = it->second;
    s->refs++;
    lru_.splice(lru_.begin(), lru_, s->lru_it);
    hits_++;
    return s;
  }

  // Acquire a slot to fill for `key` (store path or load-miss populate).
  // Evicts the LRU unreferenced slot. nullptr when every slot is busy.
  Slot* acquire(const std::string& key, size_t bytes) {
    std::lock_guard<std::mutex> g(mu_);
    if (slots_.empty() || bytes > slots_[0]->buf->size()) return nullptr;
    auto it = map_.find(key);
    if (it != map_.end()) {
      // someone already caches (or is filling) this key; don't duplicate
      if (it->second->valid) return nullptr;
      return nullptr;
    }
    for (auto rit = lru_.rbegin(); rit != lru_.rend(); ++rit) {
      Slot* s = *rit;
      if (s->refs == 0) {
        if (!s->key.empty()) map_.erase(s->key);
        s->key = key;
        s->valid = false;
        s->bytes_used = bytes;
        s->refs = 1;
        map_[key] = s;
        lru_.splice(lru_.begin(), lru_, s->lru_it);
        return s;
      }
    }
    return nullptr;
  }

  // Mark a filled slot usable by future lookups.
  void publish(Slot* s) {
    std::lock_guard<std::mutex> g(mu_);
    s->valid = true;
  }

  // Drop an acquired slot that was not successfully filled.
  void abandon(Slot* s) {
    std::lock_guard<std::mutex> g(mu_);
    map_.erase(s->key);
    s->key.clear();
    s->valid = false;
    s->refs--;
    lru_.splice(lru_.end(), lru_, s->lru_it);  // immediate eviction candidate
  }

  void release(Slot* s) {
    std::lock_guard<std::mutex> g(mu_);
    s->refs--;
  }

  // Extra reference on a held slot (e.g. handing it to an async flush).
  void addref(Slot* s) {
    std::lock_guard<std::mutex> g(mu_);
    s->refs++;
  }

  uint64_t hit_count() const { return hits_; }

 private:
  std::mutex mu_;
  std::vector<std::unique_ptr<Slot>> slots_;
  std::list<Slot*> lru_;  // front = most recent
  std::unordered_map<std::string, Slot*> map_;
  uint64_t hits_ = 0;
};

}  // namespace kvo
