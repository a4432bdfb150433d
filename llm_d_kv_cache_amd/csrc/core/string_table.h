// Interning table for pod identifiers and device-tier names.
//
// The index stores pod entries as fixed-size 12-byte records of interned ids
// instead of strings, so the Lookup/Add hot loops compare and hash u32s.
// Strings only cross this boundary at the Python API edge.
#pragma once

#include <cstdint>
#include <mutex>
#include <shared_mutex>
#include <string>
#include <string_view>
#include <unordered_map>
#include <vector>

namespace kvc {

class StringTable {
 public:
  static constexpr uint32_t kInvalid = 0xffffffffu;

  uint32_t intern(std::string_view s) {
    {
      std::shared_lock rl(mu_);
      auto it = ids_.find(std::string(s));
      if (it != ids_.end()) return it->second;
    }
    std::unique_lock wl(mu_);
    auto [it, inserted] = ids_.emplace(std::string(s), uint32_t(strings_.size()));
    if (inserted) strings_.push_back(it->first);
    return it->second;
  }

  // Returns kInvalid when the string was never interned (e.g. a pod filter
  // naming a pod the index has never seen — semantically "no entries").
  uint32_t find(std::string_view s) const {
    std::shared_lock rl(mu_);
    auto it = ids_.find(std::string(s));
    return it == ids_.end() ? kInvalid : it->second;
  }

  size_t size() const {
    std::shared_lock rl(mu_);
    return strings_.size();
  }

  std::string get(uint32_t id) const {
    std::shared_lock rl(mu_);
    return id < strings_.size() ? strings_[id] : std::string();
  }

 private:
  mutable std::shared_mutex mu_;
  std::unordered_map<std::string, uint32_t> ids_;
  std::vector<std::string> strings_;  // stable copies; id = position
};

}  // namespace kvc
