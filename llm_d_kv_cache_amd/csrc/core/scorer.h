// Longest-consecutive-prefix scorer with device-tier weights.
//
// Capability parity with the reference LongestPrefixScorer
// (pkg/kvcache/kvblock_scorer.go:60-154): per block a pod contributes the
// MAX weight across the tiers holding it; the walk stops for a pod at its
// first missing block. Operates on interned ids end-to-end; the active set
// is a flat vector updated in place (no per-key allocation).
#pragma once

#include <cstdint>
#include <unordered_map>
#include <vector>

#include "index.h"

namespace kvc {

class LongestPrefixScorer {
 public:
  // tier_weights: interned tier id -> weight. Unlisted tiers weigh 1.0.
  explicit LongestPrefixScorer(std::unordered_map<uint32_t, double> tier_weights = {})
      : tier_weights_(std::move(tier_weights)) {}

  // keys: the full ordered key list; hits: Lookup output (ordered subset).
  // Returns pod id -> accumulated weight over its consecutive prefix.
  std::unordered_map<uint32_t, double> score(
      const std::vector<uint64_t>& keys,
      const std::vector<std::pair<uint64_t, std::vector<PodEntry>>>& hits) const {
    std::unordered_map<uint32_t, double> scores;
    if (keys.empty() || hits.empty()) return scores;

    // Index hits by key for the consecutive walk (hits are ordered but may
    // skip keys; a skipped key ends every pod's chain).
    std::unordered_map<uint64_t, const std::vector<PodEntry>*> by_key;
    by_key.reserve(hits.size());
    for (const auto& [k, pods] : hits) by_key.emplace(k, &pods);

    std::unordered_map<uint32_t, double> cur;  // scratch: pod -> max weight
    std::vector<uint32_t> active;

    for (size_t i = 0; i < keys.size(); ++i) {
      auto it = by_key.find(keys[i]);
      if (it == by_key.end()) break;  // chain broken for every pod
      cur.clear();
      for (const auto& e : *it->second) {
        double w = weight(e.tier);
        auto [ci, inserted] = cur.emplace(e.pod, w);
        if (!inserted && w > ci->second) ci->second = w;
      }
      if (i == 0) {
        for (const auto& [pod, w] : cur) {
          active.push_back(pod);
          scores[pod] = w;
        }
      } else {
        size_t j = 0;
        for (size_t a = 0; a < active.size(); ++a) {
          auto ci = cur.find(active[a]);
          if (ci != cur.end()) {
            scores[active[a]] += ci->second;
            active[j++] = active[a];
          }
        }
        active.resize(j);
      }
      if (active.empty()) break;
    }
    return scores;
  }

 private:
  double weight(uint32_t tier) const {
    auto it = tier_weights_.find(tier);
    return it == tier_weights_.end() ? 1.0 : it->second;
  }

  std::unordered_map<uint32_t, double> tier_weights_;
};

}  // namespace kvc
