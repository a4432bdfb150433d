// Longest-consecutive-prefix scorer with device-tier weights.
//
// Capability parity with the reference LongestPrefixScorer
// (pkg/kvcache/kvblock_scorer.go:60-154): per block a pod contributes the
// MAX weight across the tiers holding it; the walk stops for a pod at its
// first missing block. Operates on interned ids end-to-end; the active set
// is a flat vector updated in place (no per-key allocation).
//
// Window-aware extension (the reference marks hybrid/sliding-window-aware
// scoring WIP, docs/architecture.md:291): for pods whose model uses pure
// sliding-window attention, engines evict out-of-window leading blocks, so
// the vanilla prefix walk scores such pods 0 even though the engine can
// fully reuse the prefix (vLLM sliding-window cache-hit lookup only needs
// the last ceil(window/block) blocks of a prefix present). score() accepts
// per-pod window hints and credits such pods max over positions P of
// P x best-tier-weight where the pod holds a consecutive run ending at P of
// length >= min(window_blocks, P).
#pragma once

#include <algorithm>
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
  // window_hints: optional pod id -> window length in blocks for pods with
  // pure sliding-window attention (see header comment).
  // Returns pod id -> accumulated weight over its consecutive prefix.
  std::unordered_map<uint32_t, double> score(
      const std::vector<uint64_t>& keys,
      const std::vector<std::pair<uint64_t, std::vector<PodEntry>>>& hits,
      const std::unordered_map<uint32_t, int32_t>* window_hints = nullptr) const {
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

    if (window_hints != nullptr && !window_hints->empty()) {
      for (const auto& [pod, wblocks] : *window_hints) {
        if (wblocks <= 0) continue;
        int64_t run = 0;
        double run_w = 0.0;  // best tier weight inside the current run
        double best = 0.0;
        for (size_t i = 0; i < keys.size(); ++i) {
          double w = -1.0;
          auto it = by_key.find(keys[i]);
          if (it != by_key.end()) {
            for (const auto& e : *it->second) {
              if (e.pod != pod) continue;
              double ww = weight(e.tier);
              if (ww > w) w = ww;
            }
          }
          if (w < 0.0) {
            run = 0;
            run_w = 0.0;
            continue;
          }
          ++run;
          if (w > run_w) run_w = w;
          const int64_t p = static_cast<int64_t>(i) + 1;
          if (run >= std::min<int64_t>(wblocks, p)) {
            double s = static_cast<double>(p) * run_w;
            if (s > best) best = s;
          }
        }
        if (best > 0.0) {
          auto [si, inserted] = scores.emplace(pod, best);
          if (!inserted && best > si->second) si->second = best;
        }
      }
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
