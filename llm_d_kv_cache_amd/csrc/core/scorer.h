// Longest-consecutive-prefix scorer with device-tier weights.
//
// Capability parity with the reference LongestPrefixScorer
// (pkg/kvcache/kvblock_scorer.go:60-154): per block a pod contributes the
// MAX weight across the tiers holding it; the walk stops for a pod at its
// first missing block. Operates on interned ids end-to-end; the active set
// is a flat vector updated in place (no per-key allocation).
//
// Hybrid/window-aware extension (the reference marks this WIP,
// docs/architecture.md:291): for pods whose KV-cache group structure is
// known (learned from BlockStored HMA fields), the score is computed with
// a per-group walk and the pod's reusable prefix is the MIN over its
// groups — full-attention groups need a strict prefix, sliding-window
// groups only need a consecutive run ending at P covering
// min(ceil(window/block), P) blocks (engines evict out-of-window leading
// blocks, and their cache-hit lookup tolerates exactly that). This both
// credits sliding-window pods the vanilla walk would score 0 AND removes
// the vanilla walk's over-optimism for hybrid pods that lost one group's
// blocks but not the other's. Entries without a group act as wildcards.
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

  // (group id, window length in blocks; <=0 = full attention)
  using GroupWindows = std::vector<std::pair<int32_t, int32_t>>;

  // keys: the full ordered key list; hits: Lookup output (ordered subset).
  // window_hints: optional pod id -> group/window structure; hinted pods
  // get the per-group hybrid walk (replacing their vanilla score).
  // Returns pod id -> accumulated weight over its reusable prefix.
  std::unordered_map<uint32_t, double> score(
      const std::vector<uint64_t>& keys,
      const std::vector<std::pair<uint64_t, std::vector<PodEntry>>>& hits,
      const std::unordered_map<uint32_t, GroupWindows>* window_hints =
          nullptr) const {
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
      for (const auto& [pod, groups] : *window_hints) {
        if (groups.empty()) continue;
        int64_t p_pod = static_cast<int64_t>(keys.size());
        double w_best = 0.0;
        for (const auto& [gid, wblocks] : groups) {
          int64_t p_g = 0;
          int64_t run = 0;
          double run_w = 0.0;
          for (size_t i = 0; i < keys.size(); ++i) {
            double w = -1.0;
            auto it = by_key.find(keys[i]);
            if (it != by_key.end()) {
              for (const auto& e : *it->second) {
                if (e.pod != pod) continue;
                if (e.has_group() && e.group != gid) continue;
                double ww = weight(e.tier);
                if (ww > w) w = ww;
              }
            }
            if (w < 0.0) {
              if (wblocks <= 0) break;  // full attention: strict prefix ends
              run = 0;
              run_w = 0.0;
              continue;
            }
            ++run;
            if (w > run_w) run_w = w;
            const int64_t p = static_cast<int64_t>(i) + 1;
            if (wblocks <= 0) {
              if (run == p) {  // still a strict prefix
                p_g = p;
                if (run_w > w_best) w_best = run_w;
              }
            } else if (run >= std::min<int64_t>(wblocks, p)) {
              if (p > p_g) {
                p_g = p;
                if (run_w > w_best) w_best = run_w;
              }
            }
          }
          if (p_g < p_pod) p_pod = p_g;
          if (p_pod == 0) break;
        }
        // Replace the vanilla (any-entry) score: the hybrid walk is the
        // authoritative model of what the engine can actually reuse.
        if (p_pod > 0 && w_best > 0.0) {
          scores[pod] = static_cast<double>(p_pod) * w_best;
        } else {
          scores.erase(pod);
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
