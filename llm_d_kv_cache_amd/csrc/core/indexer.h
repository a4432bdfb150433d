// Indexer orchestrator: the whole ScoreTokens read path in native code.
//
// tokens -> chained block keys -> index lookup -> longest-prefix scoring,
// one C++ call from Python with the GIL released. Capability parity with the
// reference kvcache.Indexer (pkg/kvcache/indexer.go:238-294), including the
// block-hit-ratio observability attributes.
#pragma once

#include <memory>
#include <string>
#include <unordered_map>
#include <vector>

#include "index.h"
#include "scorer.h"
#include "token_processor.h"

namespace kvc {

struct ScoreResult {
  std::unordered_map<std::string, double> scores;
  size_t total_blocks = 0;
  size_t hit_blocks = 0;  // keys with at least one (filtered) pod entry
};

class Indexer {
 public:
  Indexer(std::shared_ptr<TokenProcessor> tp, std::shared_ptr<IndexBackend> index,
          std::unordered_map<std::string, double> tier_weights)
      : tp_(std::move(tp)), index_(std::move(index)) {
    std::unordered_map<uint32_t, double> w;
    for (const auto& [tier, weight] : tier_weights)
      w.emplace(index_->strings().intern(tier), weight);
    scorer_ = std::make_unique<LongestPrefixScorer>(std::move(w));
  }

  IndexBackend& index() { return *index_; }
  TokenProcessor& token_processor() { return *tp_; }

  std::vector<uint64_t> compute_block_keys(const uint32_t* tokens, size_t n,
                                           const std::string& model) const {
    return tp_->tokens_to_block_keys(0, tokens, n, model, nullptr);
  }

  ScoreResult score_tokens(const uint32_t* tokens, size_t n, const std::string& model,
                           const std::vector<std::string>& pods,
                           const std::vector<BlockExtra>* extra = nullptr,
                           const std::unordered_map<
                               std::string, LongestPrefixScorer::GroupWindows>*
                               window_hints = nullptr) {
    ScoreResult res;
    auto keys = tp_->tokens_to_block_keys(0, tokens, n, model, extra);
    res.total_blocks = keys.size();
    if (keys.empty()) return res;

    // Intern (not find): in a multi-replica deployment this process may
    // score pods whose entries were ingested by ANOTHER process into a
    // shared backend (Redis/Valkey) — an unknown-here name must still
    // filter-match entries the backend lookup interns on decode.
    std::unordered_set<uint32_t> filter;
    for (const auto& p : pods) filter.insert(index_->strings().intern(p));

    auto hits = index_->lookup(keys, filter);
    res.hit_blocks = hits.size();
    std::unordered_map<uint32_t, LongestPrefixScorer::GroupWindows> wh;
    if (window_hints != nullptr) {
      for (const auto& [pod, groups] : *window_hints)
        wh.emplace(index_->strings().intern(pod), groups);
    }
    auto scores = scorer_->score(keys, hits, wh.empty() ? nullptr : &wh);
    for (const auto& [pod_id, s] : scores)
      res.scores.emplace(index_->strings().get(pod_id), s);
    return res;
  }

 private:
  std::shared_ptr<TokenProcessor> tp_;
  std::shared_ptr<IndexBackend> index_;
  std::unique_ptr<LongestPrefixScorer> scorer_;
};

}  // namespace kvc
