// TokenProcessor: token stream -> chained KV-block keys.
//
// Chunk tokens into fixed-size blocks (partial tail dropped), then hash each
// block as FNV-64a(canonical-CBOR([parent, chunk, extra])) with the chain
// seeded by hash(FNV-64a(hash_seed), null, model_name).
//
// Capability parity with the reference's chunkedTokenDatabase
// (pkg/kvcache/kvblock/token_processor.go:61-228); implementation is a
// zero-allocation streaming encoder (see cbor.h) rather than a
// marshal-then-hash pipeline.
#pragma once

#include <cstdint>
#include <optional>
#include <stdexcept>
#include <string>
#include <string_view>
#include <vector>

#include "../common/cbor.h"
#include "../common/fnv.h"

namespace kvc {

// Per-block extra features that taint the hash (multimodal content hashes).
// An empty optional means pure text (encoded as CBOR null).
using BlockExtra = std::optional<std::vector<std::string>>;

class TokenProcessor {
 public:
  static constexpr int kDefaultBlockSize = 16;

  explicit TokenProcessor(int block_size_tokens = kDefaultBlockSize,
                          std::string hash_seed = "")
      : block_size_(block_size_tokens), hash_seed_(std::move(hash_seed)) {
    if (block_size_ <= 0)
      throw std::invalid_argument("block_size_tokens must be > 0");
    init_hash_ = fnv64a(hash_seed_);
  }

  int block_size() const { return block_size_; }

  // hash(parent, chunk, extra): one block-key step.
  static uint64_t hash_block(uint64_t parent, const uint32_t* tokens, size_t n,
                             const BlockExtra* extra) {
    Fnv64a f;
    CborEncoder<Fnv64a> enc(f);
    enc.array(3);
    enc.uint(parent);
    if (tokens == nullptr) {
      enc.null();
    } else {
      enc.array(n);
      for (size_t i = 0; i < n; ++i) enc.uint(tokens[i]);
    }
    if (extra == nullptr || !extra->has_value()) {
      enc.null();
    } else {
      enc.array((*extra)->size());
      for (const auto& s : **extra) enc.text(s);
    }
    return f.digest();
  }

  // Chain seed for a model: hash(init, null, model).
  uint64_t init_hash(std::string_view model_name) const {
    Fnv64a f;
    CborEncoder<Fnv64a> enc(f);
    enc.array(3);
    enc.uint(init_hash_);
    enc.null();
    enc.text(model_name);
    return f.digest();
  }

  // Convert tokens to chained block keys. parent_key == 0 means "start of
  // chain" (seed from model name); non-zero continues an existing chain.
  // extra, when non-empty, must have one entry per full chunk.
  std::vector<uint64_t> tokens_to_block_keys(
      uint64_t parent_key, const uint32_t* tokens, size_t n_tokens,
      std::string_view model_name,
      const std::vector<BlockExtra>* extra = nullptr) const {
    const size_t bs = static_cast<size_t>(block_size_);
    const size_t n_chunks = n_tokens / bs;  // partial tail dropped
    std::vector<uint64_t> keys;
    if (n_chunks == 0) return keys;
    if (extra != nullptr && !extra->empty() && extra->size() != n_chunks)
      throw std::invalid_argument(
          "extra features length does not match full-chunk count");

    uint64_t prefix = parent_key != 0 ? parent_key : init_hash(model_name);
    keys.reserve(n_chunks);
    for (size_t c = 0; c < n_chunks; ++c) {
      const BlockExtra* ex =
          (extra != nullptr && !extra->empty()) ? &(*extra)[c] : nullptr;
      prefix = hash_block(prefix, tokens + c * bs, bs, ex);
      keys.push_back(prefix);
    }
    return keys;
  }

 private:
  int block_size_;
  std::string hash_seed_;
  uint64_t init_hash_;
};

}  // namespace kvc
