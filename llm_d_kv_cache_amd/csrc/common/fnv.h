// FNV-64a streaming hasher.
//
// The block-key chain is FNV-64a over the canonical-CBOR encoding of
// [parent, tokens, extra] (capability parity with the reference's
// pkg/kvcache/kvblock/token_processor.go:146-158, re-designed here as a
// zero-allocation streaming hasher: CBOR bytes are folded into the hash as
// they are produced, no intermediate buffer).
#pragma once

#include <cstddef>
#include <cstdint>
#include <string_view>

namespace kvc {

inline constexpr uint64_t kFnv64Offset = 0xcbf29ce484222325ull;
inline constexpr uint64_t kFnv64Prime = 0x100000001b3ull;

struct Fnv64a {
  uint64_t h = kFnv64Offset;

  inline void update(uint8_t b) {
    h ^= b;
    h *= kFnv64Prime;
  }
  inline void update(const void* data, size_t n) {
    const uint8_t* p = static_cast<const uint8_t*>(data);
    for (size_t i = 0; i < n; ++i) update(p[i]);
  }
  inline void update(std::string_view s) { update(s.data(), s.size()); }
  inline uint64_t digest() const { return h; }
};

inline uint64_t fnv64a(std::string_view s) {
  Fnv64a f;
  f.update(s);
  return f.digest();
}

// FNV-1a 32-bit: used for sharding pod-ids onto event-pool workers
// (parity with the reference's pool.go:161-173 sharding function).
inline uint32_t fnv32a(std::string_view s) {
  uint32_t h = 2166136261u;
  for (unsigned char c : s) {
    h ^= c;
    h *= 16777619u;
  }
  return h;
}

}  // namespace kvc
