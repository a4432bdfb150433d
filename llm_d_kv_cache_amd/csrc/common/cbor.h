// Minimal canonical CBOR (RFC 8949) *streaming encoder*, specialized for the
// block-hash payload shape [parent:uint64, tokens:[uint32...], extra].
//
// Only the subset the hash chain needs is implemented: unsigned integers
// (shortest form), text strings, arrays, null. Instead of materializing
// bytes, every emit folds directly into a sink (the FNV hasher), so the hot
// scoring loop does zero heap allocation per block.
//
// Capability parity: the reference hashes the canonical-CBOR encoding of the
// same payload (token_processor.go:146-158); this is an independent
// implementation of RFC 8949 deterministic encoding.
#pragma once

#include <cstdint>
#include <string_view>

namespace kvc {

// Sink must provide update(uint8_t) and update(const void*, size_t).
template <typename Sink>
class CborEncoder {
 public:
  explicit CborEncoder(Sink& sink) : sink_(sink) {}

  // Major type 0: unsigned integer, shortest form (deterministic encoding).
  void uint(uint64_t v) { head(0, v); }

  // Major type 3: text string header + payload bytes.
  void text(std::string_view s) {
    head(3, s.size());
    sink_.update(s.data(), s.size());
  }

  // Major type 4: array header; caller then emits `n` items.
  void array(uint64_t n) { head(4, n); }

  // Simple value 22: null.
  void null() { sink_.update(uint8_t(0xf6)); }

 private:
  void head(uint8_t major, uint64_t v) {
    const uint8_t m = major << 5;
    if (v < 24) {
      sink_.update(uint8_t(m | v));
    } else if (v <= 0xff) {
      sink_.update(uint8_t(m | 24));
      sink_.update(uint8_t(v));
    } else if (v <= 0xffff) {
      sink_.update(uint8_t(m | 25));
      be(v, 2);
    } else if (v <= 0xffffffffull) {
      sink_.update(uint8_t(m | 26));
      be(v, 4);
    } else {
      sink_.update(uint8_t(m | 27));
      be(v, 8);
    }
  }

  void be(uint64_t v, int bytes) {
    for (int i = bytes - 1; i >= 0; --i) sink_.update(uint8_t((v >> (8 * i)) & 0xff));
  }

  Sink& sink_;
};

}  // namespace kvc
