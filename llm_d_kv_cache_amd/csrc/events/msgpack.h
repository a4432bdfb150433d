// Minimal msgpack cursor decoder for the KVEvents wire format.
//
// Single-pass, zero-copy (string/bin views point into the payload buffer),
// no generic "any" tree: each event field is pulled positionally. Covers the
// subset engines emit via msgspec (array_like=True, omit_defaults=True):
// nil/bool/int/uint/float/str/bin/array/map. Unknown trailing fields are
// skippable for forward compatibility.
#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string_view>

namespace kvc {

struct MsgpackError : std::runtime_error {
  using std::runtime_error::runtime_error;
};

class MsgCursor {
 public:
  MsgCursor(const uint8_t* data, size_t n) : p_(data), end_(data + n) {}

  bool done() const { return p_ >= end_; }
  const uint8_t* pos() const { return p_; }
  size_t remaining() const { return static_cast<size_t>(end_ - p_); }

  // An N-element container needs at least N bytes of payload; a declared
  // count beyond that is malformed (and would otherwise drive unbounded
  // reserve() on attacker-controlled input).
  uint32_t bounded_len(uint32_t n) const {
    if (n > remaining()) throw MsgpackError("declared count exceeds payload");
    return n;
  }

  uint8_t peek() const {
    need(1);
    return *p_;
  }

  bool is_nil() const { return !done() && *p_ == 0xc0; }
  void nil() {
    expect(0xc0, "nil");
    ++p_;
  }

  bool is_array() const {
    if (done()) return false;
    uint8_t b = *p_;
    return (b >= 0x90 && b <= 0x9f) || b == 0xdc || b == 0xdd;
  }
  bool is_str() const {
    if (done()) return false;
    uint8_t b = *p_;
    return (b >= 0xa0 && b <= 0xbf) || b == 0xd9 || b == 0xda || b == 0xdb;
  }
  bool is_bin() const {
    if (done()) return false;
    uint8_t b = *p_;
    return b == 0xc4 || b == 0xc5 || b == 0xc6;
  }
  bool is_int() const {
    if (done()) return false;
    uint8_t b = *p_;
    return b <= 0x7f || b >= 0xe0 || (b >= 0xcc && b <= 0xd3);
  }

  uint32_t array_len() {
    uint8_t b = take();
    if (b >= 0x90 && b <= 0x9f) return b & 0x0f;
    if (b == 0xdc) return be16();
    if (b == 0xdd) return be32();
    throw MsgpackError("expected array");
  }

  uint32_t map_len() {
    uint8_t b = take();
    if (b >= 0x80 && b <= 0x8f) return b & 0x0f;
    if (b == 0xde) return be16();
    if (b == 0xdf) return be32();
    throw MsgpackError("expected map");
  }

  std::string_view str() {
    uint8_t b = take();
    uint32_t n;
    if (b >= 0xa0 && b <= 0xbf)
      n = b & 0x1f;
    else if (b == 0xd9)
      n = take();
    else if (b == 0xda)
      n = be16();
    else if (b == 0xdb)
      n = be32();
    else
      throw MsgpackError("expected str");
    need(n);
    std::string_view s(reinterpret_cast<const char*>(p_), n);
    p_ += n;
    return s;
  }

  std::string_view bin() {
    uint8_t b = take();
    uint32_t n;
    if (b == 0xc4)
      n = take();
    else if (b == 0xc5)
      n = be16();
    else if (b == 0xc6)
      n = be32();
    else
      throw MsgpackError("expected bin");
    need(n);
    std::string_view s(reinterpret_cast<const char*>(p_), n);
    p_ += n;
    return s;
  }

  // Signed/unsigned integers; uint64 values above INT64_MAX round-trip via
  // the uint64 accessor.
  int64_t int64() { return static_cast<int64_t>(uint64_impl(true)); }
  uint64_t uint64() { return uint64_impl(false); }

  double f64() {
    uint8_t b = take();
    if (b == 0xca) {
      uint32_t v = be32();
      float f;
      std::memcpy(&f, &v, 4);
      return f;
    }
    if (b == 0xcb) {
      uint64_t v = be64();
      double d;
      std::memcpy(&d, &v, 8);
      return d;
    }
    // msgspec may emit an int when the float is integral.
    --p_;
    return static_cast<double>(int64());
  }

  bool boolean() {
    uint8_t b = take();
    if (b == 0xc2) return false;
    if (b == 0xc3) return true;
    throw MsgpackError("expected bool");
  }

  void skip() {
    uint8_t b = peek();
    if (b <= 0x7f || b >= 0xe0 || b == 0xc0 || b == 0xc2 || b == 0xc3) {
      ++p_;
      return;
    }
    if (is_str()) {
      str();
      return;
    }
    if (is_bin()) {
      bin();
      return;
    }
    if (is_array()) {
      uint32_t n = array_len();
      for (uint32_t i = 0; i < n; ++i) skip();
      return;
    }
    if ((b >= 0x80 && b <= 0x8f) || b == 0xde || b == 0xdf) {
      uint32_t n = map_len();
      for (uint32_t i = 0; i < 2 * n; ++i) skip();
      return;
    }
    switch (b) {
      case 0xcc: case 0xd0: need(2); p_ += 2; return;
      case 0xcd: case 0xd1: need(3); p_ += 3; return;
      case 0xce: case 0xd2: case 0xca: need(5); p_ += 5; return;
      case 0xcf: case 0xd3: case 0xcb: need(9); p_ += 9; return;
      default: throw MsgpackError("skip: unsupported type byte");
    }
  }

 private:
  uint64_t uint64_impl(bool allow_neg) {
    uint8_t b = take();
    if (b <= 0x7f) return b;                           // positive fixint
    if (b >= 0xe0) {                                   // negative fixint
      if (!allow_neg) throw MsgpackError("negative where unsigned expected");
      return static_cast<uint64_t>(static_cast<int64_t>(static_cast<int8_t>(b)));
    }
    switch (b) {
      case 0xcc: return take();
      case 0xcd: return be16();
      case 0xce: return be32();
      case 0xcf: return be64();
      case 0xd0: return static_cast<uint64_t>(static_cast<int64_t>(static_cast<int8_t>(take())));
      case 0xd1: return static_cast<uint64_t>(static_cast<int64_t>(static_cast<int16_t>(be16())));
      case 0xd2: return static_cast<uint64_t>(static_cast<int64_t>(static_cast<int32_t>(be32())));
      case 0xd3: return be64();
      default: throw MsgpackError("expected int");
    }
  }

  uint8_t take() {
    need(1);
    return *p_++;
  }
  uint16_t be16() {
    need(2);
    uint16_t v = (uint16_t(p_[0]) << 8) | p_[1];
    p_ += 2;
    return v;
  }
  uint32_t be32() {
    need(4);
    uint32_t v = (uint32_t(p_[0]) << 24) | (uint32_t(p_[1]) << 16) |
                 (uint32_t(p_[2]) << 8) | p_[3];
    p_ += 4;
    return v;
  }
  uint64_t be64() {
    need(8);
    uint64_t v = 0;
    for (int i = 0; i < 8; ++i) v = (v << 8) | p_[i];
    p_ += 8;
    return v;
  }
  void need(size_t n) const {
    if (static_cast<size_t>(end_ - p_) < n) throw MsgpackError("truncated payload");
  }
  void expect(uint8_t b, const char* what) {
    if (done() || *p_ != b) throw MsgpackError(std::string("expected ") + what);
  }

  const uint8_t* p_;
  const uint8_t* end_;
};

}  // namespace kvc
