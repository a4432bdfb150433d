// KVEvents domain types + engine wire parsers (vLLM / SGLang).
//
// Wire format (capability parity with the reference's engineadapter
// package, pkg/kvevents/engineadapter/): topic "kv@<pod>@<model>", payload =
// msgpack array [ts, [event...], dp_rank?] where each event is a positional
// msgpack array produced by msgspec (array_like=True, omit_defaults=True):
//
//   BlockStored      [tag, hashes, parent, tokens, block_size, lora_id?,
//                     medium?, lora_name?, extra_keys?, group_idx?,
//                     spec_kind?, sliding_window?]
//   BlockRemoved     [tag, hashes, medium?, group_idx?]
//   AllBlocksCleared [tag]
//
// Hashes arrive as uint64/int64 or as bytes (last 8 bytes, big-endian).
// Decoding is single-pass positional with length guards (forward/backward
// compatible), not a generic any-tree.
#pragma once

#include <cstdint>
#include <optional>
#include <string>
#include <vector>

#include "msgpack.h"

namespace kvc {

enum class EventType { kBlockStored, kBlockRemoved, kAllBlocksCleared };

struct BlockStoredEvent {
  std::vector<uint64_t> block_hashes;
  uint64_t parent_hash = 0;  // 0 = none
  std::vector<uint32_t> tokens;
  int block_size = 0;
  std::optional<int64_t> lora_id;
  std::string medium;  // device tier; empty = default (gpu)
  std::optional<std::string> lora_name;
  // extra_keys: one entry per engine block; nullopt = no taint.
  bool has_extra_keys = false;
  std::vector<std::optional<std::vector<std::string>>> extra_keys;
  std::optional<int32_t> group_idx;
  std::string spec_kind;
  std::optional<int32_t> sliding_window;
};

struct BlockRemovedEvent {
  std::vector<uint64_t> block_hashes;
  std::string medium;
  std::optional<int32_t> group_idx;
};

struct ParsedEvent {
  EventType type;
  BlockStoredEvent stored;
  BlockRemovedEvent removed;
};

struct EventBatch {
  double timestamp = 0;
  std::optional<int32_t> dp_rank;
  std::vector<ParsedEvent> events;
};

// --- wire helpers -----------------------------------------------------------

inline uint64_t decode_hash(MsgCursor& c) {
  if (c.is_bin() || c.is_str()) {
    std::string_view b = c.is_bin() ? c.bin() : c.str();
    if (b.empty()) throw MsgpackError("empty hash bytes");
    uint64_t v = 0;
    size_t start = b.size() > 8 ? b.size() - 8 : 0;
    for (size_t i = start; i < b.size(); ++i)
      v = (v << 8) | static_cast<uint8_t>(b[i]);
    return v;
  }
  return c.uint64();
}

inline void parse_topic(std::string_view topic, std::string& pod, std::string& model) {
  // "kv@<pod>@<model>"; anything else: whole topic = pod, model empty.
  size_t a = topic.find('@');
  if (a != std::string_view::npos) {
    size_t b = topic.find('@', a + 1);
    if (b != std::string_view::npos && topic.find('@', b + 1) == std::string_view::npos) {
      pod.assign(topic.substr(a + 1, b - a - 1));
      model.assign(topic.substr(b + 1));
      return;
    }
  }
  pod.assign(topic);
  model.clear();
}

// Parse one event (a positional msgpack array). Throws MsgpackError on
// malformed input; unknown tags are reported via the return value.
inline bool parse_event(MsgCursor& c, ParsedEvent& out) {
  uint32_t n = c.array_len();
  if (n < 1) throw MsgpackError("event: empty tagged union");
  std::string_view tag = c.str();
  uint32_t consumed = 1;

  // Optional trailing fields tolerate unexpected types (skipped) for
  // forward compatibility with newer engine versions.
  auto maybe_nil_str = [&](std::string& dst) {
    if (c.is_str())
      dst.assign(c.str());
    else
      c.skip();
  };
  auto maybe_int = [&](std::optional<int32_t>& dst) {
    if (c.is_int())
      dst = static_cast<int32_t>(c.int64());
    else
      c.skip();
  };

  if (tag == "BlockStored") {
    out.type = EventType::kBlockStored;
    auto& e = out.stored;
    if (n < 5) throw MsgpackError("BlockStored: need at least 5 fields");
    uint32_t nh = c.bounded_len(c.array_len());
    e.block_hashes.reserve(nh);
    for (uint32_t i = 0; i < nh; ++i) e.block_hashes.push_back(decode_hash(c));
    if (c.is_nil())
      c.nil();
    else
      e.parent_hash = decode_hash(c);
    uint32_t nt = c.bounded_len(c.array_len());
    e.tokens.reserve(nt);
    for (uint32_t i = 0; i < nt; ++i)
      e.tokens.push_back(static_cast<uint32_t>(c.uint64()));
    e.block_size = static_cast<int>(c.int64());
    consumed = 5;
    if (consumed < n) {  // [5] lora_id
      if (c.is_int()) e.lora_id = c.int64(); else c.skip();
      ++consumed;
    }
    if (consumed < n) {  // [6] medium
      maybe_nil_str(e.medium);
      ++consumed;
    }
    if (consumed < n) {  // [7] lora_name
      if (c.is_str()) e.lora_name = std::string(c.str()); else c.skip();
      ++consumed;
    }
    if (consumed < n) {  // [8] extra_keys: [ nil | [str | [str, off] ...] ...]
      if (c.is_nil()) {
        c.nil();
      } else {
        e.has_extra_keys = true;
        uint32_t nb = c.bounded_len(c.array_len());
        e.extra_keys.resize(nb);
        for (uint32_t b = 0; b < nb; ++b) {
          if (c.is_nil()) {
            c.nil();
            continue;
          }
          uint32_t nk = c.array_len();
          std::vector<std::string> hashes;
          for (uint32_t k = 0; k < nk; ++k) {
            if (c.is_str()) {
              hashes.emplace_back(c.str());
            } else if (c.is_array()) {
              // legacy [hash, offset] tuple: keep hash, ignore offset
              uint32_t m = c.array_len();
              if (m >= 1 && c.is_str()) {
                hashes.emplace_back(c.str());
                for (uint32_t j = 1; j < m; ++j) c.skip();
              } else {
                for (uint32_t j = 0; j < m; ++j) c.skip();
              }
            } else {
              c.skip();  // unknown entry kinds (LoRA, cache salt) are ignored
            }
          }
          if (!hashes.empty()) e.extra_keys[b] = std::move(hashes);
        }
      }
      ++consumed;
    }
    if (consumed < n) {  // [9] group_idx
      maybe_int(e.group_idx);
      ++consumed;
    }
    if (consumed < n) {  // [10] kv_cache_spec_kind
      maybe_nil_str(e.spec_kind);
      ++consumed;
    }
    if (consumed < n) {  // [11] sliding window
      maybe_int(e.sliding_window);
      ++consumed;
    }
    for (; consumed < n; ++consumed) c.skip();  // newer-engine trailing fields
    return true;
  }

  if (tag == "BlockRemoved") {
    out.type = EventType::kBlockRemoved;
    auto& e = out.removed;
    if (n < 2) throw MsgpackError("BlockRemoved: need at least 2 fields");
    uint32_t nh = c.bounded_len(c.array_len());
    e.block_hashes.reserve(nh);
    for (uint32_t i = 0; i < nh; ++i) e.block_hashes.push_back(decode_hash(c));
    consumed = 2;
    if (consumed < n) {
      maybe_nil_str(e.medium);
      ++consumed;
    }
    if (consumed < n) {
      maybe_int(e.group_idx);
      ++consumed;
    }
    for (; consumed < n; ++consumed) c.skip();
    return true;
  }

  if (tag == "AllBlocksCleared") {
    out.type = EventType::kAllBlocksCleared;
    for (; consumed < n; ++consumed) c.skip();
    return true;
  }

  for (; consumed < n; ++consumed) c.skip();
  return false;  // unknown tag: skipped
}

// Parse a full batch payload: [ts, [event...], dp_rank?].
inline EventBatch parse_batch(const uint8_t* data, size_t n) {
  MsgCursor c(data, n);
  EventBatch batch;
  uint32_t nf = c.array_len();
  if (nf < 2) throw MsgpackError("batch: need [ts, events]");
  batch.timestamp = c.f64();
  uint32_t ne = c.bounded_len(c.array_len());
  batch.events.reserve(ne);
  for (uint32_t i = 0; i < ne; ++i) {
    ParsedEvent ev;
    if (parse_event(c, ev)) batch.events.push_back(std::move(ev));
  }
  if (nf > 2) {
    if (c.is_nil()) c.nil(); else batch.dp_rank = static_cast<int32_t>(c.int64());
  }
  for (uint32_t i = 3; i < nf; ++i) c.skip();
  return batch;
}

}  // namespace kvc
