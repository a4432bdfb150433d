// Network-backed index for Redis/Valkey: the shared-state backend for
// multi-replica deployments (capability parity with the reference
// pkg/kvcache/kvblock/redis.go — Valkey is wire-compatible, so one
// implementation serves both).
//
// Data model (strings on the wire, interned ids only in-process):
//   kv:r:<16-hex requestKey>  HASH   field "pod\x1ftier\x1fflags\x1fgroup" -> "1"
//   kv:e:<16-hex engineKey>   STRING comma-separated requestKey hex list
//
// Lookup pipelines one HGETALL per key (one round trip for the whole
// prefix walk); Add pipelines HSET + SET; Evict prunes the engine mapping
// once every spanned request key is empty.
#pragma once

#include <charconv>
#include <cstdio>
#include <sstream>

#include "index.h"
#include "resp.h"

namespace kvc {

struct RedisIndexConfig {
  std::string host = "127.0.0.1";
  int port = 6379;
  size_t pool_size = 4;
  std::string key_prefix = "kv";
};

class RedisIndex : public IndexBackend {
 public:
  explicit RedisIndex(const RedisIndexConfig& cfg = {})
      : cfg_(cfg), pool_(cfg.host, cfg.port, cfg.pool_size) {
    // fail fast on misconfiguration
    pool_.with([](RespConnection& c) { return c.command({"PING"}); });
  }

  std::vector<std::pair<uint64_t, std::vector<PodEntry>>> lookup(
      const std::vector<uint64_t>& request_keys,
      const std::unordered_set<uint32_t>& pod_filter) override {
    if (request_keys.empty())
      throw std::invalid_argument("no request keys provided for lookup");
    lookups_.fetch_add(1, std::memory_order_relaxed);
    std::vector<std::vector<std::string>> cmds;
    cmds.reserve(request_keys.size());
    for (uint64_t k : request_keys) cmds.push_back({"HGETALL", rkey(k)});
    auto replies = pool_.with(
        [&](RespConnection& c) { return c.pipeline(cmds); });

    std::vector<std::pair<uint64_t, std::vector<PodEntry>>> out;
    bool any = false;
    for (size_t i = 0; i < request_keys.size(); ++i) {
      if (replies[i].is_error || replies[i].is_nil()) continue;
      const auto& arr = replies[i].array();
      if (arr.empty()) continue;  // absent key: skipped (scorer breaks chain)
      std::vector<PodEntry> pods;
      for (size_t j = 0; j + 1 < arr.size(); j += 2) {
        PodEntry e;
        if (!decode_entry(arr[j].str(), &e)) continue;
        if (pod_filter.empty() || pod_filter.count(e.pod)) pods.push_back(e);
      }
      if (!pods.empty()) {
        any = true;
        out.emplace_back(request_keys[i], std::move(pods));
      }
    }
    if (any) hits_.fetch_add(1, std::memory_order_relaxed);
    return out;
  }

  void add(const std::vector<uint64_t>& engine_keys,
           const std::vector<uint64_t>& request_keys,
           const std::vector<PodEntry>& entries) override {
    if (request_keys.empty() || entries.empty())
      throw std::invalid_argument("no keys or entries provided for add");
    std::vector<std::vector<std::string>> cmds;
    for (uint64_t rk : request_keys) {
      std::vector<std::string> cmd{"HSET", rkey(rk)};
      for (const auto& e : entries) {
        cmd.push_back(encode_entry(e));
        cmd.push_back("1");
      }
      cmds.push_back(std::move(cmd));
    }
    if (!engine_keys.empty()) {
      const size_t ne = engine_keys.size(), nr = request_keys.size();
      const size_t n = std::max(ne, nr);
      uint64_t cur_ek = 0;
      std::string rks;
      bool have = false;
      auto flush = [&]() {
        if (have) cmds.push_back({"SET", ekey(cur_ek), rks});
      };
      for (size_t i = 0; i < n; ++i) {
        uint64_t ek = engine_keys[i * ne / n];
        uint64_t rk = request_keys[i * nr / n];
        if (!have || ek != cur_ek) {
          flush();
          cur_ek = ek;
          rks.clear();
          have = true;
        }
        if (!rks.empty()) rks += ",";
        rks += hex(rk);
      }
      flush();
    }
    pool_.with([&](RespConnection& c) { return c.pipeline(cmds); });
    admissions_.fetch_add(entries.size() * request_keys.size(),
                          std::memory_order_relaxed);
  }

  void evict(uint64_t key, KeyType type,
             const std::vector<PodEntry>& entries) override {
    if (entries.empty())
      throw std::invalid_argument("no entries provided for evict");
    std::vector<uint64_t> rks;
    if (type == KeyType::kRequest) {
      rks.push_back(key);
    } else {
      auto reply = pool_.with(
          [&](RespConnection& c) { return c.command({"GET", ekey(key)}); });
      if (reply.is_nil() || reply.is_error) return;
      rks = parse_rk_list(reply.str());
      if (rks.empty()) return;
    }
    // BlockRemoved may omit group/flags the stored field encodes, so the
    // exact HDEL field is not constructible client-side: read the fields,
    // apply the relaxed evict_match, delete the exact matches.
    std::vector<std::vector<std::string>> reads;
    for (uint64_t rk : rks) reads.push_back({"HKEYS", rkey(rk)});
    auto fields = pool_.with(
        [&](RespConnection& c) { return c.pipeline(reads); });
    std::vector<std::vector<std::string>> cmds;
    for (size_t ri = 0; ri < rks.size(); ++ri) {
      std::vector<std::string> cmd{"HDEL", rkey(rks[ri])};
      if (!fields[ri].is_error && !fields[ri].is_nil()) {
        for (const auto& f : fields[ri].array()) {
          PodEntry stored;
          if (!decode_entry(f.str(), &stored)) continue;
          for (const auto& e : entries) {
            if (PodEntry::evict_match(stored, e)) {
              cmd.push_back(f.str());
              break;
            }
          }
        }
      }
      if (cmd.size() > 2) cmds.push_back(std::move(cmd));
    }
    if (!cmds.empty())
      pool_.with([&](RespConnection& c) { return c.pipeline(cmds); });
    evictions_.fetch_add(1, std::memory_order_relaxed);
    if (type == KeyType::kEngine) {
      // Atomic prune-if-empty: the empty-check and the deletes run as ONE
      // server-side Lua script, so a concurrent add() interleaving between
      // "hash is empty" and "DEL engine key" cannot lose entries (parity
      // with the reference's Lua prune, redis.go:160-169; the old separate
      // HLEN -> DEL round trips had exactly that race).
      std::vector<std::string> ev{"EVAL", prune_script(),
                                  std::to_string(rks.size() + 1)};
      for (uint64_t rk : rks) ev.push_back(rkey(rk));
      ev.push_back(ekey(key));
      pool_.with([&](RespConnection& c) { return c.command(ev); });
    }
  }

  bool get_request_key(uint64_t engine_key, uint64_t* out) override {
    auto reply = pool_.with(
        [&](RespConnection& c) { return c.command({"GET", ekey(engine_key)}); });
    if (reply.is_nil() || reply.is_error) return false;
    auto rks = parse_rk_list(reply.str());
    if (rks.empty()) return false;
    *out = rks.back();
    return true;
  }

  void clear(uint32_t pod_id) override {
    // SCAN the request-key namespace and prune this pod's fields.
    std::string pod = strings_.get(pod_id);
    std::string cursor = "0";
    std::string pattern = cfg_.key_prefix + ":r:*";
    do {
      auto reply = pool_.with([&](RespConnection& c) {
        return c.command({"SCAN", cursor, "MATCH", pattern, "COUNT", "512"});
      });
      if (reply.is_error || reply.is_nil()) break;
      const auto& arr = reply.array();
      cursor = arr[0].str();
      std::vector<std::string> keys;
      for (const auto& k : arr[1].array()) keys.push_back(k.str());
      if (keys.empty()) continue;
      std::vector<std::vector<std::string>> gets;
      for (const auto& k : keys) gets.push_back({"HGETALL", k});
      auto hashes = pool_.with(
          [&](RespConnection& c) { return c.pipeline(gets); });
      std::vector<std::vector<std::string>> dels;
      for (size_t i = 0; i < keys.size(); ++i) {
        if (hashes[i].is_error || hashes[i].is_nil()) continue;
        std::vector<std::string> cmd{"HDEL", keys[i]};
        size_t matched = 0;
        const auto& fields = hashes[i].array();
        for (size_t j = 0; j + 1 < fields.size(); j += 2) {
          const std::string& f = fields[j].str();
          size_t sep = f.find('\x1f');
          if (sep != std::string::npos && f.compare(0, sep, pod) == 0) {
            cmd.push_back(f);
            ++matched;
          }
        }
        if (matched > 0) dels.push_back(std::move(cmd));
      }
      if (!dels.empty())
        pool_.with([&](RespConnection& c) { return c.pipeline(dels); });
    } while (cursor != "0");
  }

  IndexStats stats() const override {
    IndexStats s;
    s.admissions = admissions_.load(std::memory_order_relaxed);
    s.evictions = evictions_.load(std::memory_order_relaxed);
    s.lookups = lookups_.load(std::memory_order_relaxed);
    s.hits = hits_.load(std::memory_order_relaxed);
    return s;
  }

 private:
  // KEYS[1..n-1] = request-key hashes, KEYS[n] = the engine key.
  // Deletes each empty request hash and, only if ALL are empty, the engine
  // mapping — atomically (Redis runs scripts with no interleaved commands).
  static const std::string& prune_script() {
    static const std::string s =
        "local all_empty = 1\n"
        "for i = 1, #KEYS - 1 do\n"
        "  if redis.call('HLEN', KEYS[i]) == 0 then\n"
        "    redis.call('DEL', KEYS[i])\n"
        "  else\n"
        "    all_empty = 0\n"
        "  end\n"
        "end\n"
        "if all_empty == 1 then redis.call('DEL', KEYS[#KEYS]) end\n"
        "return all_empty\n";
    return s;
  }

  static std::string hex(uint64_t v) {
    char buf[17];
    snprintf(buf, sizeof(buf), "%016llx", static_cast<unsigned long long>(v));
    return buf;
  }
  std::string rkey(uint64_t k) const { return cfg_.key_prefix + ":r:" + hex(k); }
  std::string ekey(uint64_t k) const { return cfg_.key_prefix + ":e:" + hex(k); }

  std::string encode_entry(const PodEntry& e) {
    std::string out = strings_.get(e.pod);
    out += '\x1f';
    out += strings_.get(e.tier);
    out += '\x1f';
    out += std::to_string(static_cast<int>(e.flags));
    out += '\x1f';
    out += std::to_string(e.group);
    return out;
  }

  // Shared-backend fields can be written by other versions or edited by
  // operators: every numeric parse validates and reports failure instead of
  // throwing out of lookup()/evict() (which would feed the event pool's
  // process-terminating path). Undecodable entries are skipped.
  static bool parse_i64(const char* first, const char* last, int64_t* out,
                        int base = 10) {
    if (first == last) return false;
    auto res = std::from_chars(first, last, *out, base);
    return res.ec == std::errc() && res.ptr == last;
  }

  bool decode_entry(const std::string& s, PodEntry* e) {
    size_t a = s.find('\x1f');
    if (a == std::string::npos) return false;
    size_t b = s.find('\x1f', a + 1);
    if (b == std::string::npos) return false;
    size_t c = s.find('\x1f', b + 1);
    if (c == std::string::npos) return false;
    int64_t flags = 0, group = 0;
    const char* p = s.data();
    if (!parse_i64(p + b + 1, p + c, &flags) || flags < 0 || flags > 255)
      return false;
    if (!parse_i64(p + c + 1, p + s.size(), &group) || group < INT32_MIN ||
        group > INT32_MAX)
      return false;
    e->pod = strings_.intern(s.substr(0, a));
    e->tier = strings_.intern(s.substr(a + 1, b - a - 1));
    e->flags = static_cast<uint8_t>(flags);
    e->group = static_cast<int32_t>(group);
    return true;
  }

  static std::vector<uint64_t> parse_rk_list(const std::string& s) {
    std::vector<uint64_t> out;
    std::stringstream ss(s);
    std::string item;
    while (std::getline(ss, item, ',')) {
      if (item.empty()) continue;
      uint64_t u = 0;  // request keys are full 64-bit values: unsigned hex
      auto res = std::from_chars(item.data(), item.data() + item.size(), u, 16);
      if (res.ec == std::errc() && res.ptr == item.data() + item.size())
        out.push_back(u);
    }
    return out;
  }

  RedisIndexConfig cfg_;
  RespPool pool_;
  std::atomic<uint64_t> admissions_{0}, evictions_{0}, lookups_{0}, hits_{0};
};

}  // namespace kvc
