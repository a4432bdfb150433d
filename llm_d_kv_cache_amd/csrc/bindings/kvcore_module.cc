// _kvcore: pybind11 bindings for the native control plane —
// TokenProcessor, InMemoryIndex, LongestPrefixScorer/Indexer, EventPool,
// ZMTP PUB/SUB. Heavy calls release the GIL; the event-ingest path never
// takes it (C++ subscriber -> C++ pool -> C++ index).
#include <pybind11/functional.h>
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "../core/indexer.h"
#include "../core/redis_index.h"
#include "../core/token_processor.h"
#include "../events/pool.h"
#include "../events/zmtp.h"

namespace py = pybind11;
using namespace kvc;

namespace {

// Python-facing pod entry: interned ids resolved back to strings.
struct PyPodEntry {
  std::string pod;
  std::string tier;
  bool speculative = false;
  std::optional<int32_t> group;
};

PodEntry to_native(IndexBackend& idx, const PyPodEntry& e) {
  PodEntry n;
  n.pod = idx.strings().intern(e.pod);
  n.tier = idx.strings().intern(e.tier);
  if (e.speculative) n.flags |= 1;
  if (e.group.has_value()) {
    n.flags |= 2;
    n.group = *e.group;
  }
  return n;
}

PyPodEntry from_native(IndexBackend& idx, const PodEntry& e) {
  PyPodEntry p;
  p.pod = idx.strings().get(e.pod);
  p.tier = idx.strings().get(e.tier);
  p.speculative = e.speculative();
  if (e.has_group()) p.group = e.group;
  return p;
}

std::vector<uint32_t> as_tokens(py::handle obj) {
  if (py::isinstance<py::array>(obj)) {
    auto arr = py::array_t<uint32_t, py::array::c_style | py::array::forcecast>::ensure(obj);
    std::vector<uint32_t> out(arr.size());
    std::memcpy(out.data(), arr.data(), arr.size() * 4);
    return out;
  }
  return obj.cast<std::vector<uint32_t>>();
}

std::vector<BlockExtra> as_extra(py::handle obj) {
  // None -> {}; else list of (None | list[str]).
  std::vector<BlockExtra> out;
  if (obj.is_none()) return out;
  for (py::handle item : obj) {
    if (item.is_none())
      out.emplace_back(std::nullopt);
    else
      out.emplace_back(item.cast<std::vector<std::string>>());
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_kvcore, m) {
  m.doc() = "llm-d-kv-cache-amd native control plane (CPU)";

  py::class_<PyPodEntry>(m, "PodEntry")
      .def(py::init([](std::string pod, std::string tier, bool speculative,
                       std::optional<int32_t> group) {
             PyPodEntry e;
             e.pod = std::move(pod);
             e.tier = std::move(tier);
             e.speculative = speculative;
             e.group = group;
             return e;
           }),
           py::arg("pod"), py::arg("tier") = "gpu", py::arg("speculative") = false,
           py::arg("group") = py::none())
      .def_readwrite("pod", &PyPodEntry::pod)
      .def_readwrite("tier", &PyPodEntry::tier)
      .def_readwrite("speculative", &PyPodEntry::speculative)
      .def_readwrite("group", &PyPodEntry::group)
      .def("__repr__",
           [](const PyPodEntry& e) {
             std::string s = e.pod + "@" + e.tier;
             if (e.speculative) s += "[speculative]";
             if (e.group.has_value()) s += "[group=" + std::to_string(*e.group) + "]";
             return s;
           })
      .def("__eq__", [](const PyPodEntry& a, const PyPodEntry& b) {
        return a.pod == b.pod && a.tier == b.tier && a.speculative == b.speculative &&
               a.group == b.group;
      })
      .def("__hash__", [](const PyPodEntry& e) {
        return py::hash(py::make_tuple(e.pod, e.tier, e.speculative,
                                       e.group.has_value() ? *e.group : -1));
      });

  py::class_<TokenProcessor, std::shared_ptr<TokenProcessor>>(m, "TokenProcessor")
      .def(py::init<int, std::string>(), py::arg("block_size") = 16,
           py::arg("hash_seed") = "")
      .def_property_readonly("block_size", &TokenProcessor::block_size)
      .def(
          "tokens_to_block_keys",
          [](const TokenProcessor& tp, py::handle tokens, const std::string& model,
             uint64_t parent, py::handle extra) {
            auto toks = as_tokens(tokens);
            auto ex = as_extra(extra);
            py::gil_scoped_release rel;
            return tp.tokens_to_block_keys(parent, toks.data(), toks.size(), model,
                                           ex.empty() ? nullptr : &ex);
          },
          py::arg("tokens"), py::arg("model"), py::arg("parent") = 0,
          py::arg("extra") = py::none());

  m.def(
      "hash_block",
      [](uint64_t parent, py::handle tokens, py::handle extra) {
        std::vector<uint32_t> toks;
        bool has_tokens = !tokens.is_none();
        if (has_tokens) toks = as_tokens(tokens);
        BlockExtra ex;
        if (!extra.is_none()) ex = extra.cast<std::vector<std::string>>();
        return TokenProcessor::hash_block(parent, has_tokens ? toks.data() : nullptr,
                                          toks.size(), &ex);
      },
      py::arg("parent"), py::arg("tokens"), py::arg("extra") = py::none(),
      "One FNV-64a(canonical-CBOR([parent, tokens, extra])) step");

  py::class_<IndexStats>(m, "IndexStats")
      .def_readonly("admissions", &IndexStats::admissions)
      .def_readonly("evictions", &IndexStats::evictions)
      .def_readonly("rejections", &IndexStats::rejections)
      .def_readonly("lookups", &IndexStats::lookups)
      .def_readonly("hits", &IndexStats::hits)
      .def_readonly("keys", &IndexStats::keys);

  py::class_<IndexBackend, std::shared_ptr<IndexBackend>>(m, "IndexBackend")
      .def(
          "lookup",
          [](IndexBackend& idx, const std::vector<uint64_t>& keys,
             const std::vector<std::string>& pods) {
            // Intern (not find): shared backends hold entries written by
            // other processes whose pod names this one never saw.
            std::unordered_set<uint32_t> filter;
            for (const auto& p : pods) filter.insert(idx.strings().intern(p));
            std::vector<std::pair<uint64_t, std::vector<PodEntry>>> hits;
            {
              py::gil_scoped_release rel;
              hits = idx.lookup(keys, filter);
            }
            py::dict out;
            for (auto& [k, entries] : hits) {
              py::list lst;
              for (const auto& e : entries) lst.append(from_native(idx, e));
              out[py::cast(k)] = lst;
            }
            return out;
          },
          py::arg("keys"), py::arg("pods") = std::vector<std::string>{})
      .def(
          "add",
          [](IndexBackend& idx, const std::vector<uint64_t>& engine_keys,
             const std::vector<uint64_t>& request_keys,
             const std::vector<PyPodEntry>& entries) {
            std::vector<PodEntry> native;
            native.reserve(entries.size());
            for (const auto& e : entries) native.push_back(to_native(idx, e));
            py::gil_scoped_release rel;
            idx.add(engine_keys, request_keys, native);
          },
          py::arg("engine_keys"), py::arg("request_keys"), py::arg("entries"))
      .def(
          "evict",
          [](IndexBackend& idx, uint64_t key, const std::string& key_type,
             const std::vector<PyPodEntry>& entries) {
            KeyType t;
            if (key_type == "engine")
              t = KeyType::kEngine;
            else if (key_type == "request")
              t = KeyType::kRequest;
            else
              throw std::invalid_argument("key_type must be 'engine' or 'request'");
            std::vector<PodEntry> native;
            for (const auto& e : entries) native.push_back(to_native(idx, e));
            py::gil_scoped_release rel;
            idx.evict(key, t, native);
          },
          py::arg("key"), py::arg("key_type"), py::arg("entries"))
      .def("get_request_key",
           [](IndexBackend& idx, uint64_t ek) -> py::object {
             uint64_t rk;
             bool found;
             {
               py::gil_scoped_release rel;
               found = idx.get_request_key(ek, &rk);
             }
             if (found) return py::cast(rk);
             return py::none();
           })
      .def("clear",
           [](IndexBackend& idx, const std::string& pod) {
             uint32_t id = idx.strings().intern(pod);
             py::gil_scoped_release rel;
             idx.clear(id);
           })
      .def("stats", &IndexBackend::stats,
           py::call_guard<py::gil_scoped_release>());

  py::class_<InMemoryIndex, IndexBackend, std::shared_ptr<InMemoryIndex>>(
      m, "InMemoryIndex")
      .def(py::init([](size_t size, size_t pods_per_key, size_t shards,
                       size_t max_bytes) {
             InMemoryIndexConfig cfg;
             cfg.size = size;
             cfg.pods_per_key = pods_per_key;
             cfg.shards = shards;
             cfg.max_bytes = max_bytes;
             return std::make_shared<InMemoryIndex>(cfg);
           }),
           py::arg("size") = 100000000, py::arg("pods_per_key") = 10,
           py::arg("shards") = 64, py::arg("max_bytes") = 0)
      .def("save", &InMemoryIndex::save, py::arg("path"),
           py::call_guard<py::gil_scoped_release>())
      .def("load", &InMemoryIndex::load, py::arg("path"),
           py::call_guard<py::gil_scoped_release>());

  py::class_<RedisIndex, IndexBackend, std::shared_ptr<RedisIndex>>(
      m, "RedisIndex")
      .def(py::init([](const std::string& host, int port, size_t pool_size,
                       const std::string& key_prefix) {
             RedisIndexConfig cfg;
             cfg.host = host;
             cfg.port = port;
             cfg.pool_size = pool_size;
             cfg.key_prefix = key_prefix;
             py::gil_scoped_release rel;
             return std::make_shared<RedisIndex>(cfg);
           }),
           py::arg("host") = "127.0.0.1", py::arg("port") = 6379,
           py::arg("pool_size") = 4, py::arg("key_prefix") = "kv");

  py::class_<ScoreResult>(m, "ScoreResult")
      .def_readonly("scores", &ScoreResult::scores)
      .def_readonly("total_blocks", &ScoreResult::total_blocks)
      .def_readonly("hit_blocks", &ScoreResult::hit_blocks);

  py::class_<Indexer>(m, "Indexer")
      .def(py::init<std::shared_ptr<TokenProcessor>, std::shared_ptr<IndexBackend>,
                    std::unordered_map<std::string, double>>(),
           py::arg("token_processor"), py::arg("index"), py::arg("tier_weights"))
      .def(
          "score_tokens",
          [](Indexer& ix, py::handle tokens, const std::string& model,
             const std::vector<std::string>& pods, py::handle extra,
             std::unordered_map<std::string,
                                LongestPrefixScorer::GroupWindows>
                 window_hints) {
            auto toks = as_tokens(tokens);
            auto ex = as_extra(extra);
            py::gil_scoped_release rel;
            return ix.score_tokens(toks.data(), toks.size(), model, pods,
                                   ex.empty() ? nullptr : &ex,
                                   window_hints.empty() ? nullptr : &window_hints);
          },
          py::arg("tokens"), py::arg("model"),
          py::arg("pods") = std::vector<std::string>{}, py::arg("extra") = py::none(),
          py::arg("window_hints") =
              std::unordered_map<std::string,
                                 LongestPrefixScorer::GroupWindows>{})
      .def(
          "compute_block_keys",
          [](Indexer& ix, py::handle tokens, const std::string& model) {
            auto toks = as_tokens(tokens);
            py::gil_scoped_release rel;
            return ix.compute_block_keys(toks.data(), toks.size(), model);
          },
          py::arg("tokens"), py::arg("model"));

  py::class_<PoolStats>(m, "PoolStats")
      .def_readonly("enqueued", &PoolStats::enqueued)
      .def_readonly("processed", &PoolStats::processed)
      .def_readonly("parse_failures", &PoolStats::parse_failures)
      .def_readonly("handler_failures", &PoolStats::handler_failures)
      .def_readonly("dropped_parent_misses", &PoolStats::dropped_parent_misses)
      .def_readonly("dropped_backpressure", &PoolStats::dropped_backpressure);

  py::class_<EventPool, std::shared_ptr<EventPool>>(m, "EventPool")
      .def(py::init<std::shared_ptr<TokenProcessor>, std::shared_ptr<IndexBackend>,
                    size_t, bool, size_t>(),
           py::arg("token_processor"), py::arg("index"), py::arg("concurrency") = 4,
           py::arg("dp_rank_routing") = false, py::arg("max_queue_depth") = 0)
      .def("start", &EventPool::start, py::call_guard<py::gil_scoped_release>())
      .def("shutdown", &EventPool::shutdown, py::call_guard<py::gil_scoped_release>())
      .def(
          "add_task",
          [](EventPool& p, const std::string& topic, uint64_t seq, py::bytes payload) {
            RawMessage msg;
            msg.topic = topic;
            msg.seq = seq;
            msg.payload = payload.cast<std::string>();
            py::gil_scoped_release rel;
            p.add_task(std::move(msg));
          },
          py::arg("topic"), py::arg("seq"), py::arg("payload"))
      .def(
          "process",
          [](EventPool& p, const std::string& topic, uint64_t seq, py::bytes payload) {
            RawMessage msg;
            msg.topic = topic;
            msg.seq = seq;
            msg.payload = payload.cast<std::string>();
            py::gil_scoped_release rel;
            p.process(msg);
          },
          py::arg("topic"), py::arg("seq"), py::arg("payload"))
      .def("drain", &EventPool::drain, py::call_guard<py::gil_scoped_release>())
      .def("stats", &EventPool::stats)
      .def("group_metadata", [](EventPool& p, const std::string& pod, int32_t group)
               -> py::object {
        auto md = p.group_catalog().get(pod, group);
        if (!md.has_value()) return py::none();
        py::dict d;
        d["kind"] = md->kind;
        d["block_size"] = md->block_size;
        d["sliding_window"] =
            md->sliding_window.has_value() ? py::cast(*md->sliding_window) : py::none();
        return d;
      })
      .def("sliding_window_tokens",
           [](EventPool& p, const std::string& pod) {
             return p.group_catalog().sliding_window_tokens(pod);
           },
           py::arg("pod"))
      .def("group_windows",
           [](EventPool& p, const std::string& pod) {
             return p.group_catalog().group_windows(pod);
           },
           py::arg("pod"))
      .def("catalog_version",
           [](EventPool& p) { return p.group_catalog().version(); })
      .def("catalog_snapshot",
           [](EventPool& p) { return p.group_catalog().snapshot(); })
      .def("catalog_pods",
           [](EventPool& p) { return p.group_catalog().pod_count(); });

  py::class_<ZmtpPublisher>(m, "Publisher")
      .def(py::init<const std::string&, bool, std::string, std::string>(),
           py::arg("endpoint"), py::arg("bind") = true,
           py::arg("username") = "", py::arg("password") = "")
      .def_property_readonly("port", &ZmtpPublisher::port)
      .def_property_readonly("peer_count", &ZmtpPublisher::peer_count)
      .def(
          "publish",
          [](ZmtpPublisher& p, const std::string& topic, uint64_t seq,
             py::bytes payload) {
            std::string data = payload.cast<std::string>();
            py::gil_scoped_release rel;
            p.publish(topic, seq, data);
          },
          py::arg("topic"), py::arg("seq"), py::arg("payload"))
      .def("close", &ZmtpPublisher::close, py::call_guard<py::gil_scoped_release>());

  py::class_<ZmtpSubscriber>(m, "Subscriber")
      .def(py::init([](const std::string& endpoint, const std::string& topic_filter,
                       std::shared_ptr<EventPool> pool, py::object callback,
                       bool bind, int reconnect_ms,
                       const std::string& username,
                       const std::string& password) {
             ZmtpSubscriber::Handler h;
             if (pool) {
               // Native fast path: deliver straight into the pool, no GIL.
               h = [pool](std::string topic, uint64_t seq, std::string payload) {
                 RawMessage msg;
                 msg.topic = std::move(topic);
                 msg.seq = seq;
                 msg.payload = std::move(payload);
                 pool->add_task(std::move(msg));
               };
             } else if (!callback.is_none()) {
               auto cb = std::make_shared<py::object>(callback);
               h = [cb](std::string topic, uint64_t seq, std::string payload) {
                 py::gil_scoped_acquire acq;
                 (*cb)(topic, seq, py::bytes(payload));
               };
             }
             return std::make_unique<ZmtpSubscriber>(endpoint, topic_filter, std::move(h),
                                                     bind, reconnect_ms,
                                                     username, password);
           }),
           py::arg("endpoint"), py::arg("topic_filter") = "",
           py::arg("pool") = nullptr, py::arg("callback") = py::none(),
           py::arg("bind") = false, py::arg("reconnect_ms") = 5000,
           py::arg("username") = "", py::arg("password") = "")
      .def_property_readonly("port", &ZmtpSubscriber::port)
      .def("close", &ZmtpSubscriber::close, py::call_guard<py::gil_scoped_release>());
}
