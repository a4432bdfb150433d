// ThreadSanitizer stress for the native control plane (index + event pool
// + token processor), C++-only so TSan output is clean of Python noise.
//
// The reference relies on `go test -race` for its concurrency-heavy index
// (SURVEY.md §4/§5 notes the C++ side had no sanitizer coverage); this
// closes that gap for the new build.
//
// Build + run (tools/tsan_check.sh):
//   g++ -O1 -g -fsanitize=thread -std=c++17 -pthread tools/tsan_stress.cc -o tsan_stress
//   ./tsan_stress
#include <atomic>
#include <cstdio>
#include <thread>
#include <vector>

#include "../llm_d_kv_cache_amd/csrc/core/index.h"
#include "../llm_d_kv_cache_amd/csrc/core/indexer.h"
#include "../llm_d_kv_cache_amd/csrc/events/pool.h"

using namespace kvc;

int main() {
  auto tp = std::make_shared<TokenProcessor>(16, "");
  InMemoryIndexConfig cfg;
  cfg.shards = 8;
  cfg.size = 4096;  // small: force LRU evictions under contention
  auto index = std::make_shared<InMemoryIndex>(cfg);
  Indexer indexer(tp, index, {{"gpu", 1.0}, {"cpu", 0.8}});

  std::atomic<bool> stop{false};
  std::atomic<uint64_t> ops{0};
  std::vector<std::thread> threads;

  // adders: overlapping keys, engine bridges, shared pods
  for (int t = 0; t < 4; ++t) {
    threads.emplace_back([&, t] {
      PodEntry e;
      e.pod = index->strings().intern("pod-" + std::to_string(t % 2));
      e.tier = index->strings().intern(t % 2 ? "cpu" : "gpu");
      uint64_t i = 0;
      while (!stop) {
        uint64_t base = (i++ % 512) * 4;
        index->add({base + 1000}, {base, base + 1, base + 2, base + 3}, {e});
        ops++;
      }
    });
  }
  // evicters
  for (int t = 0; t < 2; ++t) {
    threads.emplace_back([&, t] {
      PodEntry e;
      e.pod = index->strings().intern("pod-" + std::to_string(t));
      e.tier = index->strings().intern(t ? "cpu" : "gpu");
      uint64_t i = 0;
      while (!stop) {
        index->evict((i++ % 512) * 4 + 1000, KeyType::kEngine, {e});
        ops++;
      }
    });
  }
  // lookups + scoring
  for (int t = 0; t < 4; ++t) {
    threads.emplace_back([&] {
      std::vector<uint32_t> tokens(64);
      for (size_t j = 0; j < tokens.size(); ++j) tokens[j] = j;
      while (!stop) {
        index->lookup({0, 1, 2, 3, 4}, {});
        indexer.score_tokens(tokens.data(), tokens.size(), "m", {});
        ops++;
      }
    });
  }
  // clear storms
  threads.emplace_back([&] {
    uint32_t p0 = index->strings().intern("pod-0");
    while (!stop) {
      index->clear(p0);
      std::this_thread::sleep_for(std::chrono::milliseconds(5));
      ops++;
    }
  });
  // byte-budget index with the TinyLFU admission sketch under contention
  InMemoryIndexConfig bcfg;
  bcfg.shards = 4;
  bcfg.max_bytes = 64 * 1024;
  auto budget_index = std::make_shared<InMemoryIndex>(bcfg);
  for (int t = 0; t < 3; ++t) {
    threads.emplace_back([&, t] {
      PodEntry e;
      e.pod = budget_index->strings().intern("bp-" + std::to_string(t));
      e.tier = budget_index->strings().intern("gpu");
      uint64_t i = 0;
      while (!stop) {
        uint64_t k = (t == 0) ? (i % 64) : (100000 + i);  // hot set + flood
        budget_index->add({}, {k}, {e});
        if (t == 0) budget_index->lookup({i % 64}, {});
        ++i;
        ops++;
      }
    });
  }

  // event pool with live workers processing synthetic batches
  EventPool pool(tp, index, 4);
  pool.start();
  threads.emplace_back([&] {
    // hand-built msgpack batch: [0.0, [["BlockStored",[h],nil,[t...],16]]]
    while (!stop) {
      for (int p = 0; p < 4; ++p) {
        std::string payload;
        uint8_t head[] = {0x92, 0xca, 0, 0, 0, 0, 0x91, 0x95};
        payload.assign(reinterpret_cast<char*>(head), sizeof(head));
        payload += '\xab';
        payload += "BlockStored";
        payload += '\x91';
        payload += '\x07';  // hashes [7]
        payload += '\xc0';  // parent nil
        payload += '\x90';  // tokens [] -> tier-update path
        payload += '\x10';  // block_size 16
        RawMessage msg;
        msg.topic = "kv@pod-" + std::to_string(p) + "@m";
        msg.payload = payload;
        pool.add_task(std::move(msg));
      }
      std::this_thread::sleep_for(std::chrono::microseconds(100));
    }
  });

  std::this_thread::sleep_for(std::chrono::seconds(8));
  stop = true;
  for (auto& t : threads) t.join();
  pool.shutdown();
  printf("tsan stress done: %llu ops, %llu events\n",
         static_cast<unsigned long long>(ops.load()),
         static_cast<unsigned long long>(pool.stats().processed));
  return 0;
}
