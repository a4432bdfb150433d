#!/usr/bin/env python3
"""Routing-value simulation: the reference's headline benchmark shape
replayed against this framework's REAL control plane.

The reference's published numbers (benchmarking/37-capacity)
measure one thing: how much TTFT/throughput a fleet gains when the
endpoint picker routes by precise KV-cache knowledge instead of load or
random. That takes a vLLM fleet; this tool reproduces the experiment's
structure as a discrete-event simulation in which the *control plane is
not simulated*: every pod's cache admissions/evictions are published as
real msgpack BlockStored/BlockRemoved events through the native ingestion
pool, and the "precise" scheduler calls the native score_tokens path
(tokens -> chained hashes -> index lookup -> prefix scoring) per request.

Workload (reference 37-capacity/README.md shape): G shared-prefix groups,
each group shares a long system prompt; every request adds a unique
question and produces output tokens. Constant request rate, multiple
schedulers compared on the same trace.

Usage: python tools/routing_sim.py [--qps 8] [--pods 4] [--groups 24]
"""
import argparse
import heapq
import json
import os
import random
import sys
from collections import OrderedDict

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.events.publisher import (
    block_removed_payload,
    block_stored_payload,
    encode_batch,
)

MODEL = "sim-llama-70b"
BLOCK = 16


class Pod:
    """One serving pod: bounded concurrency, FIFO queue, LRU block cache.

    The cache is ground truth; the index only learns about it through the
    event wire (with the same batching an engine would do), so the precise
    scheduler can be stale/wrong exactly the way it can be in production.
    """

    def __init__(self, name, pool, capacity_blocks, concurrency):
        self.name = name
        self.pool = pool
        self.capacity = capacity_blocks
        self.concurrency = concurrency
        self.cache = OrderedDict()  # engine_hash -> True (LRU)
        self.active = 0
        self.queue = []
        self.seq = 0

    def _publish(self, events):
        self.pool.process(f"kv@{self.name}@{MODEL}", self.seq,
                          encode_batch(events))
        self.seq += 1

    def cached_prefix_tokens(self, keys):
        n = 0
        for k in keys:
            if k in self.cache:
                self.cache.move_to_end(k)
                n += BLOCK
            else:
                break
        return n

    def admit(self, keys, tokens):
        """Insert the request's blocks; publish stores + evictions."""
        new = [k for k in keys if k not in self.cache]
        for k in keys:
            self.cache[k] = True
            self.cache.move_to_end(k)
        evicted = []
        while len(self.cache) > self.capacity:
            old, _ = self.cache.popitem(last=False)
            evicted.append(old)
        events = []
        if new:
            # one BlockStored for the whole chain (engine-granularity batch)
            events.append(block_stored_payload(
                keys, None, tokens[:len(keys) * BLOCK], BLOCK))
        if evicted:
            events.append(block_removed_payload(evicted))
        if events:
            self._publish(events)


class Sim:
    def __init__(self, args, seed=0):
        self.rng = random.Random(seed)
        self.args = args
        self.indexer = KVCacheIndexer(IndexerConfig())
        self.pool = KVEventsPool(EventPoolConfig(), self.indexer)
        self.pods = [Pod(f"pod-{i}", self.pool, args.capacity_blocks,
                         args.concurrency) for i in range(args.pods)]
        # shared-prefix groups: unique token streams per group
        self.prefixes = {
            g: [100_000_000 + g * 100_000 + t for t in range(args.prefix_tokens)]
            for g in range(args.groups)
        }

    def _request_tokens(self, group, rid):
        q = [1_000_000_000 + rid * 10_000 + t
             for t in range(self.args.question_tokens)]
        return self.prefixes[group] + q

    def _service_time(self, pod, tokens, keys):
        cached = pod.cached_prefix_tokens(keys)
        uncached = len(tokens) - cached
        prefill = 0.02 + uncached / self.args.prefill_tok_s
        decode = self.args.output_tokens * self.args.itl_s
        return prefill, decode, cached

    def run(self, scheduler):
        a = self.args
        # reset pods + index between schedulers (same trace via same seed)
        for p in self.pods:
            p.cache.clear()
            p.active = 0
            p.queue = []
            self.pool.process(f"kv@{p.name}@{MODEL}", p.seq,
                              encode_batch([["AllBlocksCleared"]]))
            p.seq += 1
        rng = random.Random(a.seed)
        events = []  # (time, kind, payload)
        t = 0.0
        reqs = []
        for rid in range(a.requests):
            t += rng.expovariate(a.qps)
            group = rng.randrange(a.groups)
            reqs.append((t, rid, group))
            heapq.heappush(events, (t, 0, ("arrive", rid, group)))
        ttfts = []
        cached_fracs = []
        done_t = 0.0
        seqc = [0]
        avg_service = [2.0]  # EMA, seeds the backlog penalty

        def pick(tokens, keys):
            if scheduler == "random":
                return self.rng.choice(self.pods)
            if scheduler == "load":
                return min(self.pods,
                           key=lambda p: (p.active + len(p.queue),
                                          self.rng.random()))
            # precise: the real native scoring path, blended with a
            # backlog penalty the way EPP composes scorer plugins
            # (prefix-cache scorer + queue scorer)
            scores = self.indexer.score_tokens(
                tokens, MODEL, [p.name for p in self.pods])
            best, best_cost = None, None
            for p in self.pods:
                cached = min(scores.get(p.name, 0.0) * BLOCK, len(tokens))
                prefill = 0.02 + (len(tokens) - cached) / a.prefill_tok_s
                backlog = max(0, p.active + len(p.queue)
                              - p.concurrency + 1)
                cost = (prefill + backlog * avg_service[0] / p.concurrency,
                        self.rng.random())
                if best is None or cost < best_cost:
                    best, best_cost = p, cost
            return best

        def start(now, pod, rid, group, arrival):
            tokens = self._request_tokens(group, rid)
            keys = self.indexer.compute_block_keys(tokens, MODEL)
            prefill, decode, cached = self._service_time(pod, tokens, keys)
            pod.active += 1
            avg_service[0] += 0.05 * (prefill + decode - avg_service[0])
            ttfts.append(now + prefill - arrival)
            cached_fracs.append(cached / len(tokens))
            pod.admit(keys, tokens)
            heapq.heappush(events, (now + prefill + decode, seqc[0],
                                    ("finish", pod, rid)))
            seqc[0] += 1

        while events:
            now, _, ev = heapq.heappop(events)
            if ev[0] == "arrive":
                _, rid, group = ev
                tokens = self._request_tokens(group, rid)
                keys = self.indexer.compute_block_keys(tokens, MODEL)
                self.pool.drain()  # index has seen all published events
                pod = pick(tokens, keys)
                if pod.active < pod.concurrency:
                    start(now, pod, rid, group, now)
                else:
                    pod.queue.append((now, rid, group))
            else:
                _, pod, rid = ev
                pod.active -= 1
                done_t = max(done_t, now)
                if pod.queue and pod.active < pod.concurrency:
                    arrival, qrid, qgroup = pod.queue.pop(0)
                    start(now, pod, qrid, qgroup, arrival)
        ttfts.sort()
        n = len(ttfts)
        return {
            "scheduler": scheduler,
            "ttft_mean_s": round(sum(ttfts) / n, 3),
            "ttft_p90_s": round(ttfts[int(0.9 * n)], 3),
            "ttft_max_s": round(ttfts[-1], 3),
            "cached_prefix_frac": round(sum(cached_fracs) / n, 3),
            "throughput_req_s": round(n / done_t, 2) if done_t else None,
        }


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--pods", type=int, default=4)
    ap.add_argument("--groups", type=int, default=24)
    ap.add_argument("--requests", type=int, default=600)
    ap.add_argument("--qps", type=float, default=8.0)
    ap.add_argument("--prefix-tokens", type=int, default=4096)
    ap.add_argument("--question-tokens", type=int, default=256)
    ap.add_argument("--output-tokens", type=int, default=50)
    ap.add_argument("--prefill-tok-s", type=float, default=3000.0,
                    help="uncached prefill token rate per pod slot")
    ap.add_argument("--itl-s", type=float, default=0.02)
    ap.add_argument("--concurrency", type=int, default=4)
    ap.add_argument("--capacity-blocks", type=int, default=2048,
                    help="per-pod KV blocks; the fleet cannot hold every "
                         "group everywhere, so placement matters")
    ap.add_argument("--seed", type=int, default=1)
    args = ap.parse_args(argv)

    sim = Sim(args, seed=args.seed)
    out = {"config": vars(args),
           "results": [sim.run(s) for s in ("precise", "load", "random")]}
    print(json.dumps(out, indent=1))
    return out


if __name__ == "__main__":
    main()
