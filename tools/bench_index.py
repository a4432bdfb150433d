#!/usr/bin/env python3
"""Index backend micro-benchmark: Add/Lookup rates per backend.

Parity with the reference tests/profiling/kv_cache_index harness
(InMemory vs CostAware vs Redis). The Redis arm runs against the embedded
fake server unless --redis host:port is given.

Run: python tools/bench_index.py [--keys 100000] [--redis host:port]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tests"))

from llm_d_kv_cache_amd import ensure_native

k = ensure_native()


def bench_backend(name, idx, n_keys, batch=64):
    entries = [k.PodEntry(f"pod-{i}", "gpu") for i in range(4)]
    keys = list(range(1, n_keys + 1))
    t0 = time.perf_counter()
    for i in range(0, n_keys, batch):
        idx.add([], keys[i:i + batch], entries)
    add_dt = time.perf_counter() - t0
    t0 = time.perf_counter()
    n_lookups = max(1, 20000 // batch)
    for i in range(n_lookups):
        base = (i * batch) % max(1, n_keys - batch)
        idx.lookup(keys[base:base + batch])
    lk_dt = time.perf_counter() - t0
    print(f"{name:<22} add: {n_keys / add_dt:>12,.0f} keys/s   "
          f"lookup: {n_lookups * batch / lk_dt:>12,.0f} keys/s "
          f"({n_lookups / lk_dt:,.0f} batch-lookups/s)")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--keys", type=int, default=100_000)
    ap.add_argument("--redis", default=None, help="host:port (else embedded fake)")
    args = ap.parse_args()

    bench_backend("InMemoryIndex", k.InMemoryIndex(), args.keys)
    bench_backend("CostAware (256MiB)",
                  k.InMemoryIndex(max_bytes=256 * 2**20), args.keys)
    if args.redis:
        host, port = args.redis.rsplit(":", 1)
        bench_backend("RedisIndex", k.RedisIndex(host=host, port=int(port)),
                      min(args.keys, 20_000))
    else:
        from fake_redis import FakeRedis

        srv = FakeRedis()
        bench_backend("RedisIndex (fake)",
                      k.RedisIndex(host="127.0.0.1", port=srv.port),
                      min(args.keys, 20_000))
        srv.close()


if __name__ == "__main__":
    main()
