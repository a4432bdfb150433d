#!/usr/bin/env python3
"""Event-plane micro-benchmarks: decode rate and ZMTP loopback throughput.

Parity with the reference's adapter/ZMQ bench harnesses
(pkg/kvevents/engineadapter/vllm_adapter_bench_test.go,
zmq_subscriber_bench_test.go).

Run: python tools/bench_events.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.events.publisher import (
    block_stored_payload,
    encode_batch,
)

k = ensure_native()
MODEL = "m"


def bench_decode_and_apply():
    """Synchronous decode+index-apply rate (single thread)."""
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), ix)
    payloads = []
    for j in range(2000):
        payloads.append(encode_batch([
            block_stored_payload([j * 8 + x for x in range(8)], None,
                                 list(range(128)), 16)
        ]))
    t0 = time.perf_counter()
    for i, p in enumerate(payloads):
        pool.process(f"kv@pod-{i % 8}@{MODEL}", i, p)
    dt = time.perf_counter() - t0
    print(f"decode+apply (1 thread): {len(payloads) / dt:,.0f} batches/s "
          f"({len(payloads) * 8 / dt:,.0f} blocks/s)")


def bench_pool_parallel(workers=8):
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(concurrency=workers), ix)
    pool._pool.start()
    payloads = []
    for j in range(2000):
        payloads.append((f"kv@pod-{j % 32}@{MODEL}", encode_batch([
            block_stored_payload([j * 8 + x for x in range(8)], None,
                                 list(range(128)), 16)
        ])))
    t0 = time.perf_counter()
    for i, (t, p) in enumerate(payloads):
        pool.add_task(t, i, p)
    pool.drain()
    dt = time.perf_counter() - t0
    pool.shutdown()
    print(f"pool ({workers} workers):     {len(payloads) / dt:,.0f} batches/s "
          f"({len(payloads) * 8 / dt:,.0f} blocks/s)")


def bench_zmtp_throughput():
    """Loopback PUB -> SUB -> pool -> index, wire included."""
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(zmq_endpoint="tcp://127.0.0.1:0",
                                        concurrency=4), ix)
    pool.start()
    pub = k.Publisher(f"tcp://127.0.0.1:{pool.port}", bind=False)
    deadline = time.time() + 10
    while pub.peer_count < 1 and time.time() < deadline:
        time.sleep(0.01)
    time.sleep(0.2)
    n = 20000
    payload = encode_batch([
        block_stored_payload(list(range(8)), None, list(range(128)), 16)
    ])
    t0 = time.perf_counter()
    for i in range(n):
        pub.publish(f"kv@pod-{i % 8}@{MODEL}", i, payload)
    deadline = time.time() + 60
    while pool.stats().processed < n and time.time() < deadline:
        time.sleep(0.005)
    dt = time.perf_counter() - t0
    done = pool.stats().processed
    print(f"ZMTP loopback e2e:       {done / dt:,.0f} msgs/s "
          f"({done * len(payload) / dt / 1e6:.1f} MB/s wire)")
    pub.close()
    pool.shutdown()


if __name__ == "__main__":
    bench_decode_and_apply()
    bench_pool_parallel()
    bench_zmtp_throughput()
