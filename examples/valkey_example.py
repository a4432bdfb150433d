#!/usr/bin/env python3
"""Valkey/Redis-backed shared index (parity with the reference
examples/valkey_example): several indexer replicas share one network
index, so each sees every replica's ingested events.

Run: python examples/valkey_example.py [--host 127.0.0.1 --port 6379]
(without a live server the demo starts the embedded fake from tests/)
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from llm_d_kv_cache_amd.core import (
    IndexerConfig,
    KVCacheIndexer,
    RedisIndexConfig,
)
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.events.publisher import (
    block_stored_payload,
    encode_batch,
)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default=None)
    ap.add_argument("--port", type=int, default=6379)
    args = ap.parse_args()

    fake = None
    host, port = args.host, args.port
    if host is None:
        sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))
        from fake_redis import FakeRedis

        fake = FakeRedis()
        host, port = "127.0.0.1", fake.port
        print(f"no --host given: using embedded fake valkey on :{port}")

    cfg = IndexerConfig(redis_index=RedisIndexConfig(host=host, port=port))
    replica_a = KVCacheIndexer(cfg)
    replica_b = KVCacheIndexer(cfg)

    # replica A ingests events; replica B scores against the shared state
    pool_a = KVEventsPool(EventPoolConfig(), replica_a)
    tokens = list(range(48))
    pool_a.process("kv@pod-v@m", 0, encode_batch([
        block_stored_payload([1, 2, 3], None, tokens, 16)
    ]))
    print("replica B scores:", replica_b.score_tokens(tokens, "m"))
    assert replica_b.score_tokens(tokens, "m") == {"pod-v": 3.0}
    if fake:
        fake.close()
    print("ok")


if __name__ == "__main__":
    main()
