#!/usr/bin/env python3
"""Offline demo: in-memory kvblock index + synthetic KVEvents -> Score().

Parity with the reference examples/kv_cache_index/main.go: wire the
indexer, ingest a few synthetic BlockStored batches, score a prompt's
tokens against the fleet.

Run: python examples/kv_cache_index.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.events.publisher import (
    block_stored_payload,
    encode_batch,
)

MODEL = "meta-llama/Llama-3.1-8B-Instruct"


def main():
    indexer = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), indexer)

    tokens = list(range(64))  # 4 blocks of 16
    # pod-a cached the whole prefix on GPU; pod-b cached half, offloaded to CPU
    pool.process(f"kv@pod-a@{MODEL}", 0, encode_batch([
        block_stored_payload([101, 102, 103, 104], None, tokens, 16),
    ]))
    pool.process(f"kv@pod-b@{MODEL}", 0, encode_batch([
        block_stored_payload([201, 202], None, tokens[:32], 16, medium="CPU"),
    ]))

    scores = indexer.score_tokens(tokens, MODEL)
    print("pod scores:", scores)
    assert scores["pod-a"] == 4.0
    assert abs(scores["pod-b"] - 1.6) < 1e-9  # 2 blocks x cpu weight 0.8
    print("stats:", indexer.stats().keys, "keys resident")


if __name__ == "__main__":
    main()
