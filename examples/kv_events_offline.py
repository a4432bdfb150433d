#!/usr/bin/env python3
"""Offline KVEvents demo (parity with the reference examples/kv_events/
offline): no sockets — wire-format msgpack batches are fed straight into
the ingestion pool, then the index is scored. Shows the whole write path
(topic parse -> positional decode -> hash-chain -> index) and read path
(tokens -> keys -> lookup -> prefix scoring) in one process.

Run: python examples/kv_events_offline.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.events.publisher import (
    all_blocks_cleared_payload,
    block_removed_payload,
    block_stored_payload,
    encode_batch,
)


def main():
    indexer = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), indexer)

    tokens = list(range(64))  # 4 blocks of 16
    # two pods cache the same prefix; pod-b also keeps going
    pool.process("kv@pod-a@demo-model", 0, encode_batch([
        block_stored_payload([1, 2], None, tokens[:32], 16),
    ]))
    pool.process("kv@pod-b@demo-model", 0, encode_batch([
        block_stored_payload([11, 12, 13, 14], None, tokens, 16),
    ]))
    scores = indexer.score_tokens(tokens, "demo-model")
    print("after stores:", scores)
    assert scores == {"pod-a": 2.0, "pod-b": 4.0}

    # pod-b's engine evicts its tail block
    pool.process("kv@pod-b@demo-model", 1, encode_batch([
        block_removed_payload([14]),
    ]))
    scores = indexer.score_tokens(tokens, "demo-model")
    print("after removal:", scores)
    assert scores == {"pod-a": 2.0, "pod-b": 3.0}

    # pod-a restarts
    pool.process("kv@pod-a@demo-model", 1, encode_batch([
        all_blocks_cleared_payload(),
    ]))
    scores = indexer.score_tokens(tokens, "demo-model")
    print("after clear:", scores)
    assert scores == {"pod-b": 3.0}
    print("ok")


if __name__ == "__main__":
    main()
