"""KVEvents ingestion: wire decode + pool semantics.

Payloads are built with the real msgpack library in the exact positional
layout engines emit (msgspec array_like=True), so these are wire-level
tests of the native single-pass decoder. Mirrors the reference test
strategy for pkg/kvevents/pool_test.go and engineadapter tests.
"""
import time

import msgpack
import pytest

import reference_impl as ref
from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.events.publisher import (
    all_blocks_cleared_payload,
    block_removed_payload,
    block_stored_payload,
    encode_batch,
)

k = ensure_native()

MODEL = "meta-llama/Llama-3.1-8B-Instruct"
POD = "vllm-pod-0"
TOPIC = f"kv@{POD}@{MODEL}"


@pytest.fixture
def setup():
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), ix)
    return ix, pool


def stored(tokens, hashes, parent=None, **kw):
    return block_stored_payload(hashes, parent, tokens, 16, **kw)


def test_block_stored_then_score(setup):
    ix, pool = setup
    tokens = list(range(32))
    pool.process(TOPIC, 0, encode_batch([stored(tokens, [101, 102])]))
    scores = ix.score_tokens(tokens, MODEL)
    assert scores == {POD: 2.0}
    # engine keys bridged to request keys
    rks = ix.compute_block_keys(tokens, MODEL)
    assert ix.index.get_request_key(101) == rks[0]
    assert ix.index.get_request_key(102) == rks[1]


def test_parent_chain_continuation(setup):
    ix, pool = setup
    tokens = list(range(64))
    pool.process(TOPIC, 0, encode_batch([stored(tokens[:32], [1, 2])]))
    pool.process(TOPIC, 1, encode_batch([stored(tokens[32:], [3, 4], parent=2)]))
    scores = ix.score_tokens(tokens, MODEL)
    assert scores == {POD: 4.0}


def test_unknown_parent_dropped(setup):
    ix, pool = setup
    pool.process(TOPIC, 0, encode_batch([stored(list(range(32)), [1, 2], parent=999)]))
    assert ix.score_tokens(list(range(32)), MODEL) == {}
    assert pool.stats().dropped_parent_misses == 1


def test_block_removed(setup):
    ix, pool = setup
    tokens = list(range(32))
    pool.process(TOPIC, 0, encode_batch([stored(tokens, [1, 2])]))
    pool.process(TOPIC, 1, encode_batch([block_removed_payload([2])]))
    scores = ix.score_tokens(tokens, MODEL)
    assert scores == {POD: 1.0}


def test_all_blocks_cleared(setup):
    ix, pool = setup
    tokens = list(range(32))
    pool.process(TOPIC, 0, encode_batch([stored(tokens, [1, 2])]))
    pool.process(TOPIC, 1, encode_batch([all_blocks_cleared_payload()]))
    assert ix.score_tokens(tokens, MODEL) == {}


def test_medium_maps_to_tier(setup):
    ix, pool = setup
    tokens = list(range(16))
    pool.process(TOPIC, 0, encode_batch([stored(tokens, [1], medium="CPU")]))
    keys = ix.compute_block_keys(tokens, MODEL)
    got = ix.index.lookup(keys)
    assert got[keys[0]][0].tier == "cpu"
    # scoring applies the cpu weight
    assert abs(ix.score_tokens(tokens, MODEL)[POD] - 0.8) < 1e-9


def test_lora_name_replaces_model(setup):
    ix, pool = setup
    tokens = list(range(16))
    pool.process(
        TOPIC, 0,
        encode_batch([stored(tokens, [1], lora_id=7, lora_name="my-lora")]),
    )
    assert ix.score_tokens(tokens, MODEL) == {}
    assert ix.score_tokens(tokens, "my-lora") == {POD: 1.0}


def test_token_less_cpu_offload_update(setup):
    ix, pool = setup
    tokens = list(range(32))
    pool.process(TOPIC, 0, encode_batch([stored(tokens, [1, 2])]))
    # offload event: no tokens, engine keys only, medium CPU
    pool.process(TOPIC, 1, encode_batch([stored([], [1, 2], medium="CPU")]))
    keys = ix.compute_block_keys(tokens, MODEL)
    got = ix.index.lookup(keys)
    tiers = {e.tier for kk in got for e in got[kk]}
    assert tiers == {"gpu", "cpu"}


def test_hashes_as_bytes(setup):
    ix, pool = setup
    tokens = list(range(16))
    h = (123456789).to_bytes(12, "big")  # >8 bytes: last 8 taken, BE
    pool.process(TOPIC, 0, encode_batch([stored(tokens, [h])]))
    assert ix.index.get_request_key(123456789) is not None


def test_extra_keys_taint(setup):
    ix, pool = setup
    tokens = list(range(32))
    pool.process(
        TOPIC, 0,
        encode_batch([stored(tokens, [1, 2], extra_keys=[["mm-1"], None])]),
    )
    plain = ix.score_tokens(tokens, MODEL)
    assert plain == {}  # plain hashes don't match tainted ones
    tainted = ix.score_tokens(tokens, MODEL, extra_features=[["mm-1"], None])
    assert tainted == {POD: 2.0}


def test_extra_keys_realign_1_to_many(setup):
    # engine block size 32 (1 engine key), canonical 16 (2 chunks):
    # engine extras replicate onto both canonical blocks.
    ix, pool = setup
    tokens = list(range(32))
    ev = block_stored_payload([11], None, tokens, 32, extra_keys=[["mm-A"]])
    pool.process(TOPIC, 0, encode_batch([ev]))
    got = ix.score_tokens(tokens, MODEL, extra_features=[["mm-A"], ["mm-A"]])
    assert got == {POD: 2.0}


def test_hma_group_metadata(setup):
    ix, pool = setup
    tokens = list(range(16))
    ev = block_stored_payload(
        [1], None, tokens, 16, group_idx=1, spec_kind="sliding_window",
        sliding_window=1024,
    )
    pool.process(TOPIC, 0, encode_batch([ev]))
    md = pool.group_metadata(POD, 1)
    assert md == {"kind": "sliding_window", "block_size": 16, "sliding_window": 1024}
    keys = ix.compute_block_keys(tokens, MODEL)
    got = ix.index.lookup(keys)
    assert got[keys[0]][0].group == 1


def test_malformed_payload_counted_not_fatal(setup):
    ix, pool = setup
    pool.process(TOPIC, 0, b"\xde\xad\xbe\xef")
    assert pool.stats().parse_failures == 1


def test_unknown_tag_skipped(setup):
    ix, pool = setup
    batch = msgpack.packb(
        [time.time(), [["FutureEvent", 1, 2], stored(list(range(16)), [1])]],
        use_bin_type=True,
    )
    pool.process(TOPIC, 0, batch)
    assert ix.score_tokens(list(range(16)), MODEL) == {POD: 1.0}


def test_trailing_fields_ignored(setup):
    ix, pool = setup
    ev = stored(list(range(16)), [1]) + [None, None, ["future", {"x": 1}]]
    pool.process(TOPIC, 0, encode_batch([ev]))
    assert ix.score_tokens(list(range(16)), MODEL) == {POD: 1.0}


def test_short_event_vllm_min_fields(setup):
    # Older engines omit all trailing defaults: 5 positional fields only.
    ix, pool = setup
    ev = ["BlockStored", [1], None, list(range(16)), 16]
    pool.process(TOPIC, 0, encode_batch([ev]))
    assert ix.score_tokens(list(range(16)), MODEL) == {POD: 1.0}


def test_dp_rank_batch_field(setup):
    ix, pool = setup
    batch = encode_batch([stored(list(range(16)), [1])], dp_rank=3)
    pool.process(TOPIC, 0, batch)
    assert ix.score_tokens(list(range(16)), MODEL) == {POD: 1.0}


def test_async_pool_ordered_processing(setup):
    ix, pool = setup
    pool._pool.start()
    try:
        tokens = list(range(64))
        # store then remove, 100x alternating: final state deterministic
        # because same pod -> same shard -> ordered.
        for i in range(100):
            pool.add_task(TOPIC, 2 * i, encode_batch([stored(tokens[:32], [1, 2])]))
            pool.add_task(TOPIC, 2 * i + 1, encode_batch([block_removed_payload([1, 2])]))
        pool.drain()
        assert ix.score_tokens(tokens, MODEL) == {}
        assert pool.stats().processed == 200
    finally:
        pool.shutdown()


def test_multi_pod_parallel_ingest():
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(concurrency=4), ix)
    pool._pool.start()
    try:
        tokens = list(range(32))
        for p in range(8):
            t = f"kv@pod-{p}@{MODEL}"
            pool.add_task(t, 0, encode_batch([stored(tokens, [p * 10, p * 10 + 1])]))
        pool.drain()
        scores = ix.score_tokens(tokens, MODEL)
        assert scores == {f"pod-{p}": 2.0 for p in range(8)}
    finally:
        pool.shutdown()


def test_sglang_layout_decodes():
    # SGLang: same positional layout, no HMA fields (9 fields max).
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(engine_type="sglang"), ix)
    ev = ["BlockStored", [5], None, list(range(16)), 16, None, None, None, None]
    pool.process(TOPIC, 0, encode_batch([ev]))
    assert ix.score_tokens(list(range(16)), MODEL) == {POD: 1.0}


def test_dp_rank_routing():
    """Opt-in DP-rank routing: each data-parallel rank becomes its own
    scoring target (beyond the reference, which decodes but drops the
    field)."""
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(dp_rank_routing=True), ix)
    tokens = list(range(32))
    pool.process(TOPIC, 0, encode_batch([stored(tokens, [1, 2])], dp_rank=0))
    pool.process(TOPIC, 1,
                 encode_batch([stored(tokens[:16], [3])], dp_rank=1))
    scores = ix.score_tokens(tokens, MODEL)
    assert scores == {f"{POD}-dp0": 2.0, f"{POD}-dp1": 1.0}
    # default stays merged per pod
    ix2 = KVCacheIndexer(IndexerConfig())
    pool2 = KVEventsPool(EventPoolConfig(), ix2)
    pool2.process(TOPIC, 0, encode_batch([stored(tokens, [1, 2])], dp_rank=3))
    assert ix2.score_tokens(tokens, MODEL) == {POD: 2.0}


def test_backpressure_bounded_queue():
    """max_queue_depth bounds memory under floods (oldest dropped; the
    index converges from later events)."""
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(concurrency=1, max_queue_depth=16), ix)
    # not started: messages pile up in the single shard, bounded at 16
    for i in range(100):
        pool.add_task(TOPIC, i, encode_batch([stored(list(range(16)), [i])]))
    s = pool.stats()
    assert s.dropped_backpressure == 84
    pool._pool.start()
    pool.drain()
    assert pool.stats().processed == 16
    pool.shutdown()


def test_hybrid_swa_eviction_keeps_full_attention_chain(setup):
    """HMA entry model: a hybrid (full-attention + sliding-window) pod
    that evicted old SWA-group blocks still scores full prefix credit
    through its full-attention entries — the per-(entry, group) model
    keeps the chain intact (the forgiving-absent-window-blocks refinement
    the reference lists as future work falls out of the design)."""
    ix, pool = setup
    tokens = list(range(64))  # 4 canonical blocks
    # group 0 = full attention, group 1 = sliding window 32 tokens
    pool.process(TOPIC, 0, encode_batch([
        block_stored_payload([1, 2, 3, 4], None, tokens, 16, group_idx=0,
                             spec_kind="full_attention"),
        block_stored_payload([11, 12, 13, 14], None, tokens, 16, group_idx=1,
                             spec_kind="sliding_window", sliding_window=32),
    ]))
    assert ix.score_tokens(tokens, MODEL) == {POD: 4.0}
    # the engine drops the SWA blocks outside the window (first two chunks)
    pool.process(TOPIC, 1, encode_batch([
        block_removed_payload([11, 12], group_idx=1),
    ]))
    scores = ix.score_tokens(tokens, MODEL)
    assert scores == {POD: 4.0}, \
        "full-attention entries must keep the prefix chain intact"
    # removing the full-attention blocks does break the chain
    pool.process(TOPIC, 2, encode_batch([
        block_removed_payload([2], group_idx=0),
    ]))
    assert ix.score_tokens(tokens, MODEL) == {POD: 1.0}


def test_clear_forgets_group_catalog():
    """AllBlocksCleared also drops the pod's learned group structure: a
    restarted pod may run a different model config, so stale window
    metadata must not shape scoring until new events re-teach it."""
    from llm_d_kv_cache_amd.events.publisher import (
        all_blocks_cleared_payload,
        block_stored_payload,
        encode_batch,
    )

    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), ix)
    pool.process("kv@pod-r@m", 0, encode_batch([
        block_stored_payload([1, 2], None, list(range(32)), 16, group_idx=0,
                             spec_kind="sliding_window", sliding_window=64)]))
    assert pool.native.sliding_window_tokens("pod-r") == 64
    assert pool.native.catalog_pods() == 1
    pool.process("kv@pod-r@m", 1, encode_batch([all_blocks_cleared_payload()]))
    assert pool.native.sliding_window_tokens("pod-r") == 0
    assert pool.native.catalog_pods() == 0
