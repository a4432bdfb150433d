"""Longest-prefix scorer + Indexer orchestrator (read path)."""
from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import (
    IndexerConfig,
    KVCacheIndexer,
    TokenProcessorConfig,
)

k = ensure_native()

MODEL = "m"


def make_indexer():
    return KVCacheIndexer(IndexerConfig(token_processor=TokenProcessorConfig(16, "")))


def add_prefix(ix, pod, tokens, tier="gpu"):
    keys = ix.compute_block_keys(tokens, MODEL)
    ix.index.add([], keys, [k.PodEntry(pod, tier)])
    return keys


def test_longest_prefix_scoring():
    ix = make_indexer()
    tokens = list(range(64))  # 4 blocks
    add_prefix(ix, "pod-full", tokens)
    add_prefix(ix, "pod-half", tokens[:32])
    scores = ix.score_tokens(tokens, MODEL)
    assert scores["pod-full"] == 4.0
    assert scores["pod-half"] == 2.0


def test_chain_break_stops_scoring():
    ix = make_indexer()
    tokens = list(range(64))
    keys = ix.compute_block_keys(tokens, MODEL)
    # pod holds blocks 0 and 2 but not 1: only block 0 counts
    ix.index.add([], [keys[0], keys[2]], [k.PodEntry("pod-gap", "gpu")])
    scores = ix.score_tokens(tokens, MODEL)
    assert scores == {"pod-gap": 1.0}


def test_tier_weights():
    ix = make_indexer()  # default weights: gpu 1.0, cpu 0.8, peer-gpu 0.95
    tokens = list(range(32))
    add_prefix(ix, "pod-gpu", tokens, tier="gpu")
    add_prefix(ix, "pod-cpu", tokens, tier="cpu")
    add_prefix(ix, "pod-peer", tokens, tier="peer-gpu")
    scores = ix.score_tokens(tokens, MODEL)
    assert scores["pod-gpu"] == 2.0
    assert abs(scores["pod-cpu"] - 1.6) < 1e-9
    assert abs(scores["pod-peer"] - 1.9) < 1e-9


def test_max_weight_across_tiers_per_block():
    ix = make_indexer()
    tokens = list(range(16))
    add_prefix(ix, "pod-a", tokens, tier="cpu")
    add_prefix(ix, "pod-a", tokens, tier="gpu")
    scores = ix.score_tokens(tokens, MODEL)
    assert scores["pod-a"] == 1.0  # max(gpu, cpu), not sum


def test_pod_filter():
    ix = make_indexer()
    tokens = list(range(16))
    add_prefix(ix, "pod-a", tokens)
    add_prefix(ix, "pod-b", tokens)
    scores = ix.score_tokens(tokens, MODEL, pod_identifiers=["pod-b"])
    assert set(scores) == {"pod-b"}
    # filter naming only unknown pods -> empty scores
    assert ix.score_tokens(tokens, MODEL, pod_identifiers=["nope"]) == {}


def test_no_blocks_no_scores():
    ix = make_indexer()
    assert ix.score_tokens(list(range(8)), MODEL) == {}  # < 1 block
    assert ix.score_tokens(list(range(16)), MODEL) == {}  # no index entries


def test_detailed_hit_ratio():
    ix = make_indexer()
    tokens = list(range(64))
    add_prefix(ix, "pod-a", tokens[:32])
    scores, total, hits = ix.score_tokens_detailed(tokens, MODEL)
    assert total == 4
    assert hits == 2
    assert scores["pod-a"] == 2.0


def test_unknown_tier_defaults_to_weight_1():
    ix = make_indexer()
    tokens = list(range(16))
    add_prefix(ix, "pod-x", tokens, tier="exotic-tier")
    assert ix.score_tokens(tokens, MODEL)["pod-x"] == 1.0


# ---- multimodal read side ---------------------------------------------------

def test_compute_block_extra_features():
    from llm_d_kv_cache_amd.core.extra_keys import (
        PlaceholderRange,
        compute_block_extra_features,
    )

    # 64 tokens, 16-token blocks; image A covers tokens [10, 30),
    # image B covers [30, 34)
    out = compute_block_extra_features(
        64, 16, ["img-A", "img-B"],
        [PlaceholderRange(10, 20), PlaceholderRange(30, 4)],
    )
    assert out == [["img-A"], ["img-A", "img-B"], ["img-B"], None]
    # pure text
    assert compute_block_extra_features(64, 16, [], []) is None
    # partial tail dropped: only full blocks get entries
    out = compute_block_extra_features(40, 16, ["x"], [PlaceholderRange(38, 2)])
    assert out == [None, None]


def test_mm_features_read_write_agree():
    """The read-side features reproduce the write-side tainted hashes: a
    scorer query with mm features matches blocks stored with engine
    extra_keys."""
    from llm_d_kv_cache_amd.core.extra_keys import (
        PlaceholderRange,
        compute_block_extra_features,
    )
    from llm_d_kv_cache_amd.events.publisher import (
        block_stored_payload,
        encode_batch,
    )
    from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool

    ix = make_indexer()
    pool = KVEventsPool(EventPoolConfig(), ix)
    tokens = list(range(48))
    extra = compute_block_extra_features(
        48, 16, ["mm-7"], [PlaceholderRange(20, 10)])
    assert extra == [None, ["mm-7"], None]
    pool.process("kv@pod-mm@m", 0, encode_batch([
        block_stored_payload([1, 2, 3], None, tokens, 16, extra_keys=extra)
    ]))
    # untainted query matches only the untainted first block; the chain
    # diverges at the tainted block
    assert ix.score_tokens(tokens, MODEL) == {"pod-mm": 1.0}
    assert ix.score_tokens(tokens, MODEL, extra_features=extra) == {"pod-mm": 3.0}


def test_window_aware_scoring_sliding_window_pod():
    """Reference WIP feature (docs/architecture.md:291), implemented here:
    a pure sliding-window pod keeps a high score after its engine evicts
    out-of-window leading blocks — the vanilla prefix walk would score 0."""
    from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
    from llm_d_kv_cache_amd.events.publisher import (
        block_removed_payload,
        block_stored_payload,
        encode_batch,
    )

    ix = make_indexer()
    pool = KVEventsPool(EventPoolConfig(), ix)
    tokens = list(range(128))  # 8 blocks of 16
    eng = list(range(1, 9))
    pool.process("kv@pod-swa@m", 0, encode_batch([
        block_stored_payload(eng, None, tokens, 16, group_idx=0,
                             spec_kind="sliding_window", sliding_window=32)
    ]))
    assert pool.native.sliding_window_tokens("pod-swa") == 32
    assert ix.score_tokens(tokens, MODEL) == {"pod-swa": 8.0}

    # engine evicts the 6 out-of-window leading blocks
    pool.process("kv@pod-swa@m", 1, encode_batch([
        block_removed_payload(eng[:6])
    ]))
    scores, total, hits = ix.score_tokens_detailed(
        tokens, MODEL, ["pod-swa"])
    assert total == 8 and hits == 2
    # trailing run of 2 blocks covers the 32-token window at P=8
    assert scores == {"pod-swa": 8.0}

    # with window-aware scoring disabled the pod drops to 0 (prefix broken)
    ix.config.window_aware_scoring = False
    assert ix.score_tokens(tokens, MODEL, ["pod-swa"]) == {}


def test_window_hints_only_for_pure_sliding_pods():
    """A hybrid pod (any learned group without a window) gets no hint, and a
    partial tail shorter than the window earns nothing."""
    from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
    from llm_d_kv_cache_amd.events.publisher import (
        block_removed_payload,
        block_stored_payload,
        encode_batch,
    )

    ix = make_indexer()
    pool = KVEventsPool(EventPoolConfig(), ix)
    tokens = list(range(128))
    # hybrid pod: sliding group 0 + full-attention group 1
    pool.process("kv@pod-hy@m", 0, encode_batch([
        block_stored_payload(list(range(1, 9)), None, tokens, 16, group_idx=0,
                             spec_kind="sliding_window", sliding_window=32),
        block_stored_payload(list(range(11, 19)), None, tokens, 16,
                             group_idx=1, spec_kind="full_attention"),
    ]))
    assert pool.native.sliding_window_tokens("pod-hy") == 0

    # pure sliding pod with window 64 (4 blocks) holding only 2 tail blocks:
    # the window is not covered anywhere -> no windowed credit
    pool.process("kv@pod-sw2@m", 0, encode_batch([
        block_stored_payload(list(range(21, 29)), None, tokens, 16,
                             group_idx=0, spec_kind="sliding_window",
                             sliding_window=64),
        block_removed_payload(list(range(21, 27))),
    ]))
    assert pool.native.sliding_window_tokens("pod-sw2") == 64
    assert ix.score_tokens(tokens, MODEL, ["pod-sw2"]) == {}


def test_hybrid_per_group_scoring():
    """Per-group hybrid walk (reference WIP): the pod's reusable prefix is
    the MIN over its KV-cache groups — losing sliding-window leading
    blocks is fine (engines evict those by design), losing full-attention
    leading blocks kills reuse even while sliding entries remain."""
    from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
    from llm_d_kv_cache_amd.events.publisher import (
        block_removed_payload,
        block_stored_payload,
        encode_batch,
    )

    ix = make_indexer()
    pool = KVEventsPool(EventPoolConfig(), ix)
    tokens = list(range(128))  # 8 blocks
    swa = list(range(1, 9))       # engine hashes, sliding group 0
    full = list(range(11, 19))    # engine hashes, full-attn group 1
    pool.process("kv@pod-h2@m", 0, encode_batch([
        block_stored_payload(swa, None, tokens, 16, group_idx=0,
                             spec_kind="sliding_window", sliding_window=32),
        block_stored_payload(full, None, tokens, 16, group_idx=1,
                             spec_kind="full_attention"),
    ]))
    assert ix.score_tokens(tokens, MODEL) == {"pod-h2": 8.0}

    # sliding group evicts its out-of-window leading blocks: full reuse
    pool.process("kv@pod-h2@m", 1, encode_batch([
        block_removed_payload(swa[:6], group_idx=0),
    ]))
    assert ix.score_tokens(tokens, MODEL, ["pod-h2"]) == {"pod-h2": 8.0}

    # a second hybrid pod loses its full-attention FIRST block while every
    # sliding entry remains: nothing is reusable — but the vanilla
    # any-entry walk would report 8 (every key still has an entry)
    pool.process("kv@pod-h3@m", 0, encode_batch([
        block_stored_payload(list(range(21, 29)), None, tokens, 16,
                             group_idx=0, spec_kind="sliding_window",
                             sliding_window=32),
        block_stored_payload(list(range(31, 39)), None, tokens, 16,
                             group_idx=1, spec_kind="full_attention"),
        block_removed_payload([31], group_idx=1),
    ]))
    assert ix.score_tokens(tokens, MODEL, ["pod-h3"]) == {}
    ix.config.window_aware_scoring = False
    assert ix.score_tokens(tokens, MODEL, ["pod-h3"]) == {"pod-h3": 8.0}
