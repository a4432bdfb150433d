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
