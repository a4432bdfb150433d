"""Token processor: chained FNV-64a/CBOR block hashing.

Cross-checked against an independent pure-Python CBOR+FNV implementation
(tests/reference_impl.py). Mirrors the reference test strategy for
pkg/kvcache/kvblock/token_processor_test.go.
"""
import numpy as np
import pytest

import reference_impl as ref
from llm_d_kv_cache_amd import ensure_native

k = ensure_native()

MODEL = "meta-llama/Llama-3.1-8B-Instruct"


def test_matches_pure_python_reference():
    tp = k.TokenProcessor(16, "")
    tokens = list(range(1000, 1064))
    got = tp.tokens_to_block_keys(tokens, MODEL)
    want = ref.block_keys(tokens, MODEL, block_size=16)
    assert got == want


def test_partial_tail_dropped():
    tp = k.TokenProcessor(16, "")
    assert tp.tokens_to_block_keys(list(range(15)), MODEL) == []
    assert len(tp.tokens_to_block_keys(list(range(16)), MODEL)) == 1
    assert len(tp.tokens_to_block_keys(list(range(47)), MODEL)) == 2


def test_chain_is_prefix_dependent():
    tp = k.TokenProcessor(16, "")
    a = tp.tokens_to_block_keys(list(range(32)), MODEL)
    b = tp.tokens_to_block_keys(list(range(16)) + list(range(100, 116)), MODEL)
    assert a[0] == b[0]
    assert a[1] != b[1]


def test_parent_key_continuation():
    tp = k.TokenProcessor(16, "")
    tokens = list(range(48))
    full = tp.tokens_to_block_keys(tokens, MODEL)
    head = tp.tokens_to_block_keys(tokens[:16], MODEL)
    tail = tp.tokens_to_block_keys(tokens[16:], MODEL, parent=head[-1])
    assert full == head + tail


def test_model_name_seeds_chain():
    tp = k.TokenProcessor(16, "")
    a = tp.tokens_to_block_keys(list(range(16)), "model-a")
    b = tp.tokens_to_block_keys(list(range(16)), "model-b")
    assert a != b


def test_hash_seed_changes_hashes():
    a = k.TokenProcessor(16, "").tokens_to_block_keys(list(range(16)), MODEL)
    b = k.TokenProcessor(16, "seed-1").tokens_to_block_keys(list(range(16)), MODEL)
    assert a != b
    want = ref.block_keys(list(range(16)), MODEL, hash_seed="seed-1")
    assert b == want


def test_block_size_validation():
    with pytest.raises(Exception):
        k.TokenProcessor(0, "")
    with pytest.raises(Exception):
        k.TokenProcessor(-4, "")


def test_extra_features_taint_hash():
    tp = k.TokenProcessor(16, "")
    tokens = list(range(32))
    plain = tp.tokens_to_block_keys(tokens, MODEL)
    tainted = tp.tokens_to_block_keys(tokens, MODEL, extra=[["mm-hash-1"], None])
    assert plain[0] != tainted[0]
    # Second block depends on the first through the chain even though its
    # own extra is None.
    assert plain[1] != tainted[1]
    want = ref.block_keys(tokens, MODEL, extra=[["mm-hash-1"], None])
    assert tainted == want


def test_extra_features_length_mismatch_raises():
    tp = k.TokenProcessor(16, "")
    with pytest.raises(Exception):
        tp.tokens_to_block_keys(list(range(32)), MODEL, extra=[["x"]])


def test_numpy_tokens_fast_path():
    tp = k.TokenProcessor(16, "")
    tokens = np.arange(64, dtype=np.uint32)
    assert tp.tokens_to_block_keys(tokens, MODEL) == tp.tokens_to_block_keys(
        tokens.tolist(), MODEL
    )


def test_large_token_values():
    tp = k.TokenProcessor(16, "")
    tokens = [2**31 + i for i in range(16)]
    got = tp.tokens_to_block_keys(tokens, MODEL)
    assert got == ref.block_keys(tokens, MODEL)


def test_hash_block_primitive():
    got = k.hash_block(12345, list(range(8)), ["a", "b"])
    assert got == ref.hash_block(12345, list(range(8)), ["a", "b"])
    assert k.hash_block(0, None, None) == ref.hash_block(0, None, None)


def test_determinism_across_instances():
    vals = set()
    for _ in range(4):
        tp = k.TokenProcessor(16, "test-seed")
        vals.add(tuple(tp.tokens_to_block_keys(list(range(16)), MODEL)))
    assert len(vals) == 1


def test_custom_block_size():
    tp = k.TokenProcessor(64, "")
    tokens = list(range(200))
    keys = tp.tokens_to_block_keys(tokens, MODEL)
    assert len(keys) == 3
    assert keys == ref.block_keys(tokens, MODEL, block_size=64)
