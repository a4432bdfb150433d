"""Frozen wire-format + hash-chain fixtures.

The byte string below is a recorded `KVEventBatch` in the vLLM KVEvents
wire format (msgspec `array_like=True, omit_defaults=False` positional
arrays — reference vllm_adapter.go:133-149) and the key constants are the
FNV-64a/CBOR chain values for its token stream. Both are frozen as
literals: any change to the msgpack field order, the CBOR canonical
encoding, the seed mixing, or the chain rule breaks these asserts loudly
instead of silently diverging from every already-deployed peer.

(The reference repo ships no golden hash constants to borrow —
token_processor_test.go asserts determinism only — so these literals are
this repo's own recorded values, cross-checked at recording time against
the independent CPU oracle in tests/reference_impl.py.)
"""
import pytest

from llm_d_kv_cache_amd import ensure_native
from reference_impl import block_keys

k = ensure_native()

# One batch: BlockStored([0x1111,0x2222,0x3333], parent=None, tokens=0..47,
# block_size=16, medium="gpu") + BlockRemoved([0x1111]); ts frozen at
# recording time.
GOLDEN_BATCH = (
    b"\x92\xcbA\xda\xa9\xe4\xa6\xcd\xbd\xf6\x92\x99\xabBlockStored"
    b"\x93\xcd\x11\x11\xcd\"\"\xcd33\xc0\xdc\x000"
    b"\x00\x01\x02\x03\x04\x05\x06\x07\x08\t\n\x0b\x0c\r\x0e\x0f"
    b"\x10\x11\x12\x13\x14\x15\x16\x17\x18\x19\x1a\x1b\x1c\x1d\x1e\x1f"
    b" !\"#$%&'()*+,-./"
    b"\x10\xc0\xa3gpu\xc0\xc0"
    b"\x93\xacBlockRemoved\x91\xcd\x11\x11\xc0"
)

TOKENS = list(range(48))
MODEL = "meta-llama/Llama-3-8B"

# tokens 0..47, block 16, model "meta-llama/Llama-3-8B", empty hash seed
GOLDEN_KEYS = [0xE251F9E737DBBF94, 0x68150529FE26B475, 0xB2D97119A87D974B]
# 16 tokens, model "m", seed "seed1", parent chain 0xDEADBEEF
GOLDEN_CHAINED_KEY = 0x15EBF3E30A863A30


def test_golden_batch_bytes_are_stable():
    from llm_d_kv_cache_amd.events.publisher import (
        block_removed_payload,
        block_stored_payload,
        encode_batch,
    )

    fresh = encode_batch([
        block_stored_payload([0x1111, 0x2222, 0x3333], None, TOKENS, 16,
                             medium="gpu"),
        block_removed_payload([0x1111]),
    ])
    # everything but the leading float64 timestamp must match byte-for-byte
    assert fresh[:1] == GOLDEN_BATCH[:1]
    assert fresh[10:] == GOLDEN_BATCH[10:], "wire encoding drifted"


def test_golden_batch_parses_to_expected_index_state():
    ix = k.InMemoryIndex()
    tp = k.TokenProcessor(16, "")
    pool = k.EventPool(tp, ix, 1)
    pool.process("kv@pod-g@" + MODEL, 0, GOLDEN_BATCH)
    st = pool.stats()
    assert st.parse_failures == 0 and st.handler_failures == 0
    # the BlockStored keys land under the frozen request-key constants;
    # 0x1111's BlockRemoved evicted the first block's entry
    got = ix.lookup(GOLDEN_KEYS)
    assert set(got.keys()) == set(GOLDEN_KEYS[1:])
    assert [e.pod for e in got[GOLDEN_KEYS[1]]] == ["pod-g"]
    assert ix.get_request_key(0x2222) == GOLDEN_KEYS[1]
    assert ix.get_request_key(0x3333) == GOLDEN_KEYS[2]


def test_golden_hash_constants():
    tp = k.TokenProcessor(16, "")
    assert tp.tokens_to_block_keys(TOKENS, MODEL, 0) == GOLDEN_KEYS
    tp2 = k.TokenProcessor(16, "seed1")
    assert tp2.tokens_to_block_keys(TOKENS[:16], "m", 0xDEADBEEF) == \
        [GOLDEN_CHAINED_KEY]
    # and the independent CPU oracle agrees with the frozen constants
    assert block_keys(TOKENS, MODEL, 16, "") == GOLDEN_KEYS
