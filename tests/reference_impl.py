"""Pure-Python reference implementations used to cross-check the native code.

Independent of the C++ implementation: canonical CBOR (RFC 8949
deterministic encoding) + FNV-64a, written from the spec. Numerics/test
oracle only — never imported by the framework itself.
"""
from __future__ import annotations

from typing import Optional, Sequence

FNV_OFFSET = 0xCBF29CE484222325
FNV_PRIME = 0x100000001B3
U64 = (1 << 64) - 1


def fnv64a(data: bytes) -> int:
    h = FNV_OFFSET
    for b in data:
        h = ((h ^ b) * FNV_PRIME) & U64
    return h


def cbor_head(major: int, value: int) -> bytes:
    m = major << 5
    if value < 24:
        return bytes([m | value])
    if value <= 0xFF:
        return bytes([m | 24, value])
    if value <= 0xFFFF:
        return bytes([m | 25]) + value.to_bytes(2, "big")
    if value <= 0xFFFFFFFF:
        return bytes([m | 26]) + value.to_bytes(4, "big")
    return bytes([m | 27]) + value.to_bytes(8, "big")


def cbor_uint(v: int) -> bytes:
    return cbor_head(0, v)


def cbor_text(s: str) -> bytes:
    b = s.encode("utf-8")
    return cbor_head(3, len(b)) + b


def cbor_array(n: int) -> bytes:
    return cbor_head(4, n)


CBOR_NULL = b"\xf6"


def hash_block(parent: int, tokens: Optional[Sequence[int]],
               extra: Optional[Sequence[str]]) -> int:
    buf = cbor_array(3) + cbor_uint(parent)
    if tokens is None:
        buf += CBOR_NULL
    else:
        buf += cbor_array(len(tokens)) + b"".join(cbor_uint(t) for t in tokens)
    if extra is None:
        buf += CBOR_NULL
    else:
        buf += cbor_array(len(extra)) + b"".join(cbor_text(s) for s in extra)
    return fnv64a(buf)


def block_keys(tokens: Sequence[int], model: str, block_size: int = 16,
               hash_seed: str = "", parent: int = 0,
               extra: Optional[Sequence[Optional[Sequence[str]]]] = None) -> list:
    if parent == 0:
        parent = hash_block(fnv64a(hash_seed.encode()), None, None) if False else \
            fnv64a_chain_seed(model, hash_seed)
    keys = []
    n_chunks = len(tokens) // block_size
    for c in range(n_chunks):
        chunk = tokens[c * block_size:(c + 1) * block_size]
        ex = extra[c] if extra is not None else None
        parent = hash_block(parent, chunk, ex)
        keys.append(parent)
    return keys


def fnv64a_chain_seed(model: str, hash_seed: str = "") -> int:
    init = fnv64a(hash_seed.encode())
    buf = cbor_array(3) + cbor_uint(init) + CBOR_NULL + cbor_text(model)
    return fnv64a(buf)
