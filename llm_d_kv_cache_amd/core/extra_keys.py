"""Multimodal extra features (read side).

Capability parity with the reference pkg/kvcache/kvblock/extra_keys.go
(ComputeBlockExtraFeatures, mirroring vLLM's _gen_mm_extra_hash_keys):
convert tokenizer-provided multimodal placeholder ranges into per-block
extra-feature lists that taint the hash chain exactly like the engine's
write side does — a block's extra is the ordered identifiers of every
multimodal item whose placeholder range overlaps it.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple


@dataclass(frozen=True)
class PlaceholderRange:
    """Contiguous placeholder-token span of one multimodal item."""
    offset: int  # absolute start token index
    length: int  # number of placeholder tokens


def compute_block_extra_features(
    n_tokens: int,
    block_size: int,
    mm_hashes: Sequence[str],
    placeholders: Sequence[PlaceholderRange],
) -> Optional[List[Optional[List[str]]]]:
    """Per-full-block extra features for score_tokens/tokens_to_block_keys.

    mm_hashes[i] identifies the item occupying placeholders[i]. Returns
    None when there are no items (pure text — no taint), else one entry
    per full block: None or the ordered overlapping identifiers.
    """
    if len(mm_hashes) != len(placeholders):
        raise ValueError("mm_hashes and placeholders must align")
    if not mm_hashes:
        return None
    items: List[Tuple[int, int, str]] = sorted(
        (p.offset, p.offset + p.length, h)
        for p, h in zip(placeholders, mm_hashes)
    )
    n_blocks = n_tokens // block_size
    out: List[Optional[List[str]]] = [None] * n_blocks
    for b in range(n_blocks):
        lo, hi = b * block_size, (b + 1) * block_size
        hashes = [h for (s, e, h) in items if s < hi and e > lo]
        if hashes:
            out[b] = hashes
    return out
