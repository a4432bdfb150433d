"""KVEvents publisher: vLLM-wire-format msgpack batches over ZMTP PUB.

Builds the exact positional msgpack arrays engines emit (msgspec
array_like=True) and ships them as 3-frame ZMQ messages
[topic, 8-byte BE seq, payload] on topic "kv@<pod>@<model>".

Serves both as the test-fleet simulator (reference examples/helper/
publisher.go parity) and as the base transport of the storage-tier event
publisher (reference llmd_fs_backend/event_publisher.py parity).
"""
from __future__ import annotations

import time
from typing import Iterable, List, Optional, Sequence

import msgpack

from .. import ensure_native

UINT64_MASK = (1 << 64) - 1


def block_stored_payload(
    block_hashes: Sequence[int],
    parent_hash: Optional[int],
    token_ids: Sequence[int],
    block_size: int,
    lora_id: Optional[int] = None,
    medium: Optional[str] = None,
    lora_name: Optional[str] = None,
    extra_keys: Optional[Sequence[Optional[Sequence[str]]]] = None,
    group_idx: Optional[int] = None,
    spec_kind: Optional[str] = None,
    sliding_window: Optional[int] = None,
) -> list:
    def mask(h):
        return (h & UINT64_MASK) if isinstance(h, int) else h

    ev = [
        "BlockStored",
        [mask(h) for h in block_hashes],
        mask(parent_hash) if parent_hash is not None else None,
        list(token_ids),
        block_size,
        lora_id,
        medium,
        lora_name,
        list(extra_keys) if extra_keys is not None else None,
    ]
    if group_idx is not None or spec_kind is not None or sliding_window is not None:
        ev += [group_idx, spec_kind, sliding_window]
    return ev


def block_removed_payload(
    block_hashes: Sequence[int], medium: Optional[str] = None,
    group_idx: Optional[int] = None,
) -> list:
    ev = ["BlockRemoved", [h & UINT64_MASK for h in block_hashes], medium]
    if group_idx is not None:
        ev.append(group_idx)
    return ev


def all_blocks_cleared_payload() -> list:
    return ["AllBlocksCleared"]


def encode_batch(events: Iterable[list], ts: Optional[float] = None,
                 dp_rank: Optional[int] = None) -> bytes:
    batch: List = [ts if ts is not None else time.time(), list(events)]
    if dp_rank is not None:
        batch.append(dp_rank)
    return msgpack.packb(batch, use_bin_type=True)


class EventPublisher:
    """ZMTP PUB endpoint emitting KVEvents batches for one pod."""

    def __init__(self, endpoint: str, pod_id: str, model: str,
                 bind: bool = True, username: str = "", password: str = ""):
        self._k = ensure_native()
        self._pub = self._k.Publisher(endpoint, bind=bind, username=username,
                                      password=password)
        self.pod_id = pod_id
        self.model = model
        self._seq = 0

    @property
    def port(self) -> int:
        return self._pub.port

    @property
    def topic(self) -> str:
        return f"kv@{self.pod_id}@{self.model}"

    def publish_events(self, events: Iterable[list], ts: Optional[float] = None,
                       dp_rank: Optional[int] = None) -> int:
        payload = encode_batch(events, ts=ts, dp_rank=dp_rank)
        seq = self._seq
        self._pub.publish(self.topic, seq, payload)
        self._seq += 1
        return seq

    def close(self) -> None:
        self._pub.close()
