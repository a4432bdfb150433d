"""Storage-tier KVEvents publisher.

Emits BlockStored / BlockRemoved with medium=SHARED_STORAGE (or
OBJECT_STORE) on topic ``kv@<medium>@<model>`` in the exact engine wire
format, so the global indexer ingests storage-tier locality through the
same adapter as GPU events. Capability parity with the reference
llmd_fs_backend/event_publisher.py:31-158.
"""
from __future__ import annotations

from typing import Optional, Sequence

from ..events.publisher import (
    block_removed_payload,
    block_stored_payload,
    encode_batch,
)
from .. import ensure_native


class StorageMedium:
    SHARED_STORAGE = "SHARED_STORAGE"
    OBJECT_STORE = "OBJECT_STORE"


class StorageEventPublisher:
    def __init__(self, endpoint: str, model: str,
                 medium: str = StorageMedium.SHARED_STORAGE,
                 offloaded_block_tokens: int = 256, bind: bool = False):
        k = ensure_native()
        self._pub = k.Publisher(endpoint, bind=bind)
        self.model = model
        self.medium = medium
        self.offloaded_block_tokens = offloaded_block_tokens
        self._seq = 0

    @property
    def topic(self) -> str:
        # the "pod" slot carries the medium: storage is a shared tier, not a pod
        return f"kv@{self.medium}@{self.model}"

    def _publish(self, events) -> None:
        payload = encode_batch(events)
        self._pub.publish(self.topic, self._seq, payload)
        self._seq += 1

    def publish_block_stored(
        self,
        chunk_hashes: Sequence[int],
        token_ids: Sequence[int],
        parent_chunk_hash: Optional[int] = None,
    ) -> None:
        self._publish([
            block_stored_payload(
                list(chunk_hashes), parent_chunk_hash, list(token_ids),
                self.offloaded_block_tokens, medium=self.medium,
            )
        ])

    def publish_block_removed(self, chunk_hashes: Sequence[int]) -> None:
        self._publish([block_removed_payload(list(chunk_hashes), medium=self.medium)])

    def close(self) -> None:
        self._pub.close()
