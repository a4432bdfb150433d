"""Scheduler-side offload manager: stateless lookup / prepare / complete.

Capability parity with the reference SharedStorageOffloadingManager
(llmd_fs_backend/manager.py): ``lookup`` = file existence scan (stops at
the first gap — prefix semantics), ``prepare_store`` always accepts and
never evicts (the PVC evictor owns disk space), ``complete_store``
publishes BlockStored storage events so the global indexer learns the
storage tier.
"""
from __future__ import annotations

import os
from typing import List, Optional, Sequence

from .events import StorageEventPublisher
from .file_mapper import FileMapper


class SharedStorageOffloadManager:
    def __init__(self, mapper: FileMapper, num_groups: int = 1,
                 publisher: Optional[StorageEventPublisher] = None):
        self.mapper = mapper
        self.num_groups = num_groups
        self.publisher = publisher

    def lookup(self, chunk_hashes: Sequence[int]) -> int:
        """Number of leading chunks whose files all exist (every group)."""
        hits = 0
        for h in chunk_hashes:
            if all(
                os.path.exists(self.mapper.file_name(h, g))
                for g in range(self.num_groups)
            ):
                hits += 1
            else:
                break
        return hits

    def prepare_store(self, chunk_hashes: Sequence[int]) -> List[int]:
        """Always accepts; returns the hashes to store (all of them —
        dedupe happens at the engine via skip-if-exists)."""
        return list(chunk_hashes)

    def complete_store(
        self,
        chunk_hashes: Sequence[int],
        token_ids: Sequence[int],
        parent_chunk_hash: Optional[int] = None,
    ) -> None:
        if self.publisher is not None and chunk_hashes:
            self.publisher.publish_block_stored(
                chunk_hashes, token_ids, parent_chunk_hash
            )

    def complete_remove(self, chunk_hashes: Sequence[int]) -> None:
        if self.publisher is not None and chunk_hashes:
            self.publisher.publish_block_removed(chunk_hashes)
