"""Tiered KV resolution: one call walks the whole hierarchy.

A chunk requested for a request's prefix is resolved in cost order:

  1. local filesystem/DRAM tier (the offload engine's load path — a
     pinned-cache hit never touches the filesystem),
  2. peer GPUs over RCCL/xGMI (cheaper than any host round trip when a
     same-node rank still holds the blocks in HBM),
  3. miss -> the caller recomputes (prefill).

This is the node-side consumer of the global index: the scheduler's
scorer says WHICH pod likely holds a prefix; TieredKVLoader turns that
into bytes in local KV pages.
"""
from __future__ import annotations

import logging
from typing import Optional, Sequence

log = logging.getLogger(__name__)


def make_dram_lookup(engine, mapper):
    """Build a PeerMigrationService dram_lookup callback from an offload
    engine + file mapper: serves PULL requests out of the pinned host-DRAM
    cache when the blocks are no longer registered in HBM. Returns
    (host_tensor, is_fp8) or None."""
    fp8 = engine.config.serialize == "fp8_e4m3"

    def lookup(chunk_hash, group, n_blocks):
        path = mapper.file_name(chunk_hash, group)
        t = engine.dram_chunk(path, n_blocks, group)
        if t is None:
            return None
        return t, fp8

    return lookup


class TieredKVLoader:
    def __init__(self, load_handler=None, peer_service=None,
                 peer_ranks: Sequence[int] = (), pull_timeout_s: float = 30.0):
        """load_handler: StorageToGPUHandler (or None when the node has no
        storage tier). peer_service: PeerMigrationService (or None).
        peer_ranks: same-node ranks to try, in preference order."""
        self.load_handler = load_handler
        self.peer = peer_service
        self.peer_ranks = list(peer_ranks)
        self.pull_timeout_s = pull_timeout_s

    def resolve(self, chunk_hash: int, block_ids: Sequence[int],
                group: int = 0,
                preferred_ranks: Optional[Sequence[int]] = None) -> str:
        """Fill local pages `block_ids` with the chunk's KV. Returns the
        tier that served it: 'storage' (covers the DRAM cache), 'peer', or
        'miss'."""
        if self.load_handler is not None:
            job = self.load_handler.transfer_async([chunk_hash],
                                                   {group: list(block_ids)})
            if self._wait(self.load_handler, job):
                return "storage"
        if self.peer is not None:
            for rank in (preferred_ranks if preferred_ranks is not None
                         else self.peer_ranks):
                try:
                    ok = self.peer.pull(chunk_hash, group, list(block_ids),
                                        src_rank=rank,
                                        timeout=self.pull_timeout_s
                                        ).result(timeout=self.pull_timeout_s + 5)
                except Exception as e:
                    log.warning("peer pull from rank %d failed: %s", rank, e)
                    continue
                if ok:
                    return "peer"
        return "miss"

    def resolve_prefix(self, chunk_hashes: Sequence[int],
                       block_ids: Sequence[int], blocks_per_chunk: int,
                       group: int = 0) -> int:
        """Resolve a prefix chunk by chunk; stops at the first miss (a
        prefix must be contiguous). Returns the number of chunks filled.

        All storage loads are issued CONCURRENTLY up front (the engine's
        I/O pool parallelizes them); the longest contiguous success prefix
        counts. Gaps are then offered to the peer tier in batched
        pull_many calls (one control round trip + one xGMI transfer per
        batch); the granted prefix of each batch extends the fill up to
        the next storage-loaded chunk. The first chunk no tier serves
        ends the walk."""
        import time as _time

        from . import MAX_BATCH

        n = len(chunk_hashes)

        def ids_of(ci):
            return list(
                block_ids[ci * blocks_per_chunk:(ci + 1) * blocks_per_chunk])

        status = [None] * n  # True: storage-loaded, False/None: not
        if self.load_handler is not None:
            jobs = {}
            for ci in range(n):
                ids = ids_of(ci)
                if not ids:
                    break
                jid = self.load_handler.transfer_async(
                    [chunk_hashes[ci]], {group: ids})
                jobs[jid] = ci
            deadline = _time.time() + self.pull_timeout_s
            pending = set(jobs)
            while pending and _time.time() < deadline:
                for res in self.load_handler.get_finished():
                    ci = jobs.get(res.job_id)
                    if ci is not None:
                        status[ci] = res.success
                        pending.discard(res.job_id)
                if pending:
                    _time.sleep(0.002)

        filled = 0
        while filled < n:
            if status[filled]:
                filled += 1
                continue
            if not ids_of(filled) or self.peer is None:
                break
            end = filled
            while (end < n and end - filled < MAX_BATCH
                   and not status[end] and ids_of(end)):
                end += 1
            batch = [(chunk_hashes[ci], group, ids_of(ci))
                     for ci in range(filled, end)]
            granted_prefix = 0
            for rank in self.peer_ranks:
                try:
                    res = self.peer.pull_many(
                        batch, src_rank=rank, timeout=self.pull_timeout_s
                    ).result(timeout=self.pull_timeout_s + 5)
                except Exception as e:
                    log.warning("peer batch pull from rank %d failed: %s",
                                rank, e)
                    continue
                k = 0
                while k < len(res) and res[k]:
                    k += 1
                if k:
                    granted_prefix = k
                    break
            if granted_prefix == 0:
                break
            filled += granted_prefix
        return filled

    def _wait(self, handler, job_id) -> bool:
        import time

        deadline = time.time() + self.pull_timeout_s
        while time.time() < deadline:
            for res in handler.get_finished():
                if res.job_id == job_id:
                    return res.success
            time.sleep(0.002)
        return False
