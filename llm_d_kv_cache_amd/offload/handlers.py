"""Transfer handlers: map offload requests to per-file engine transfers.

Capability parity with the reference worker.py handlers
(GPUToStorageHandler / StorageToGPUHandler, worker.py:186-323): a request
covers N offloaded chunks (``offloaded_block_tokens`` each, one file per
chunk per group); engine-granularity GPU block ids are split per file, the
tail chunk may be partial (short file), and loads may skip an
already-on-GPU head both at file granularity and inside the first file
(slot_offset tail-seek).
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

from .engine import TorchOffloadEngine
from .file_mapper import FileMapper


@dataclass
class TransferResult:
    job_id: int
    success: bool
    dropped: bool = False


@dataclass
class _JobInfo:
    submitted: float
    bytes: int
    kind: str  # "store" | "load"


class _BaseHandler:
    def __init__(self, engine: TorchOffloadEngine, mapper: FileMapper,
                 blocks_per_file: Sequence[int],
                 group_block_tokens: Optional[Sequence[int]] = None):
        """blocks_per_file: per group, engine blocks per offloaded chunk
        (= offloaded_block_tokens // group_block_size_tokens).
        group_block_tokens: per group, engine block size in tokens — needed
        to convert a token-level load skip into each group's block units in
        the hybrid multi-group case."""
        self.engine = engine
        self.mapper = mapper
        self.blocks_per_file = list(blocks_per_file)
        self.group_block_tokens = (
            list(group_block_tokens) if group_block_tokens is not None else None)
        self._jobs: Dict[int, _JobInfo] = {}

    def _file_bytes(self, group: int, n_blocks: int) -> int:
        geo = self.engine.group_geometry[group]
        return n_blocks * geo["num_layers"] * geo.get("record_bytes",
                                                      geo["block_bytes"])

    def get_finished(self) -> List[TransferResult]:
        out = []
        for job_id, success, dropped in self.engine.poll_finished(self._jobs.keys()):
            info = self._jobs.pop(job_id, None)
            if info is not None and success and not dropped:
                dt = max(time.time() - info.submitted, 1e-9)
                import logging

                logging.getLogger(__name__).debug(
                    "%s job %d: %.1f MB in %.1f ms (%.2f GB/s)", info.kind,
                    job_id, info.bytes / 1e6, dt * 1e3, info.bytes / dt / 1e9,
                )
            out.append(TransferResult(job_id, success, dropped))
        return out

    def wait_job(self, job_id: int) -> bool:
        self._jobs.pop(job_id, None)
        return self.engine.wait_job(job_id)


class GPUToStorageHandler(_BaseHandler):
    """Store: GPU blocks -> files. One file per (chunk, group); the tail
    chunk may cover fewer blocks (short head-partial file)."""

    def transfer_async(
        self,
        chunk_hashes: Sequence[int],
        block_ids_per_group: Dict[int, Sequence[int]],
        stream: Optional[int] = None,
    ) -> int:
        files: List[Tuple[int, str, List[int], int]] = []
        total_bytes = 0
        for group, block_ids in block_ids_per_group.items():
            bpf = self.blocks_per_file[group]
            block_ids = list(block_ids)
            if len(block_ids) > len(chunk_hashes) * bpf:
                raise ValueError(
                    f"group {group}: {len(block_ids)} blocks exceed "
                    f"{len(chunk_hashes)} chunks x {bpf} blocks/chunk")
            for ci, chunk_hash in enumerate(chunk_hashes):
                ids = block_ids[ci * bpf:(ci + 1) * bpf]
                if not ids:
                    break
                path = self.mapper.file_name(chunk_hash, group)
                files.append((group, path, [int(b) for b in ids], 0))
                total_bytes += self._file_bytes(group, len(ids))
        job_id = self.engine.async_store(files, stream)
        self._jobs[job_id] = _JobInfo(time.time(), total_bytes, "store")
        return job_id


class StorageToGPUHandler(_BaseHandler):
    """Load: files -> GPU blocks, HIGH I/O priority.

    ``skip_leading_blocks`` skips engine blocks already on the GPU: whole
    leading files are dropped and the first remaining file is tail-seeked
    via slot_offset. It is expressed in the group's own block units, which
    is only well-defined when every group uses the same block size; for
    hybrid multi-group configs pass ``skip_leading_tokens`` instead and the
    skip is converted per group (skip_g = tokens // group_block_tokens[g],
    the reference worker's per-group logical start index).
    """

    def transfer_async(
        self,
        chunk_hashes: Sequence[int],
        block_ids_per_group: Dict[int, Sequence[int]],
        skip_leading_blocks: int = 0,
        skip_leading_tokens: Optional[int] = None,
    ) -> int:
        heterogeneous = len(set(self.blocks_per_file)) > 1
        if (skip_leading_blocks and heterogeneous
                and skip_leading_tokens is None):
            raise ValueError(
                "skip_leading_blocks is ambiguous with heterogeneous group "
                "block sizes; pass skip_leading_tokens instead")
        if skip_leading_tokens is not None and self.group_block_tokens is None:
            raise ValueError(
                "skip_leading_tokens needs group_block_tokens at construction")
        files: List[Tuple[int, str, List[int], int]] = []
        total_bytes = 0
        for group, block_ids in block_ids_per_group.items():
            bpf = self.blocks_per_file[group]
            block_ids = list(block_ids)
            # block_ids correspond to blocks AFTER the skip: the caller
            # passes only the ids it wants filled.
            if skip_leading_tokens is not None:
                skip = skip_leading_tokens // self.group_block_tokens[group]
            else:
                skip = skip_leading_blocks
            first_file = skip // bpf
            slot = skip % bpf
            cursor = 0
            for ci in range(first_file, len(chunk_hashes)):
                so = slot if ci == first_file else 0
                ids = block_ids[cursor:cursor + (bpf - so)]
                cursor += len(ids)
                if not ids:
                    break
                path = self.mapper.file_name(chunk_hashes[ci], group)
                files.append((group, path, [int(b) for b in ids], so))
                total_bytes += self._file_bytes(group, len(ids))
        job_id = self.engine.async_load(files)
        self._jobs[job_id] = _JobInfo(time.time(), total_bytes, "load")
        return job_id
