"""Torch-facing construction of the native offload engine.

Tensors are handed to C++ as raw (pointer, stride, bytes) descriptors; the
native module has no libtorch dependency. On a GPU box the engine REQUIRES
the HIP extension and CUDA(=HIP) tensors — there is no silent eager
fallback; ``copy_path='host'`` on CPU tensors is an explicit mode used by
CPU-only CI.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Sequence

DEFAULT_STAGING_BUDGET_BYTES = 150 * 1024**3  # parity with reference worker.py:70


@dataclass
class OffloadEngineConfig:
    io_threads: int = 16
    gpu_blocks_per_file: int = 16
    read_preferring_ratio: float = 0.75
    max_write_queued_seconds: float = 30.0
    copy_path: str = "staged"  # staged | zero_copy | host
    serialize: str = "raw"     # raw | fp8_e4m3 (bf16 pages -> fp8 + scale)
    host_cache_bytes: int = 0  # pinned-DRAM cache tier (0 = off)
    write_policy: str = "through"  # through | back (flush async via DRAM tier)
    direct_io: bool = False    # O_DIRECT for NVMe (auto-fallback elsewhere)
    device: int = 0
    staging_budget_bytes: int = DEFAULT_STAGING_BUDGET_BYTES


class TorchOffloadEngine:
    """Owns the native StorageOffloadEngine built from torch KV tensors.

    groups: one entry per KV-cache group; each a list of per-layer tensors
    shaped (num_blocks, ...) with block-contiguous rows.
    """

    def __init__(self, groups: Sequence[Sequence], config: OffloadEngineConfig):
        import torch

        from .. import ensure_offload_native

        self._ko = ensure_offload_native()
        self.config = config

        if not groups or not groups[0]:
            raise ValueError("need at least one group with one layer tensor")
        first = groups[0][0]
        gpu_mode = first.is_cuda
        for g in groups:
            for t in g:
                if t.is_cuda != gpu_mode:
                    raise ValueError("all KV tensors must live on the same device kind")
        if gpu_mode and config.copy_path == "host":
            raise ValueError("copy_path='host' is invalid for GPU tensors")
        if not gpu_mode and config.copy_path != "host":
            # Explicit: CPU tensors run the host path. Constructing with GPU
            # copy paths on CPU tensors is a config error, not a fallback.
            raise ValueError(
                f"copy_path='{config.copy_path}' requires GPU tensors; "
                "use copy_path='host' for CPU tensors"
            )

        native_groups = []
        self.group_geometry: List[dict] = []
        max_file_bytes = 0
        for g in groups:
            ptrs = [t.data_ptr() for t in g]
            strides = [t.stride(0) * t.element_size() for t in g]
            block_bytes = {t.stride(0) * t.element_size() for t in g}
            if len(block_bytes) != 1:
                raise ValueError("all layers of a group must share block_bytes")
            bb = block_bytes.pop()
            # the actual payload per block may be smaller than the stride;
            # canonical layouts are dense so stride == payload
            native_groups.append((ptrs, strides, bb, int(g[0].shape[0])))
            record_bytes = bb // 2 + 4 if config.serialize == "fp8_e4m3" else bb
            self.group_geometry.append(
                {
                    "num_layers": len(g),
                    "block_bytes": bb,
                    "record_bytes": record_bytes,
                    "num_device_blocks": g[0].shape[0],
                }
            )
            max_file_bytes = max(
                max_file_bytes, config.gpu_blocks_per_file * len(g) * bb
            )

        # staging budget clamp: per worker we allocate host staging (and in
        # staged mode an equal device bounce)
        per_thread = max_file_bytes * (2 if config.copy_path == "staged" else 1)
        io_threads = config.io_threads
        if per_thread > 0:
            budget_threads = max(1, config.staging_budget_bytes // per_thread)
            io_threads = min(io_threads, budget_threads)

        stream = 0
        if gpu_mode:
            torch.cuda.init()
        self.gpu_mode = gpu_mode
        self._engine = self._ko.StorageOffloadEngine(
            native_groups,
            io_threads=int(io_threads),
            gpu_blocks_per_file=config.gpu_blocks_per_file,
            read_preferring_ratio=config.read_preferring_ratio,
            max_write_queued_seconds=config.max_write_queued_seconds,
            gpu_mode=gpu_mode,
            device=config.device,
            copy_path=config.copy_path,
            serialize=config.serialize,
            host_cache_bytes=config.host_cache_bytes,
            write_policy=config.write_policy,
            direct_io=config.direct_io,
        )
        del stream
        # keep tensor refs: the native engine holds raw pointers
        self._tensors = [list(g) for g in groups]
        import threading

        self._fin_lock = threading.Lock()
        self._fin_buffer = {}

    @property
    def native(self):
        return self._engine

    def current_stream_handle(self) -> int:
        if not self.gpu_mode:
            return 0
        import torch

        return torch.cuda.current_stream().cuda_stream

    def async_store(self, files, stream: int = None) -> int:
        if stream is None:
            stream = self.current_stream_handle()
        return self._engine.async_store(files, stream)

    def async_load(self, files) -> int:
        return self._engine.async_load(files)

    def get_finished(self):
        return self._engine.get_finished()

    def poll_finished(self, job_ids):
        """Drain engine completions into a shared buffer and pop the ones in
        job_ids — several handlers can share one engine without stealing
        each other's completions."""
        with self._fin_lock:
            for jid, success, dropped in self._engine.get_finished():
                self._fin_buffer[jid] = (success, dropped)
            out = []
            for jid in list(job_ids):
                if jid in self._fin_buffer:
                    success, dropped = self._fin_buffer.pop(jid)
                    out.append((jid, success, dropped))
            return out

    def wait_job(self, job_id: int) -> bool:
        return self._engine.wait_job(job_id)

    def cancel_job(self, job_id: int) -> bool:
        """Non-blocking cancel: queued tasks bail when dequeued. Set flags
        for every job being preempted first, then wait_job each."""
        return self._engine.cancel_job(job_id)

    def stats(self):
        return self._engine.stats()

    def dram_chunk(self, path: str, n_blocks: int, group: int = 0):
        """Read a chunk file's payload out of the pinned host-DRAM cache
        (peer/DRAM-tier bridge). Returns a host uint8 tensor of exactly the
        first n_blocks' packed bytes, or None on a cache miss / short
        entry. The payload layout matches the file layout (raw gather
        layout, or fp8 tile records when serialize="fp8_e4m3")."""
        import torch

        geo = self.group_geometry[group]
        want = n_blocks * geo["num_layers"] * geo["record_bytes"]
        cap = (self.config.gpu_blocks_per_file * geo["num_layers"]
               * geo["record_bytes"])
        buf = torch.empty(cap, dtype=torch.uint8)
        got = self._engine.host_cache_read(path, buf.data_ptr(), cap)
        if got < want:
            return None
        return buf[:want]
