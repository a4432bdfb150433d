"""Hot-op kernel API: direct access to the CDNA4 kernels the data plane is
built from, for embedding in other MI355X services.

- gather/scatter: paged KV pages <-> contiguous slabs at the HBM roof
  (BlockCopier; see csrc/offload/kernels.hip).
- prefix_hash: batched chained block hashing on device token buffers
  (one lane per sequence).
- fp8 serialize: quantizing gather / dequantizing scatter are methods on
  the copier (gather_fp8 / scatter_fp8 with packed_bytes_fp8 /
  fp8_scratch_bytes sizing); the offload engine's serialize="fp8_e4m3"
  rides the same kernels.
"""
from __future__ import annotations

from typing import List, Optional, Sequence


def block_copier(groups: Sequence[Sequence], device: Optional[int] = None):
    """BlockCopier over torch page tensors (GPU or CPU twins)."""
    from .. import ensure_offload_native

    ko = ensure_offload_native()
    gpu = groups[0][0].is_cuda
    native = [
        ([t.data_ptr() for t in g],
         [t.stride(0) * t.element_size() for t in g],
         g[0].stride(0) * g[0].element_size(),
         int(g[0].shape[0]))
        for g in groups
    ]
    dev = device if device is not None else (
        groups[0][0].device.index or 0 if gpu else 0)
    return ko.BlockCopier(native, gpu, dev)


def prefix_hash(tokens, seq_offsets, seeds, block_size: int = 16,
                stream: Optional[int] = None) -> List[List[int]]:
    """Batched chained block hashing on the GPU.

    tokens: int32 CUDA tensor (flat); seq_offsets: int64 CUDA tensor
    [n_seq+1]; seeds: int64 CUDA tensor [n_seq] (chain seeds, two's
    complement of the uint64 parent/model hash). Returns per-sequence key
    lists (uint64 as Python ints). Must match TokenProcessor bit-exactly
    (tested in tests/test_offload_gpu.py).
    """
    import torch

    from .. import ensure_offload_native

    ko = ensure_offload_native()
    n_seq = seeds.numel()
    lens = (seq_offsets[1:] - seq_offsets[:-1]) // block_size
    key_off = torch.zeros(n_seq + 1, dtype=torch.int64, device=tokens.device)
    torch.cumsum(lens, 0, out=key_off[1:])
    keys = torch.zeros(int(key_off[-1].item()), dtype=torch.int64,
                       device=tokens.device)
    if stream is None:
        stream = torch.cuda.current_stream().cuda_stream
    ko.prefix_hash(tokens.data_ptr(), seq_offsets.data_ptr(), seeds.data_ptr(),
                   keys.data_ptr(), key_off.data_ptr(), block_size, n_seq,
                   stream)
    torch.cuda.synchronize()
    off = key_off.cpu().tolist()
    vals = keys.cpu().numpy().astype("uint64").tolist()
    return [vals[off[i]:off[i + 1]] for i in range(n_seq)]
