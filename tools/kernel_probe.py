#!/usr/bin/env python3
"""Kernel microbench: gather/scatter HBM bandwidth on Llama-3-8B KV geometry.

Times the CDNA4 gather (paged KV pages -> contiguous slab) and scatter
kernels in isolation (HBM->HBM, both directions r+w) — the compute-side
ceiling of the offload path, independent of PCIe/file I/O. Run under
rocprofv3 (--stats or --pmc) for counter evidence.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from llm_d_kv_cache_amd import ensure_offload_native  # noqa: E402

# default: Llama-3-8B geometry; --small switches to the 70B TP=8 shard
# (8 KiB tiles x 80 layers — the small-tile regime)
SMALL = "--small" in sys.argv
NUM_LAYERS = 80 if SMALL else 32
BLOCK_BYTES = (8 if SMALL else 64) * 1024
NUM_BLOCKS = 2048
BPF = 16
args = [a for a in sys.argv[1:] if not a.startswith("-")]
ITERS = int(args[0]) if args else 50


def main():
    ko = ensure_offload_native()
    assert torch.cuda.is_available()
    group = [
        torch.randint(0, 255, (NUM_BLOCKS, BLOCK_BYTES), dtype=torch.uint8,
                      device="cuda")
        for _ in range(NUM_LAYERS)
    ]
    copier = ko.BlockCopier(
        [([t.data_ptr() for t in group], [t.stride(0) for t in group],
          BLOCK_BYTES)],
        gpu_mode=True,
    )
    slab_bytes = copier.packed_bytes(0, BPF)
    slab = torch.empty(slab_bytes, dtype=torch.uint8, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    ids = list(range(0, BPF * 8, 8))  # strided pages: realistic gather

    # warmup
    copier.gather(0, ids, slab.data_ptr(), stream)
    copier.scatter(0, ids, slab.data_ptr(), stream)
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    for _ in range(ITERS):
        copier.gather(0, ids, slab.data_ptr(), stream)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    moved = slab_bytes * ITERS
    print(f"gather:  {moved / dt / 1e9:.1f} GB/s payload "
          f"({2 * moved / dt / 1e9:.1f} GB/s r+w), {dt / ITERS * 1e6:.0f} us/launch")

    t0 = time.perf_counter()
    for _ in range(ITERS):
        copier.scatter(0, ids, slab.data_ptr(), stream)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"scatter: {moved / dt / 1e9:.1f} GB/s payload "
          f"({2 * moved / dt / 1e9:.1f} GB/s r+w), {dt / ITERS * 1e6:.0f} us/launch")

    # fp8 split gather (amax + quantize) and dequantizing scatter, isolated
    nb8 = copier.packed_bytes_fp8(0, BPF)
    slab8 = torch.empty(nb8, dtype=torch.uint8, device="cuda")
    scratch = torch.empty(copier.fp8_scratch_bytes(0, BPF),
                          dtype=torch.uint8, device="cuda")
    copier.gather_fp8(0, ids, slab8.data_ptr(), scratch.data_ptr(), stream)
    copier.scatter_fp8(0, ids, slab8.data_ptr(), stream)
    torch.cuda.synchronize()
    src_bytes = copier.packed_bytes(0, BPF)
    t0 = time.perf_counter()
    for _ in range(ITERS):
        copier.gather_fp8(0, ids, slab8.data_ptr(), scratch.data_ptr(),
                          stream)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    rd = 2 * src_bytes * ITERS  # amax pass + quantize pass both read source
    wr = nb8 * ITERS
    print(f"fp8 gather:  {src_bytes * ITERS / dt / 1e9:.1f} GB/s logical "
          f"({(rd + wr) / dt / 1e9:.1f} GB/s r+w), "
          f"{dt / ITERS * 1e6:.0f} us/launch")
    t0 = time.perf_counter()
    for _ in range(ITERS):
        copier.scatter_fp8(0, ids, slab8.data_ptr(), stream)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"fp8 scatter: {src_bytes * ITERS / dt / 1e9:.1f} GB/s logical "
          f"({(nb8 + src_bytes) * ITERS / dt / 1e9:.1f} GB/s r+w), "
          f"{dt / ITERS * 1e6:.0f} us/launch")




def prefix_hash_probe(n_seq=4096, toks=2048, iters=20):
    """Batched prefix-hash kernel throughput (sequences hashed /s)."""
    ko = ensure_offload_native()
    tokens = torch.randint(0, 128000, (n_seq * toks,), dtype=torch.int32,
                           device="cuda")
    seq_off = torch.arange(0, (n_seq + 1) * toks, toks, dtype=torch.int64,
                           device="cuda")
    seeds = torch.full((n_seq,), 12345, dtype=torch.int64, device="cuda")
    nchunks = toks // 16
    key_off = torch.arange(0, (n_seq + 1) * nchunks, nchunks,
                           dtype=torch.int64, device="cuda")
    keys = torch.zeros(n_seq * nchunks, dtype=torch.int64, device="cuda")
    s = torch.cuda.current_stream().cuda_stream
    ko.prefix_hash(tokens.data_ptr(), seq_off.data_ptr(), seeds.data_ptr(),
                   keys.data_ptr(), key_off.data_ptr(), 16, n_seq, s)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ko.prefix_hash(tokens.data_ptr(), seq_off.data_ptr(), seeds.data_ptr(),
                       keys.data_ptr(), key_off.data_ptr(), 16, n_seq, s)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"prefix_hash: {n_seq * iters / dt:,.0f} seq/s "
          f"({n_seq * toks * iters / dt / 1e9:.2f} Gtok/s, "
          f"{n_seq}x{toks} tokens per launch)")


if __name__ == "__main__":
    main()
    prefix_hash_probe()
