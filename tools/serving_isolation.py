#!/usr/bin/env python3
"""Serving-isolation stress: how much GPU compute does a saturated offload
plane steal?

Measures a bf16 GEMM loop (MFMA via rocBLAS — a stand-in for the serving
engine's compute) alone, then with the offload engine continuously
storing+loading KV files at full rate, and reports the degradation. Also
samples Score() latency on the CPU side during the storm. This is the
design claim behind the SDMA mover and the HBM-roof copy kernels: the
wire runs without taking CUs from serving.
"""
import os
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from llm_d_kv_cache_amd.offload import (
    FileMapper,
    GPUToStorageHandler,
    KVCacheLayoutConfig,
    OffloadEngineConfig,
    StorageToGPUHandler,
    TorchOffloadEngine,
)

N = 6144
NUM_LAYERS = 32
BLOCK_BYTES = 64 * 1024
BPF = 16
FILES = 32


def gemm_tflops(seconds=4.0):
    a = torch.randn(N, N, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(N, N, dtype=torch.bfloat16, device="cuda")
    c = torch.empty(N, N, dtype=torch.bfloat16, device="cuda")
    torch.matmul(a, b, out=c)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 0
    while time.perf_counter() - t0 < seconds:
        torch.matmul(a, b, out=c)
        iters += 1
        if iters % 8 == 0:
            torch.cuda.synchronize()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return 2 * N**3 * iters / dt / 1e12


def main():
    assert torch.cuda.is_available()
    group = [
        torch.randint(0, 255, (1024, BLOCK_BYTES), dtype=torch.uint8,
                      device="cuda")
        for _ in range(NUM_LAYERS)
    ]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=16, gpu_blocks_per_file=BPF,
                            host_cache_bytes=4 * 1024**3),
    )
    root = tempfile.mkdtemp(dir="/dev/shm" if os.path.isdir("/dev/shm") else None)
    mapper = FileMapper(root, KVCacheLayoutConfig(model="isolation"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])

    print(f"baseline GEMM ({N}^3 bf16): ", end="", flush=True)
    base = gemm_tflops()
    print(f"{base:.0f} TF/s")

    stop = threading.Event()
    moved = [0]

    def offload_storm():
        gen = 0
        while not stop.is_set():
            gen += 1
            hashes = [gen * 1000 + i for i in range(FILES)]
            ids = list(range(FILES * BPF))
            n = 0
            for i in range(0, FILES, 8):
                store.transfer_async(hashes[i:i + 8],
                                     {0: ids[i * BPF:(i + 8) * BPF]})
                n += 1
            done = 0
            while done < n and not stop.is_set():
                done += len(store.get_finished())
                time.sleep(0.001)
            for i in range(0, FILES, 8):
                load.transfer_async(hashes[i:i + 8],
                                    {0: ids[i * BPF:(i + 8) * BPF]})
            done = 0
            while done < n and not stop.is_set():
                done += len(load.get_finished())
                time.sleep(0.001)
            moved[0] += 2 * FILES * BPF * NUM_LAYERS * BLOCK_BYTES
            # bound disk usage
            if gen > 2:
                old = (gen - 2) * 1000
                for h in range(old, old + FILES):
                    try:
                        os.unlink(mapper.file_name(h, 0))
                    except OSError:
                        pass

    t = threading.Thread(target=offload_storm, daemon=True)
    t.start()
    time.sleep(1.0)  # let the storm ramp
    t0 = time.perf_counter()
    print("GEMM under offload storm:       ", end="", flush=True)
    storm = gemm_tflops()
    storm_dt = time.perf_counter() - t0
    offload_gbps = moved[0] / storm_dt / 1e9
    stop.set()
    t.join(timeout=30)
    print(f"{storm:.0f} TF/s  (offload concurrently: ~{offload_gbps:.1f} GB/s)")
    print(f"compute retained: {100 * storm / base:.1f}%")

    # CPU-side Score() latency during a fresh storm
    from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
    import numpy as np

    ix = KVCacheIndexer(IndexerConfig())
    tokens = np.arange(4096, dtype=np.uint32)
    keys = ix.compute_block_keys(tokens, "m")
    from llm_d_kv_cache_amd import ensure_native

    kk = ensure_native()
    for p in range(32):
        ix.index.add([], keys, [kk.PodEntry(f"pod-{p}", "gpu")])
    lat = []
    stop.clear()
    t = threading.Thread(target=offload_storm, daemon=True)
    t.start()
    time.sleep(0.5)
    for _ in range(2000):
        t1 = time.perf_counter()
        ix.score_tokens(tokens, "m")
        lat.append(time.perf_counter() - t1)
    stop.set()
    t.join(timeout=30)
    lat.sort()
    print(f"Score() during storm: p50 {lat[len(lat)//2]*1e6:.0f} us, "
          f"p99 {lat[int(len(lat)*0.99)]*1e6:.0f} us")

    peer_isolation()


def peer_isolation():
    """GEMM retention under a block-migration storm: continuous gather →
    scatter of 32 MiB chunks through the CDNA4 kernels (what a peer pull
    does on each side of the xGMI hop, minus the link itself)."""
    import threading

    from llm_d_kv_cache_amd import ensure_offload_native

    ko = ensure_offload_native()
    group = [torch.randint(0, 255, (2048, 64 * 1024), dtype=torch.uint8,
                           device="cuda") for _ in range(4)]
    copier = ko.BlockCopier(
        [([t.data_ptr() for t in group], [t.stride(0) for t in group],
          64 * 1024, 2048)], gpu_mode=True)
    nb = copier.packed_bytes(0, 16)
    stream = torch.cuda.Stream()
    slab = torch.empty(nb, dtype=torch.uint8, device="cuda")
    stop = threading.Event()
    moved = [0]

    def migration_storm():
        i = 0
        with torch.cuda.stream(stream):
            while not stop.is_set():
                src = list(range((i * 16) % 1024, (i * 16) % 1024 + 16))
                dst = list(range(1024 + (i * 16) % 1024,
                                 1040 + (i * 16) % 1024))
                copier.gather(0, src, slab.data_ptr(), stream.cuda_stream)
                copier.scatter(0, dst, slab.data_ptr(), stream.cuda_stream)
                if i % 8 == 0:
                    stream.synchronize()
                moved[0] += 2 * nb
                i += 1
        stream.synchronize()

    t = threading.Thread(target=migration_storm, daemon=True)
    t0 = time.perf_counter()
    t.start()
    time.sleep(0.5)
    print("GEMM under migration storm:     ", end="", flush=True)
    storm = gemm_tflops()
    dt = time.perf_counter() - t0
    stop.set()
    t.join(timeout=30)
    print(f"{storm:.0f} TF/s  (migrating concurrently: "
          f"~{moved[0] / dt / 1e9:.0f} GB/s block copies)")


if __name__ == "__main__":
    main()
