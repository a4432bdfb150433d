#!/usr/bin/env python3
"""Chaos soak: randomized store/load/cancel/dedupe traffic against the
offload engine for N seconds, verifying the engine stays healthy and
device/pinned memory stays flat (leak hunt).

Disk usage is HARD-BOUNDED (DISK_BUDGET): old generations are unlinked
synchronously once the budget is exceeded, so the soak can never exhaust
the backing filesystem (tmpfs exhaustion kills the host).

Run: python tools/chaos_soak.py [seconds]
"""
import os
import random
import sys
import tempfile
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

SECONDS = float(sys.argv[1]) if len(sys.argv) > 1 else 60.0
SERIALIZE = sys.argv[2] if len(sys.argv) > 2 else "raw"  # raw | fp8_e4m3
COPY_PATH = sys.argv[3] if len(sys.argv) > 3 else ""  # staged | zero_copy
WRITE_POLICY = sys.argv[4] if len(sys.argv) > 4 else "through"
DISK_BUDGET = 4 * 1024**3  # hard cap on bytes resident in the root
NUM_LAYERS = 16
BLOCK_BYTES = 64 * 1024
NUM_BLOCKS = 1024
BPF = 16


def mem_mb():
    if torch.cuda.is_available():
        free, total = torch.cuda.mem_get_info()
        return (total - free) / 1e6
    return 0.0


def main():
    gpu = torch.cuda.is_available()
    dev = "cuda" if gpu else "cpu"
    rng = random.Random(42)
    group = [
        torch.randint(0, 255, (NUM_BLOCKS, BLOCK_BYTES), dtype=torch.uint8,
                      device=dev)
        for _ in range(NUM_LAYERS)
    ]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(
            io_threads=8, gpu_blocks_per_file=BPF,
            copy_path=(COPY_PATH or "staged") if gpu else "host",
            host_cache_bytes=1 * 1024**3,
            serialize=SERIALIZE,
            write_policy=WRITE_POLICY,
            max_write_queued_seconds=2.0,  # provoke drops under storms
        ),
    )
    root = tempfile.mkdtemp(dir="/dev/shm" if os.path.isdir("/dev/shm") else None)
    mapper = FileMapper(root, KVCacheLayoutConfig(model="chaos"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])

    stored_hashes = []
    live_bytes = [0]
    file_bytes = {}
    outstanding = {"store": 0, "load": 0}

    def account_store(h, n):
        b = n * NUM_LAYERS * BLOCK_BYTES
        file_bytes[h] = b
        live_bytes[0] += b

    def enforce_budget():
        while live_bytes[0] > DISK_BUDGET and len(stored_hashes) > 8:
            h, _ = stored_hashes.pop(0)
            try:
                os.unlink(mapper.file_name(h, 0))
            except OSError:
                pass
            live_bytes[0] -= file_bytes.pop(h, 0)
    mem0 = mem_mb()
    t0 = time.time()
    ops = 0
    next_hash = 1
    while time.time() - t0 < SECONDS:
        r = rng.random()
        if r < 0.4 or not stored_hashes:
            n = rng.randint(1, BPF)
            base = rng.randrange(0, NUM_BLOCKS - n)
            h = next_hash
            next_hash += 1
            job = store.transfer_async([h], {0: list(range(base, base + n))})
            stored_hashes.append((h, n))
            account_store(h, n)
            enforce_budget()
            outstanding["store"] += 1
            if rng.random() < 0.1:
                store.wait_job(job)  # cancellation path
                outstanding["store"] -= 1
        elif r < 0.8:
            h, n = rng.choice(stored_hashes)
            base = rng.randrange(0, NUM_BLOCKS - n)
            job = load.transfer_async([h], {0: list(range(base, base + n))})
            outstanding["load"] += 1
            if rng.random() < 0.05:
                load.wait_job(job)
                outstanding["load"] -= 1
        elif r < 0.9:
            # duplicate store (dedupe path)
            h, n = rng.choice(stored_hashes)
            store.transfer_async([h], {0: list(range(0, n))})
            outstanding["store"] += 1
        else:
            # delete a random file behind the engine's back
            h, _ = rng.choice(stored_hashes)
            try:
                os.unlink(mapper.file_name(h, 0))
            except OSError:
                pass
        outstanding["store"] -= len(store.get_finished())
        outstanding["load"] -= len(load.get_finished())
        ops += 1
        if ops % 200 == 0:
            time.sleep(0.05)  # let the pool breathe

    # drain
    deadline = time.time() + 60
    while (outstanding["store"] > 0 or outstanding["load"] > 0) \
            and time.time() < deadline:
        outstanding["store"] -= len(store.get_finished())
        outstanding["load"] -= len(load.get_finished())
        time.sleep(0.01)
    s = eng.stats()
    mem1 = mem_mb()
    import shutil

    shutil.rmtree(root, ignore_errors=True)
    print(f"chaos: {ops} ops in {SECONDS:.0f}s | written {s.files_written} "
          f"read {s.files_read} deduped {s.files_deduped} dropped "
          f"{s.writes_dropped} cancelled {s.tasks_cancelled} errors {s.errors} "
          f"cache_hits {s.host_cache_hits}")
    print(f"device mem: {mem0:.0f} -> {mem1:.0f} MB (delta {mem1 - mem0:+.0f})")
    # errors are EXPECTED (loads of deleted files); crashes/hangs are not
    print("OK")


if __name__ == "__main__":
    main()
