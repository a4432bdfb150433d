#!/usr/bin/env python3
"""BASELINE configs[4] demonstration: Llama-3-70B 128k-context KV,
fp8-serialized blocks, Valkey-backed distributed kvblock.Index.

Wires all three pieces end to end on one GPU (or CPU with the host copy
path): a 128k-token context's KV shard (TP=8 per-rank geometry) is
offloaded as fp8 chunk files while the shared Valkey index learns the
storage tier through the KVEvents plane; a second indexer replica then
scores the full 128k prefix from the shared state.

Run: python examples/config_70b_fp8_valkey.py [--valkey host:port]
(embedded fake Valkey when no server is given)
"""
import argparse
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from llm_d_kv_cache_amd.core import (
    IndexerConfig,
    KVCacheIndexer,
    RedisIndexConfig,
    TokenProcessorConfig,
)
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.offload import (
    FileMapper,
    GPUToStorageHandler,
    KVCacheLayoutConfig,
    OffloadEngineConfig,
    StorageToGPUHandler,
    TorchOffloadEngine,
)
from llm_d_kv_cache_amd.offload.events import StorageEventPublisher

# Llama-3-70B KV shard, TP=8: 80 layers, 1 of 8 KV heads per rank
MODEL = "meta-llama/Llama-3-70B"
LAYERS, KV_HEADS, HEAD = 80, 1, 128
BLOCK_TOKENS = 16
BLOCK_BYTES = 2 * BLOCK_TOKENS * KV_HEADS * HEAD * 2  # 8 KiB
CONTEXT = 128 * 1024
BPF = 16  # 256-token chunks


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--valkey", default=None)
    ap.add_argument("--context", type=int, default=CONTEXT)
    args = ap.parse_args()

    fake = None
    if args.valkey:
        host, port = args.valkey.rsplit(":", 1)
        port = int(port)
    else:
        sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))
        from fake_redis import FakeRedis

        fake = FakeRedis()
        host, port = "127.0.0.1", fake.port
        print(f"using embedded fake valkey on :{port}")

    n_blocks = args.context // BLOCK_TOKENS
    n_chunks = n_blocks // BPF
    gpu = torch.cuda.is_available()
    dev = "cuda" if gpu else "cpu"
    print(f"{args.context} tokens -> {n_blocks} blocks x {LAYERS} layers x "
          f"{BLOCK_BYTES // 1024} KiB = "
          f"{n_blocks * LAYERS * BLOCK_BYTES / 1e9:.1f} GB KV shard ({dev})")
    group = [
        torch.randint(0, 255, (n_blocks, BLOCK_BYTES), dtype=torch.uint8,
                      device=dev)
        for _ in range(LAYERS)
    ]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=16, gpu_blocks_per_file=BPF,
                            copy_path="staged" if gpu else "host",
                            serialize="fp8_e4m3"),
    )
    root = tempfile.mkdtemp(
        dir="/dev/shm" if os.path.isdir("/dev/shm") else None)
    mapper = FileMapper(root, KVCacheLayoutConfig(
        model=MODEL, dtype="fp8_e4m3_serialized", tp_size=8, tp_rank=0))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])

    # shared Valkey index fed by storage-tier KVEvents
    idx_cfg = IndexerConfig(
        token_processor=TokenProcessorConfig(block_size_tokens=BLOCK_TOKENS),
        redis_index=RedisIndexConfig(host=host, port=port),
    )
    writer_replica = KVCacheIndexer(idx_cfg)
    pool = KVEventsPool(EventPoolConfig(zmq_endpoint="tcp://127.0.0.1:0"),
                        writer_replica)
    pool.start()
    publisher = StorageEventPublisher(
        f"tcp://127.0.0.1:{pool.port}", MODEL,
        offloaded_block_tokens=BPF * BLOCK_TOKENS, bind=False)
    time.sleep(0.3)

    # offload the whole 128k context as fp8 chunks
    tokens = list(range(args.context))
    chunk_hashes = writer_replica.compute_block_keys(tokens, MODEL)
    chunk_hashes = chunk_hashes[BPF - 1::BPF][:n_chunks]  # one key per chunk
    t0 = time.perf_counter()
    ids = list(range(n_blocks))
    jobs = 0
    for c in range(0, n_chunks, 8):
        store.transfer_async(
            chunk_hashes[c:c + 8],
            {0: ids[c * BPF:(c + 8) * BPF]})
        jobs += 1
    done = 0
    while done < jobs:
        done += len(store.get_finished())
        time.sleep(0.002)
    dt = time.perf_counter() - t0
    raw_bytes = n_blocks * LAYERS * BLOCK_BYTES
    print(f"offloaded 128k-context fp8: {raw_bytes / dt / 1e9:.1f} GB/s logical "
          f"({eng.stats().bytes_stored / 1e9:.2f} GB on disk, "
          f"{eng.stats().bytes_stored / raw_bytes * 100:.0f}% of raw)")

    publisher.publish_block_stored(chunk_hashes, tokens, None)
    deadline = time.time() + 10
    while pool.stats().processed < 1 and time.time() < deadline:
        time.sleep(0.01)

    # a SECOND replica scores the prefix from the shared Valkey state
    reader_replica = KVCacheIndexer(idx_cfg)
    scores = reader_replica.score_tokens(tokens, MODEL)
    print("replica scores over shared Valkey:", scores)
    assert scores, "expected the storage tier in the shared index"

    # load a tail slice back (tail-seek within the fp8 records)
    for t in group:
        pass
    load.transfer_async(chunk_hashes[-2:], {0: ids[-2 * BPF:]})
    while not load.get_finished():
        time.sleep(0.01)
    print("tail chunks restored from fp8")

    publisher.close()
    pool.shutdown()
    if fake:
        fake.close()
    import shutil

    shutil.rmtree(root, ignore_errors=True)
    print("ok")


if __name__ == "__main__":
    main()
