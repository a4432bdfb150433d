"""Disk-management loop under LIVE offload churn: engine stores files,
the PVC evictor enforces the disk budget and publishes BlockRemoved, the
events pool evicts the index — the complete
store → announce → evict → un-route cycle across four components
(reference counterpart: the pvc_evictor N+2 design + storage events; here
wired end to end over real ZMTP instead of per-process unit tests)."""
import os
import time

import pytest
import torch

from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.offload import (
    FileMapper,
    GPUToStorageHandler,
    KVCacheLayoutConfig,
    OffloadEngineConfig,
    StorageToGPUHandler,
    TorchOffloadEngine,
)
from llm_d_kv_cache_amd.offload.events import StorageEventPublisher

k = ensure_native()

MODEL = "loop-model"
BPF = 4
BLOCK_TOKENS = 16


def count_bins(root):
    n = 0
    for dirpath, _dirs, files in os.walk(root):
        n += sum(1 for f in files if f.endswith(".bin"))
    return n


class FileBudgetUtilization:
    """Picklable utilization callback: 'disk pressure' = file count over a
    budget (the activator runs in a spawned process)."""

    def __init__(self, root, budget):
        self.root = str(root)
        self.budget = budget

    def __call__(self, _root):
        return 1.0 if count_bins(self.root) > self.budget else 0.0


@pytest.mark.timeout(180)
def test_offload_churn_with_evictor_closes_the_index_loop(tmp_path):
    from llm_d_kv_cache_amd.evictor import EvictorConfig, PvcEvictor

    # --- control plane: index + pool fed by a bound ZMTP SUB ---------------
    ix = KVCacheIndexer(IndexerConfig())
    pool = k.EventPool(ix.token_processor, ix.index, 1)
    pool.start()
    sub = k.Subscriber("tcp://127.0.0.1:0", "",
                       callback=lambda t, s, p: pool.add_task(t, s, p),
                       bind=True)
    ep = f"tcp://127.0.0.1:{sub.port}"

    # --- data plane: host-mode engine churning generations -----------------
    group = [torch.randint(0, 255, (64, 2048), dtype=torch.uint8)
             for _ in range(2)]
    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BPF,
                                     copy_path="host"))
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model=MODEL))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])
    announcer = StorageEventPublisher(ep, MODEL,
                                      offloaded_block_tokens=BPF * BLOCK_TOKENS)
    time.sleep(0.3)  # let the PUB/SUB handshake land

    def store_chunk(i, parent):
        h = 0x100 + i
        store.transfer_async([h], {0: list(range(BPF))})
        deadline = time.time() + 10
        while not store.get_finished() and time.time() < deadline:
            time.sleep(0.005)
        tokens = list(range(i * BPF * BLOCK_TOKENS,
                            (i + 1) * BPF * BLOCK_TOKENS))
        announcer.publish_block_stored([h], tokens,
                                       parent_chunk_hash=parent)
        return h

    # a prefix of chunks, announced as a chain
    hashes = []
    parent = None
    for i in range(12):
        parent = store_chunk(i, parent)
        hashes.append(parent)
    tokens_all = list(range(12 * BPF * BLOCK_TOKENS))
    deadline = time.time() + 15
    while time.time() < deadline:
        pool.drain()
        if ix.score_tokens(tokens_all, MODEL).get("SHARED_STORAGE"):
            break
        time.sleep(0.05)
    scores = ix.score_tokens(tokens_all, MODEL)
    assert scores.get("SHARED_STORAGE", 0) > 0, scores

    # --- evictor: budget = 8 files; everything colder than 1 s is fair game
    cfg = EvictorConfig(root=str(tmp_path), crawlers=1,
                        atime_threshold_s=1.0, crawl_interval_s=0.1,
                        check_interval_s=0.05, delete_batch=8,
                        events_endpoint=ep, events_model=MODEL)
    budget = 8
    ev = PvcEvictor(cfg, utilization=FileBudgetUtilization(tmp_path, budget))
    ev.start()
    try:
        # live churn: keep storing fresh generations while touching the
        # newest one (loads refresh atime -> the warm tail survives)
        newest = None
        for i in range(12, 30):
            newest = store_chunk(i, None)
            load.transfer_async([newest], {0: list(range(BPF))})
            deadline = time.time() + 10
            while not load.get_finished() and time.time() < deadline:
                time.sleep(0.005)
            time.sleep(0.1)

        # the budget holds under churn (with slack for in-flight deletes)
        deadline = time.time() + 30
        while count_bins(tmp_path) > budget and time.time() < deadline:
            time.sleep(0.2)
        assert count_bins(tmp_path) <= budget + 4, \
            f"{count_bins(tmp_path)} files left against a budget of {budget}"
        assert ev.deleted.value > 0

        # BlockRemoved flowed back: the early chain is un-routed
        deadline = time.time() + 20
        while time.time() < deadline:
            pool.drain()
            early = ix.score_tokens(tokens_all[: 4 * BPF * BLOCK_TOKENS],
                                    MODEL)
            if not early:
                break
            time.sleep(0.2)
        assert not early, f"evicted chunks still routed: {early}"
        assert ix.index.stats().evictions > 0
    finally:
        ev.shutdown()
        announcer.close()
        sub.close()
        pool.shutdown()
