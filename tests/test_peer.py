"""Peer KV-block migration: multi-process (gloo, world_size=2) CPU tests.

The same event-loop/protocol code drives RCCL over xGMI on a GPU node;
here the data plane is gloo and packing is memcpy. GPU twin lives in
test_peer_gpu.py.
"""
import multiprocessing as mp
import os

import pytest
import torch


def _run_rank(rank, world, init_file, fn_name, q):
    try:
        import torch.distributed as dist

        dist.init_process_group(
            "gloo", init_method=f"file://{init_file}", rank=rank,
            world_size=world,
        )
        from llm_d_kv_cache_amd.peer import PeerMigrationService

        # dedicated groups: the service thread must never share a group
        # with main-thread collectives
        ctrl_pg = dist.new_group(backend="gloo")
        data_pg = dist.new_group(backend="gloo")
        torch.manual_seed(100 + rank)
        group = [torch.randint(0, 255, (32, 4096), dtype=torch.uint8)
                 for _ in range(2)]
        group_b = [torch.randint(0, 255, (32, 2048), dtype=torch.uint8)]
        svc = PeerMigrationService([group, group_b], data_group=data_pg,
                                   control_group=ctrl_pg)
        fn = globals()[fn_name]
        fn(rank, svc, group)
        dist.barrier()
        svc.close()
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def spawn_world(fn_name, tmp_path, world):
    init_file = str(tmp_path / "pg_init")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_rank, args=(r, world, init_file, fn_name, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status = q.get(timeout=240)
        results[rank] = status
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    assert all(v == "ok" for v in results.values()), results


def spawn2(fn_name, tmp_path):
    spawn_world(fn_name, tmp_path, 2)


# ---- scenarios (run inside worker processes) --------------------------------

def scenario_basic_pull(rank, svc, group):
    import torch.distributed as dist

    CHUNK = 0xC0FFEE
    if rank == 0:
        svc.register_blocks(CHUNK, 0, [3, 5, 7, 9])
        golden = torch.cat([group[0][[3, 5, 7, 9]].reshape(4, 1, -1),
                            group[1][[3, 5, 7, 9]].reshape(4, 1, -1)], dim=1)
        # publish golden bytes for rank 1 to verify
        dist.broadcast(golden.contiguous(), src=0)
        dist.barrier()
    else:
        golden = torch.zeros(4, 2, 4096, dtype=torch.uint8)
        dist.broadcast(golden, src=0)
        ok = svc.pull(CHUNK, 0, [10, 11, 12, 13], src_rank=0).result(timeout=60)
        assert ok is True
        got = torch.stack([
            torch.stack([group[0][10 + i], group[1][10 + i]]) for i in range(4)
        ])
        assert torch.equal(got, golden)
        dist.barrier()


def scenario_missing_chunk(rank, svc, group):
    import torch.distributed as dist

    if rank == 1:
        ok = svc.pull(0xDEAD, 0, [1], src_rank=0).result(timeout=60)
        assert ok is False
        assert svc.stats().pulls_failed == 1
    dist.barrier()


def scenario_bidirectional(rank, svc, group):
    """Both ranks pull from each other at the same time: the non-blocking
    loop must not deadlock."""
    import torch.distributed as dist

    CHUNK = 0xAB00 + rank
    svc.register_blocks(CHUNK, 0, [0, 1])
    dist.barrier()
    other = 1 - rank
    futs = [svc.pull(0xAB00 + other, 0, [20 + 2 * i, 21 + 2 * i], src_rank=other)
            for i in range(3)]
    assert all(f.result(timeout=60) for f in futs)
    assert svc.stats().pulls_served >= 1
    dist.barrier()


def scenario_batched_pull(rank, svc, group):
    """pull_many: one control round trip + one data transfer for several
    chunks across groups, with partial grants (missing chunk -> False)."""
    import torch.distributed as dist

    if rank == 0:
        svc.register_blocks(0x21, 0, [1, 2])
        svc.register_blocks(0x22, 1, [5])
        svc.register_blocks(0x23, 0, [9])
        golden = [group[0][[1, 2]].clone(), group[1][[1, 2]].clone()]
        dist.broadcast(golden[0], src=0)
        dist.broadcast(golden[1], src=0)
        dist.barrier()
    else:
        golden = [torch.zeros(2, 4096, dtype=torch.uint8) for _ in range(2)]
        dist.broadcast(golden[0], src=0)
        dist.broadcast(golden[1], src=0)
        res = svc.pull_many(
            [(0x21, 0, [14, 15]),      # granted
             (0x9999, 0, [16]),        # missing -> False
             (0x23, 0, [17])],         # granted
            src_rank=0).result(timeout=60)
        assert res == [True, False, True]
        assert torch.equal(group[0][14], golden[0][0])
        assert torch.equal(group[0][15], golden[0][1])
        assert torch.equal(group[1][14], golden[1][0])
        assert torch.equal(group[1][15], golden[1][1])
        # all-missing batch resolves all-False without a data transfer
        res2 = svc.pull_many([(0x77, 0, [18]), (0x78, 1, [3])],
                             src_rank=0).result(timeout=60)
        assert res2 == [False, False]
        dist.barrier()


def scenario_fp8_wire(rank, svc, group):
    """fp8-on-the-wire pulls of HBM-resident blocks: requester opts in,
    peer quantizes on gather, half the data-plane bytes, e4m3 tolerance."""
    import torch.distributed as dist

    torch.manual_seed(300 + rank)
    group_b = svc._tensors[1]
    vals = (torch.randn(32, 1024) * 2).to(torch.bfloat16)
    group_b[0].copy_(vals.view(torch.uint8).reshape(32, 2048))
    svc.register_blocks(0x41 + rank, 1, [0, 1])
    golden = vals[[0, 1]].float()
    peer_golden = [torch.zeros(2, 1024) for _ in range(2)]
    for r in range(2):
        src = golden.clone() if r == rank else peer_golden[r]
        dist.broadcast(src, src=r)
        peer_golden[r] = src
    other = 1 - rank
    before = svc.stats().bytes_received
    ok = svc.pull(0x41 + other, 1, [10, 11], src_rank=other,
                  fp8=True).result(timeout=60)
    assert ok is True
    moved = svc.stats().bytes_received - before
    assert moved == svc._copier.packed_bytes_fp8(1, 2)  # half + scales
    got = group_b[0][[10, 11]].view(torch.bfloat16).float()
    want = peer_golden[other]
    assert (got - want).abs().max() <= 0.07 * want.abs().amax()
    # batched fp8 + a miss
    res = svc.pull_many([(0x41 + other, 1, [14, 15]), (0x9E9E, 1, [16])],
                        src_rank=other, fp8=True).result(timeout=60)
    assert res == [True, False]
    got2 = group_b[0][[14, 15]].view(torch.bfloat16).float()
    assert (got2 - want).abs().max() <= 0.07 * want.abs().amax()
    dist.barrier()


def scenario_self_pull(rank, svc, group):
    """src_rank == own rank takes the local short circuit (RCCL cannot
    send to self): HBM hit, miss, and batched variants."""
    import torch.distributed as dist

    svc.register_blocks(0x31 + rank, 0, [1, 2])
    golden = [group[0][[1, 2]].clone(), group[1][[1, 2]].clone()]
    ok = svc.pull(0x31 + rank, 0, [28, 29], src_rank=rank).result(timeout=60)
    assert ok is True
    assert torch.equal(group[0][[28, 29]], golden[0])
    assert torch.equal(group[1][[28, 29]], golden[1])
    assert svc.pull(0x777, 0, [30], src_rank=rank).result(timeout=60) is False
    res = svc.pull_many([(0x31 + rank, 0, [24, 25]), (0x778, 0, [26])],
                        src_rank=rank).result(timeout=60)
    assert res == [True, False]
    dist.barrier()


def scenario_invalid_pull_sizes(rank, svc, group):
    import pytest as _pytest
    import torch.distributed as dist

    with _pytest.raises(ValueError):
        svc.pull(0x1, 0, [], src_rank=0)
    with _pytest.raises(ValueError):
        svc.pull(0x1, 0, list(range(65)), src_rank=0)
    with _pytest.raises(ValueError):
        svc.pull_many([(0x1, 0, list(range(65)))], src_rank=0)
    dist.barrier()


def scenario_multi_group(rank, svc, group):
    import torch.distributed as dist

    if rank == 0:
        svc.register_blocks(0x11, 1, [2, 4])
    dist.barrier()
    if rank == 1:
        ok = svc.pull(0x11, 1, [6, 8], src_rank=0).result(timeout=60)
        assert ok is True
    dist.barrier()


def scenario_dram_tier(rank, svc, group):
    """Peer serving from the pinned host-DRAM cache: rank 0 offloads chunks
    (write-through populates the cache), never registers them in HBM, and
    deletes the files — rank 1's pulls are served from DRAM over the data
    plane, including an fp8-serialized group in a mixed batched pull."""
    import os as _os
    import tempfile
    import time

    import torch.distributed as dist

    from llm_d_kv_cache_amd.offload import (
        FileMapper,
        GPUToStorageHandler,
        KVCacheLayoutConfig,
        OffloadEngineConfig,
        TorchOffloadEngine,
    )
    from llm_d_kv_cache_amd.peer.tiered import make_dram_lookup

    # make group_b rows valid bf16 payloads for the fp8 codec
    torch.manual_seed(900 + rank)
    group_b = svc._tensors[1]
    vals = (torch.randn(32, 1024) * 2).to(torch.bfloat16)
    group_b[0].copy_(vals.view(torch.uint8).reshape(32, 2048))

    CH_RAW, CH_FP8, CH_MISS = 0x61, 0x62, 0xBAD
    if rank == 0:
        root = tempfile.mkdtemp(prefix="dramtier0_")
        groups = [svc._tensors[0], group_b]
        eng_raw = TorchOffloadEngine(groups, OffloadEngineConfig(
            io_threads=2, gpu_blocks_per_file=4, copy_path="host",
            host_cache_bytes=64 << 20))
        eng_fp8 = TorchOffloadEngine(groups, OffloadEngineConfig(
            io_threads=2, gpu_blocks_per_file=4, copy_path="host",
            host_cache_bytes=64 << 20, serialize="fp8_e4m3"))
        m_raw = FileMapper(root + "/raw", KVCacheLayoutConfig(model="dt"))
        m_fp8 = FileMapper(root + "/fp8", KVCacheLayoutConfig(model="dt8"))
        st_raw = GPUToStorageHandler(eng_raw, m_raw, [4, 4])
        st_fp8 = GPUToStorageHandler(eng_fp8, m_fp8, [4, 4])
        st_raw.transfer_async([CH_RAW], {0: [0, 1, 2, 3]})
        st_fp8.transfer_async([CH_FP8], {1: [0, 1, 2, 3]})
        for h in (st_raw, st_fp8):
            deadline = time.time() + 10
            while not h.get_finished() and time.time() < deadline:
                time.sleep(0.01)
        # prove DRAM (not fs) serves: remove the files
        _os.remove(m_raw.file_name(CH_RAW, 0))
        _os.remove(m_fp8.file_name(CH_FP8, 1))
        lk_raw = make_dram_lookup(eng_raw, m_raw)
        lk_fp8 = make_dram_lookup(eng_fp8, m_fp8)
        svc._dram_lookup = (lambda h, g, n:
                            lk_raw(h, g, n) if g == 0 else lk_fp8(h, g, n))
        golden = svc._tensors[0][0][[0, 1, 2, 3]].clone()
        dist.broadcast(golden, src=0)
        golden_b = vals[[0, 1, 2, 3]].float()
        dist.broadcast(golden_b, src=0)
        dist.barrier()
        dist.barrier()
        assert svc.stats().pulls_served_dram >= 3
    else:
        golden = torch.zeros(4, 4096, dtype=torch.uint8)
        dist.broadcast(golden, src=0)
        golden_b = torch.zeros(4, 1024)
        dist.broadcast(golden_b, src=0)
        # single pull from DRAM
        ok = svc.pull(CH_RAW, 0, [10, 11, 12, 13], src_rank=0).result(
            timeout=60)
        assert ok is True
        assert torch.equal(svc._tensors[0][0][[10, 11, 12, 13]], golden)
        # mixed batched pull: raw DRAM + fp8 DRAM + miss
        res = svc.pull_many(
            [(CH_RAW, 0, [20, 21, 22, 23]),
             (CH_FP8, 1, [8, 9, 10, 11]),
             (CH_MISS, 0, [24])], src_rank=0).result(timeout=60)
        assert res == [True, True, False]
        assert torch.equal(svc._tensors[0][0][[20, 21, 22, 23]], golden)
        got = group_b[0][[8, 9, 10, 11]].view(torch.bfloat16).float()
        amax = golden_b.abs().amax()
        assert (got - golden_b).abs().max() <= 0.07 * amax
        dist.barrier()
        dist.barrier()


def scenario_index_driven_peer_resolution(rank, svc, group):
    """The full serving-path flow (SURVEY §2.5 'third StorageHandler-like
    path'): KVEvents register a peer pod's blocks under the peer-gpu tier,
    the scorer picks that pod, a pod->rank map drives the TieredKVLoader's
    pull over the data plane (bit-compared), and chunks no longer
    registered on the peer fall back to the storage tier or miss."""
    import tempfile
    import time

    import torch.distributed as dist

    from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
    from llm_d_kv_cache_amd.events.publisher import (
        block_stored_payload,
        encode_batch,
    )
    from llm_d_kv_cache_amd.offload import (
        FileMapper,
        GPUToStorageHandler,
        KVCacheLayoutConfig,
        OffloadEngineConfig,
        StorageToGPUHandler,
        TorchOffloadEngine,
    )
    from llm_d_kv_cache_amd.peer.tiered import TieredKVLoader

    MODEL = "m"
    tokens = list(range(64))  # 4 hash blocks of 16 tokens = 1 offload chunk
    POD_RANK = {"pod-r0": 0, "pod-r1": 1}

    ix = KVCacheIndexer(IndexerConfig())
    keys = ix.compute_block_keys(tokens, MODEL)
    chunk_hash = keys[-1]  # chunk addressed by its last covered block key

    if rank == 0:
        # the peer holds the chunk's blocks in HBM (data plane registry)
        svc.register_blocks(chunk_hash, 0, [0, 1, 2, 3])
        golden = [group[0][[0, 1, 2, 3]].clone(), group[1][[0, 1, 2, 3]].clone()]
        dist.broadcast(golden[0], src=0)
        dist.broadcast(golden[1], src=0)
        dist.barrier()   # rank 1 resolves
        dist.barrier()   # done
        assert svc.stats().pulls_served >= 1
        return

    golden = [torch.zeros(4, 4096, dtype=torch.uint8),
              torch.zeros(4, 4096, dtype=torch.uint8)]  # two layers of group 0
    dist.broadcast(golden[0], src=0)
    dist.broadcast(golden[1], src=0)

    # control plane: rank 0's engine announced the stores as KVEvents with
    # the peer-gpu tier; this replica's pool ingests them
    from llm_d_kv_cache_amd import ensure_native

    k = ensure_native()
    pool = k.EventPool(ix.token_processor, ix.index, 1)
    pool.process(f"kv@pod-r0@{MODEL}", 0, encode_batch([
        block_stored_payload(list(keys), None, tokens, 16, medium="peer-gpu")
    ]))
    scores = ix.score_tokens(tokens, MODEL)
    assert scores == {"pod-r0": pytest.approx(4 * 0.95)}
    best_pod = max(scores, key=scores.get)
    assert POD_RANK[best_pod] == 0

    # local storage tier for the fallback chunk
    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=4,
                                     copy_path="host"))
    root = tempfile.mkdtemp(prefix="ixpeer_")
    mapper = FileMapper(root, KVCacheLayoutConfig(model="ixpeer"))
    store = GPUToStorageHandler(eng, mapper, [4])
    load = StorageToGPUHandler(eng, mapper, [4])
    CH_LOCAL, CH_GONE = 0x71, 0x72
    store.transfer_async([CH_LOCAL], {0: [8, 9, 10, 11]})
    deadline = time.time() + 10
    while not store.get_finished() and time.time() < deadline:
        time.sleep(0.01)

    loader = TieredKVLoader(load_handler=load, peer_service=svc,
                            peer_ranks=[POD_RANK[best_pod]])
    # scored chunk: pulled from the peer's HBM over the data plane
    assert loader.resolve(chunk_hash, [16, 17, 18, 19]) == "peer"
    assert torch.equal(group[0][[16, 17, 18, 19]], golden[0])
    assert torch.equal(group[1][[16, 17, 18, 19]], golden[1])
    # chunk the peer never registered: served by the local storage tier
    assert loader.resolve(CH_LOCAL, [20, 21, 22, 23]) == "storage"
    # nowhere: miss -> the caller recomputes
    assert loader.resolve(CH_GONE, [24]) == "miss"
    dist.barrier()
    dist.barrier()


def scenario_mesh_pull_world4(rank, svc, group):
    """Full-mesh pulls at world 4 (the 8-GPU node's topology in miniature):
    every rank registers its own chunk and pulls every other rank's,
    concurrently — no deadlock, every payload bit-exact."""
    import torch.distributed as dist

    world = dist.get_world_size()
    CH = 0x4000
    svc.register_blocks(CH + rank, 0, [0, 1, 2, 3])
    # publish every rank's golden rows
    goldens = []
    for r in range(world):
        g = (group[0][[0, 1, 2, 3]].clone() if r == rank
             else torch.zeros(4, 4096, dtype=torch.uint8))
        dist.broadcast(g, src=r)
        goldens.append(g)
    dist.barrier()
    futs = {}
    for i, src in enumerate([r for r in range(world) if r != rank]):
        dst = [8 + 4 * i + j for j in range(4)]
        futs[src] = (dst, svc.pull(CH + src, 0, dst, src_rank=src,
                                   timeout=120))
    for src, (dst, f) in futs.items():
        assert f.result(timeout=150) is True, f"pull from {src} failed"
        assert torch.equal(group[0][dst], goldens[src])
    assert svc.stats().pulls_served >= 1
    dist.barrier()


# ---- tests ------------------------------------------------------------------

@pytest.mark.parametrize("scenario", [
    "scenario_basic_pull",
    "scenario_missing_chunk",
    "scenario_bidirectional",
    "scenario_multi_group",
    "scenario_batched_pull",
    "scenario_dram_tier",
    "scenario_self_pull",
    "scenario_fp8_wire",
    "scenario_invalid_pull_sizes",
    "scenario_cmd_error_resolves_future",
    "scenario_tiered_loader",
    "scenario_index_driven_peer_resolution",
])
def test_peer_migration(scenario, tmp_path):
    spawn2(scenario, tmp_path)


@pytest.mark.timeout(300)
def test_peer_mesh_world4(tmp_path):
    spawn_world("scenario_mesh_pull_world4", tmp_path, 4)


def test_block_copier_roundtrip():
    from llm_d_kv_cache_amd import _kvoffload as ko

    g = [torch.randint(0, 255, (16, 1024), dtype=torch.uint8) for _ in range(3)]
    copier = ko.BlockCopier(
        [([t.data_ptr() for t in g], [t.stride(0) for t in g], 1024)],
        gpu_mode=False,
    )
    packed = torch.zeros(copier.packed_bytes(0, 4), dtype=torch.uint8)
    copier.gather(0, [1, 3, 5, 7], packed.data_ptr(), 0)
    view = packed.view(4, 3, 1024)
    for bi, b in enumerate([1, 3, 5, 7]):
        for l in range(3):
            assert torch.equal(view[bi, l], g[l][b])
    g2 = [torch.zeros_like(t) for t in g]
    copier2 = ko.BlockCopier(
        [([t.data_ptr() for t in g2], [t.stride(0) for t in g2], 1024)],
        gpu_mode=False,
    )
    copier2.scatter(0, [0, 2, 4, 6], packed.data_ptr(), 0)
    for bi, b in enumerate([0, 2, 4, 6]):
        for l in range(3):
            assert torch.equal(g2[l][b], view[bi, l])


def scenario_tiered_loader(rank, svc, group):
    """storage -> peer -> miss resolution across the tier hierarchy."""
    import tempfile

    import torch.distributed as dist

    from llm_d_kv_cache_amd.offload import (
        FileMapper,
        GPUToStorageHandler,
        KVCacheLayoutConfig,
        OffloadEngineConfig,
        StorageToGPUHandler,
        TorchOffloadEngine,
    )
    from llm_d_kv_cache_amd.peer.tiered import TieredKVLoader

    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=4,
                                     copy_path="host"),
    )
    root = tempfile.mkdtemp(prefix=f"tiered_r{rank}_")
    mapper = FileMapper(root, KVCacheLayoutConfig(model="tiered"))
    store = GPUToStorageHandler(eng, mapper, [4])
    load = StorageToGPUHandler(eng, mapper, [4])

    CH_STORAGE, CH_PEER, CH_MISS = 0x51, 0x52, 0x53
    if rank == 0:
        # rank 0 holds CH_PEER in "HBM" (registry) only
        svc.register_blocks(CH_PEER, 0, [0, 1, 2, 3])
    else:
        # rank 1 has CH_STORAGE on its local filesystem
        store.transfer_async([CH_STORAGE], {0: [0, 1, 2, 3]})
        import time

        deadline = time.time() + 10
        while not store.get_finished() and time.time() < deadline:
            time.sleep(0.01)
    dist.barrier()

    if rank == 1:
        loader = TieredKVLoader(load_handler=load, peer_service=svc,
                                peer_ranks=[0])
        assert loader.resolve(CH_STORAGE, [8, 9, 10, 11]) == "storage"
        assert loader.resolve(CH_PEER, [12, 13, 14, 15]) == "peer"
        assert loader.resolve(CH_MISS, [16, 17, 18, 19]) == "miss"
        # prefix semantics: stop at the first gap
        filled = loader.resolve_prefix([CH_STORAGE, CH_MISS, CH_PEER],
                                       list(range(20, 32)), 4)
        assert filled == 1
        # mixed tiers resolve across one batched peer attempt:
        # storage, then a peer run, then storage again, then miss
        filled = loader.resolve_prefix(
            [CH_STORAGE, CH_PEER, CH_STORAGE, CH_MISS],
            list(range(8, 24)), 4)
        assert filled == 3
    dist.barrier()


def scenario_cmd_error_resolves_future(rank, svc, group):
    """A command that blows up inside the service thread must resolve its
    future with the exception (regression: the future moved to payload[-2]
    when the fp8 flag was added)."""
    import torch.distributed as dist

    import concurrent.futures
    import time as _time

    if rank == 0:
        # an invalid destination rank makes dist.send raise inside the
        # service thread; the loop's error isolation must resolve the
        # future instead of dying
        fut = concurrent.futures.Future()
        with svc._q_mu:
            svc._cmd_q.append(("pull", 0x1, 0, [0], 99,
                               _time.time() + 30, fut, False))
        try:
            fut.result(timeout=30)
            raise AssertionError("expected an exception")
        except AssertionError:
            raise
        except Exception:
            pass  # surfaced correctly
        # the service loop survived: a normal miss still resolves
        assert svc.pull(0x1, 0, [0], src_rank=1 - rank).result(
            timeout=30) is False
    dist.barrier()
