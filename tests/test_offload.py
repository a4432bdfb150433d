"""Offload data plane on the host path: the full control logic (transfer
building, file layout, QoS, dedupe, cancellation) runs CPU-only; the HIP
copy path is exercised by the gpu-marked twin tests in test_offload_gpu.py.
"""
import os
import time

import pytest
import torch

from llm_d_kv_cache_amd.offload import (
    FileMapper,
    GPUToStorageHandler,
    KVCacheLayoutConfig,
    OffloadEngineConfig,
    SharedStorageOffloadManager,
    StorageToGPUHandler,
    TorchOffloadEngine,
)

NUM_BLOCKS = 64
NUM_LAYERS = 4
BLOCK_BYTES = 4096  # 16-token block, small geometry for tests
BLOCKS_PER_FILE = 8


def make_group(seed=0, num_layers=NUM_LAYERS, block_bytes=BLOCK_BYTES):
    g = torch.Generator().manual_seed(seed)
    return [
        torch.randint(0, 255, (NUM_BLOCKS, block_bytes), dtype=torch.uint8, generator=g)
        for _ in range(num_layers)
    ]


@pytest.fixture
def setup(tmp_path):
    group = make_group()
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=4, gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="host"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="test/model"))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    return group, eng, mapper, store, load


def wait_finished(handler, n=1, timeout=10.0):
    out = []
    deadline = time.time() + timeout
    while len(out) < n and time.time() < deadline:
        out.extend(handler.get_finished())
        time.sleep(0.005)
    assert len(out) >= n, f"only {len(out)} of {n} jobs finished"
    return out


def test_store_load_roundtrip(setup):
    group, eng, mapper, store, load = setup
    hashes = [0xABC123, 0xDEF456]
    ids = list(range(16))
    job = store.transfer_async(hashes, {0: ids})
    res = wait_finished(store)[0]
    assert res.success and not res.dropped
    for h in hashes:
        p = mapper.file_name(h, 0)
        assert os.path.exists(p)
        assert os.path.getsize(p) == BLOCKS_PER_FILE * NUM_LAYERS * BLOCK_BYTES

    orig = [t.clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async(hashes, {0: ids})
    res = wait_finished(load)[0]
    assert res.success
    for t, o in zip(group, orig):
        assert torch.equal(t[:16], o[:16])
        assert (t[16:] == 0).all()


def test_partial_tail_chunk(setup):
    group, eng, mapper, store, load = setup
    # 12 blocks = 1 full file + 4-block head-partial file
    hashes = [1111, 2222]
    job = store.transfer_async(hashes, {0: list(range(12))})
    assert wait_finished(store)[0].success
    assert os.path.getsize(mapper.file_name(1111, 0)) == 8 * NUM_LAYERS * BLOCK_BYTES
    assert os.path.getsize(mapper.file_name(2222, 0)) == 4 * NUM_LAYERS * BLOCK_BYTES

    orig = [t.clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async(hashes, {0: list(range(12))})
    assert wait_finished(load)[0].success
    for t, o in zip(group, orig):
        assert torch.equal(t[:12], o[:12])


def test_load_with_skip_and_slot_offset(setup):
    group, eng, mapper, store, load = setup
    hashes = [7, 8, 9]
    store.transfer_async(hashes, {0: list(range(24))})
    assert wait_finished(store)[0].success
    orig = [t.clone() for t in group]
    for t in group:
        t.zero_()
    # skip the first 10 engine blocks (1 full file + slot 2 of file 1)
    load.transfer_async(hashes, {0: list(range(10, 24))}, skip_leading_blocks=10)
    assert wait_finished(load)[0].success
    for t, o in zip(group, orig):
        assert torch.equal(t[10:24], o[10:24])
        assert (t[:10] == 0).all()


def test_dedupe_skip_existing(setup):
    group, eng, mapper, store, load = setup
    store.transfer_async([42], {0: list(range(8))})
    assert wait_finished(store)[0].success
    assert eng.stats().files_written == 1
    store.transfer_async([42], {0: list(range(8))})
    assert wait_finished(store)[0].success
    s = eng.stats()
    assert s.files_written == 1
    assert s.files_deduped == 1


def test_missing_file_load_fails(setup):
    group, eng, mapper, store, load = setup
    load.transfer_async([999999], {0: list(range(8))})
    res = wait_finished(load)[0]
    assert not res.success


def test_multi_group_hma(tmp_path):
    # two groups with different geometry (full attention + sliding window)
    g0 = make_group(seed=1, num_layers=4, block_bytes=4096)
    g1 = make_group(seed=2, num_layers=2, block_bytes=2048)
    eng = TorchOffloadEngine(
        [g0, g1],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=8, copy_path="host"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="hma"))
    store = GPUToStorageHandler(eng, mapper, [8, 4])
    load = StorageToGPUHandler(eng, mapper, [8, 4])
    h = [555]
    store.transfer_async(h, {0: list(range(8)), 1: list(range(4))})
    assert wait_finished(store)[0].success
    p0, p1 = mapper.file_name(555, 0), mapper.file_name(555, 1)
    assert os.path.exists(p0) and os.path.exists(p1)
    assert "_g0" in p0 and "_g1" in p1
    orig0 = [t.clone() for t in g0]
    orig1 = [t.clone() for t in g1]
    for t in g0 + g1:
        t.zero_()
    load.transfer_async(h, {0: list(range(8)), 1: list(range(4))})
    assert wait_finished(load)[0].success
    for t, o in zip(g0, orig0):
        assert torch.equal(t[:8], o[:8])
    for t, o in zip(g1, orig1):
        assert torch.equal(t[:4], o[:4])


def test_multi_group_skip_token_units(tmp_path):
    # Heterogeneous group block sizes: a load skip must be translated into
    # each group's own block units (skip_g = tokens // group_block_tokens[g])
    # — applying one block count to every group loads the wrong file spans.
    g0 = make_group(seed=3, num_layers=2, block_bytes=1024)  # 16-token blocks
    g1 = make_group(seed=4, num_layers=2, block_bytes=2048)  # 32-token blocks
    eng = TorchOffloadEngine(
        [g0, g1],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=8, copy_path="host"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="het"))
    store = GPUToStorageHandler(eng, mapper, [8, 4], group_block_tokens=[16, 32])
    load = StorageToGPUHandler(eng, mapper, [8, 4], group_block_tokens=[16, 32])
    hashes = [71, 72]  # 2 chunks x 128 tokens
    store.transfer_async(hashes, {0: list(range(16)), 1: list(range(8))})
    assert wait_finished(store)[0].success
    orig0 = [t.clone() for t in g0]
    orig1 = [t.clone() for t in g1]
    for t in g0 + g1:
        t.zero_()
    # skip the first 192 tokens: g0 skips 12 blocks (file 1 + slot 4),
    # g1 skips 6 blocks (file 1 + slot 2)
    load.transfer_async(hashes, {0: list(range(12, 16)), 1: list(range(6, 8))},
                        skip_leading_tokens=192)
    assert wait_finished(load)[0].success
    for t, o in zip(g0, orig0):
        assert torch.equal(t[12:16], o[12:16])
        assert (t[:12] == 0).all()
    for t, o in zip(g1, orig1):
        assert torch.equal(t[6:8], o[6:8])
        assert (t[:6] == 0).all()
    # a bare block-unit skip is ambiguous across heterogeneous groups
    with pytest.raises(ValueError):
        load.transfer_async(hashes, {0: [12], 1: [6]}, skip_leading_blocks=6)


def test_wait_job_cancels_queued(setup):
    group, eng, mapper, store, load = setup
    # flood the 4-thread pool, then cancel the last job: its queued tasks bail
    jobs = [
        store.transfer_async([10_000 + i], {0: list(range(8))}) for i in range(50)
    ]
    ok = store.wait_job(jobs[-1])
    assert isinstance(ok, bool)
    # wait for everything else
    deadline = time.time() + 10
    while eng.native.pending_writes > 0 and time.time() < deadline:
        time.sleep(0.01)
    s = eng.stats()
    assert s.files_written + s.tasks_cancelled + s.files_deduped >= 50


def test_manager_lookup_prefix(setup):
    group, eng, mapper, store, load = setup
    mgr = SharedStorageOffloadManager(mapper, num_groups=1)
    store.transfer_async([1, 2], {0: list(range(16))})
    assert wait_finished(store)[0].success
    assert mgr.lookup([1, 2, 3]) == 2
    assert mgr.lookup([3, 1, 2]) == 0  # gap at the front stops the scan
    assert mgr.prepare_store([5, 6]) == [5, 6]


def test_file_mapper_layout(tmp_path):
    cfg = KVCacheLayoutConfig(model="meta-llama/Llama-3-8B", tp_size=2, tp_rank=1)
    m = FileMapper(str(tmp_path), cfg)
    p = m.file_name(0xDEADBEEF12345678, group=2)
    assert p.startswith(str(tmp_path))
    assert "meta-llama_Llama-3-8B" in p
    assert "_r1/" in p
    assert "/dea/db_g2/deadbeef12345678.bin" in p
    # config changes -> different run dir
    m2 = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="meta-llama/Llama-3-8B",
                                                       tp_size=4, tp_rank=1))
    assert m.run_dir != m2.run_dir
    # rank does not change the config hash, only the _r suffix
    m3 = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="meta-llama/Llama-3-8B",
                                                       tp_size=2, tp_rank=0))
    assert m.run_dir.rsplit("_r", 1)[0] == m3.run_dir.rsplit("_r", 1)[0]
    cfgp = m.write_run_config()
    assert os.path.exists(cfgp)


def test_parallel_agnostic_collapses_ranks(tmp_path):
    a = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="m", tp_size=2, tp_rank=0,
                                                      parallel_agnostic=True))
    b = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="m", tp_size=4, tp_rank=3,
                                                      parallel_agnostic=True))
    assert a.run_dir == b.run_dir


def test_atime_touch_on_dedupe(setup):
    group, eng, mapper, store, load = setup
    store.transfer_async([77], {0: list(range(8))})
    assert wait_finished(store)[0].success
    p = mapper.file_name(77, 0)
    old = os.stat(p)
    os.utime(p, (time.time() - 3600, old.st_mtime))  # age the atime
    store.transfer_async([77], {0: list(range(8))})
    assert wait_finished(store)[0].success
    st = os.stat(p)
    assert st.st_atime > time.time() - 60  # refreshed
    assert abs(st.st_mtime - old.st_mtime) < 1e-3  # preserved


def test_cpu_tensors_require_host_mode():
    group = make_group()
    with pytest.raises(ValueError, match="copy_path='host'"):
        TorchOffloadEngine([group], OffloadEngineConfig(copy_path="staged"))


def test_fp8_serialize_roundtrip(tmp_path):
    """fp8 e4m3fn serialization: files are ~half size, values return within
    quantization tolerance (per-tile scaling => rel err <= ~6%)."""
    torch.manual_seed(3)
    nl = 2
    group = [
        (torch.randn(NUM_BLOCKS, BLOCK_BYTES // 2) * 4).to(torch.bfloat16)
        for _ in range(nl)
    ]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="host", serialize="fp8_e4m3"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="fp8"))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    ids = list(range(8))
    store.transfer_async([0xF8], {0: ids})
    assert wait_finished(store)[0].success
    raw_bytes = BLOCKS_PER_FILE * nl * BLOCK_BYTES
    fp8_bytes = os.path.getsize(mapper.file_name(0xF8, 0))
    assert fp8_bytes == BLOCKS_PER_FILE * nl * (BLOCK_BYTES // 2 + 4)
    assert fp8_bytes < raw_bytes * 0.51

    orig = [t.clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0xF8], {0: ids})
    assert wait_finished(load)[0].success
    for t, o in zip(group, orig):
        got = t[:8].float()
        want = o[:8].float()
        amax = want.abs().amax()
        assert (got - want).abs().max() <= 0.07 * amax
        # untouched blocks stay zero
        assert (t[8:] == 0).all()


def test_fp8_software_codec_exactness():
    """Values exactly representable in e4m3 round-trip bit-exact through
    the host-mode software codec (scale 1.0 when amax == 448)."""
    import struct

    nl = 1
    vals = [448.0, 1.0, -2.0, 0.5, 0.0, 240.0, -448.0, 0.001953125]
    data = torch.tensor(vals * (BLOCK_BYTES // 2 // len(vals)),
                        dtype=torch.bfloat16).unsqueeze(0).repeat(NUM_BLOCKS, 1)
    group = [data.clone()]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=1, gpu_blocks_per_file=4,
                            copy_path="host", serialize="fp8_e4m3"),
    )
    import tempfile

    d = tempfile.mkdtemp()
    mapper = FileMapper(d, KVCacheLayoutConfig(model="fp8exact"))
    store = GPUToStorageHandler(eng, mapper, [4])
    load = StorageToGPUHandler(eng, mapper, [4])
    store.transfer_async([1], {0: [0, 1, 2, 3]})
    assert wait_finished(store)[0].success
    orig = group[0][:4].clone()
    group[0].zero_()
    load.transfer_async([1], {0: [0, 1, 2, 3]})
    assert wait_finished(load)[0].success
    assert torch.equal(group[0][:4], orig)


def test_host_cache_tier(tmp_path):
    """Pinned-DRAM tier: write-through stores; loads hit the cache even
    after the file is gone (proof the filesystem read was skipped)."""
    group = make_group(seed=9)
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="host",
                            host_cache_bytes=8 * BLOCKS_PER_FILE * NUM_LAYERS * BLOCK_BYTES),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="hostcache"))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    ids = list(range(8))
    store.transfer_async([0xCC], {0: ids})
    assert wait_finished(store)[0].success
    assert eng.stats().host_cache_stores == 1
    path = mapper.file_name(0xCC, 0)
    assert os.path.exists(path)  # write-through
    os.unlink(path)

    orig = [t[:8].clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0xCC], {0: ids})
    assert wait_finished(load)[0].success
    assert eng.stats().host_cache_hits == 1
    for t, o in zip(group, orig):
        assert torch.equal(t[:8], o)


def test_host_cache_partial_offset_hit(tmp_path):
    group = make_group(seed=10)
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="host",
                            host_cache_bytes=4 * BLOCKS_PER_FILE * NUM_LAYERS * BLOCK_BYTES),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="hc2"))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    store.transfer_async([0xCD], {0: list(range(8))})
    assert wait_finished(store)[0].success
    os.unlink(mapper.file_name(0xCD, 0))
    orig = [t[:8].clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0xCD], {0: [5, 6, 7]}, skip_leading_blocks=5)
    assert wait_finished(load)[0].success
    assert eng.stats().host_cache_hits == 1
    for t, o in zip(group, orig):
        assert torch.equal(t[5:8], o[5:8])
        assert (t[:5] == 0).all()


def test_host_cache_eviction_falls_back_to_file(tmp_path):
    # cache holds exactly 2 slots; store 4 files; oldest 2 must come from disk
    slot = BLOCKS_PER_FILE * NUM_LAYERS * BLOCK_BYTES
    group = make_group(seed=11)
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=1, gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="host", host_cache_bytes=2 * slot),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="hc3"))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    for i in range(4):
        store.transfer_async([0xD0 + i], {0: list(range(i * 8, i * 8 + 8))})
    assert len(wait_finished(store, n=4)) == 4
    orig = [t.clone() for t in group]
    for t in group:
        t.zero_()
    for i in range(4):
        load.transfer_async([0xD0 + i], {0: list(range(i * 8, i * 8 + 8))})
    assert len(wait_finished(load, n=4)) == 4
    s = eng.stats()
    assert s.host_cache_hits >= 2  # the resident generation
    assert s.files_read + s.host_cache_hits >= 4
    for t, o in zip(group, orig):
        assert torch.equal(t[:32], o[:32])


def test_writeback_policy(tmp_path):
    """write_policy=back: the store completes once the slab is in the
    DRAM tier; the file appears asynchronously; loads hit the tier."""
    group = make_group(seed=12)
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="host", write_policy="back",
                            host_cache_bytes=8 * BLOCKS_PER_FILE * NUM_LAYERS * BLOCK_BYTES),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="wb"))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    ids = list(range(8))
    store.transfer_async([0xE1], {0: ids})
    assert wait_finished(store)[0].success
    # flush lands eventually
    path = mapper.file_name(0xE1, 0)
    deadline = time.time() + 10
    while not os.path.exists(path) and time.time() < deadline:
        time.sleep(0.01)
    assert os.path.exists(path)
    deadline = time.time() + 10
    while eng.stats().writeback_flushes < 1 and time.time() < deadline:
        time.sleep(0.01)
    assert eng.stats().writeback_flushes == 1

    orig = [t[:8].clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0xE1], {0: ids})
    assert wait_finished(load)[0].success
    assert eng.stats().host_cache_hits == 1
    for t, o in zip(group, orig):
        assert torch.equal(t[:8], o)


def test_writeback_without_cache_falls_through(tmp_path):
    """No DRAM tier -> write-back degrades to write-through gracefully."""
    group = make_group(seed=13)
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=1, gpu_blocks_per_file=4,
                            copy_path="host", write_policy="back",
                            host_cache_bytes=0),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="wb2"))
    store = GPUToStorageHandler(eng, mapper, [4])
    store.transfer_async([0xE2], {0: [0, 1, 2, 3]})
    assert wait_finished(store)[0].success
    assert os.path.exists(mapper.file_name(0xE2, 0))
    assert eng.stats().writeback_flushes == 0


def test_direct_io_fallback(tmp_path):
    """direct_io=True degrades gracefully on filesystems without O_DIRECT
    (tmp dirs here) and on unaligned fp8 records — round trips stay
    correct."""
    group = make_group(seed=14)
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="host", direct_io=True),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="direct"))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    ids = list(range(8))
    store.transfer_async([0xD10], {0: ids})
    assert wait_finished(store)[0].success
    orig = [t[:8].clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0xD10], {0: ids})
    assert wait_finished(load)[0].success
    for t, o in zip(group, orig):
        assert torch.equal(t[:8], o)


def test_mla_geometry_roundtrip(tmp_path):
    """MLA (DeepSeek-style) KV: one 576-wide latent per token — an odd,
    non-power-of-two block geometry (18 KiB tiles) through the same
    engine, proving layout-agnosticism."""
    latent = 576
    bb = 16 * latent * 2  # 18432 B per (block, layer)
    g = torch.Generator().manual_seed(21)
    group = [
        torch.randint(0, 255, (32, bb), dtype=torch.uint8, generator=g)
        for _ in range(4)
    ]
    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=8,
                                     copy_path="host"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(
        model="deepseek-v3-mla",
        kv_cache_groups=(("mla_attention", 16, bb),)))
    store = GPUToStorageHandler(eng, mapper, [8])
    load = StorageToGPUHandler(eng, mapper, [8])
    store.transfer_async([0x3A], {0: list(range(8))})
    assert wait_finished(store)[0].success
    assert os.path.getsize(mapper.file_name(0x3A, 0)) == 8 * 4 * bb
    orig = [t[:8].clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0x3A], {0: list(range(8))})
    assert wait_finished(load)[0].success
    for t, o in zip(group, orig):
        assert torch.equal(t[:8], o)


def test_fp8_with_dram_cache_and_writeback(tmp_path):
    """fp8 serialize composed with the pinned cache + write-back policy:
    the cache slot holds the fp8 payload, a hit load dequantizes from DRAM
    (file deleted to prove it), and the async flush writes the fp8 file."""
    import time

    torch.manual_seed(11)
    nl = 2
    group = [
        (torch.randn(NUM_BLOCKS, BLOCK_BYTES // 2) * 2).to(torch.bfloat16)
        for _ in range(nl)
    ]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="host", serialize="fp8_e4m3",
                            host_cache_bytes=64 << 20, write_policy="back"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="fp8wb"))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    ids = list(range(8))
    store.transfer_async([0xFB], {0: ids})
    assert wait_finished(store)[0].success
    deadline = time.time() + 10
    while eng.stats().writeback_flushes < 1 and time.time() < deadline:
        time.sleep(0.01)
    path = mapper.file_name(0xFB, 0)
    assert os.path.getsize(path) == BLOCKS_PER_FILE * nl * (BLOCK_BYTES // 2 + 4)
    os.unlink(path)  # DRAM hit must serve the load
    orig = [t[:8].float().clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0xFB], {0: ids})
    assert wait_finished(load)[0].success
    assert eng.stats().host_cache_hits >= 1
    for t, o in zip(group, orig):
        amax = o.abs().amax()
        assert (t[:8].float() - o).abs().max() <= 0.07 * amax
