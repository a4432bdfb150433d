"""GPU twin of test_offload.py: the HIP copy paths, bit-exact, on MI355X.

Numerics contract: bytes written/read by the CDNA4 gather/scatter kernels
must equal a plain PyTorch copy of the same blocks (fp32-safe because the
payload is opaque bytes — exactness is required, not tolerance).
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
    StorageToGPUHandler,
    TorchOffloadEngine,
)

pytestmark = pytest.mark.gpu

NUM_BLOCKS = 128
NUM_LAYERS = 8
BLOCK_BYTES = 64 * 1024  # Llama-3-8B geometry: 16 tok x 8 kv-heads x 128 x 2 x bf16
BPF = 16


def make_group(num_layers=NUM_LAYERS, block_bytes=BLOCK_BYTES):
    return [
        torch.randint(0, 255, (NUM_BLOCKS, block_bytes), dtype=torch.uint8,
                      device="cuda")
        for _ in range(num_layers)
    ]


def wait_finished(handler, n=1, timeout=30.0):
    out = []
    deadline = time.time() + timeout
    while len(out) < n and time.time() < deadline:
        out.extend(handler.get_finished())
        time.sleep(0.002)
    assert len(out) >= n
    return out


@pytest.mark.parametrize("copy_path", ["staged", "zero_copy"])
def test_gpu_roundtrip_bit_exact(tmp_path, copy_path):
    group = make_group()
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=4, gpu_blocks_per_file=BPF,
                            copy_path=copy_path),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model=f"gpu-{copy_path}"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])

    ids = list(range(32))
    hashes = [0xA1, 0xA2]
    store.transfer_async(hashes, {0: ids})
    assert wait_finished(store)[0].success
    # reference: plain torch copy of the same blocks (CPU golden)
    golden = [t[:32].cpu().clone() for t in group]

    for t in group:
        t.zero_()
    torch.cuda.synchronize()
    load.transfer_async(hashes, {0: ids})
    assert wait_finished(load)[0].success
    torch.cuda.synchronize()
    for t, g in zip(group, golden):
        assert torch.equal(t[:32].cpu(), g)
        assert (t[32:] == 0).all()


def test_gpu_store_respects_stream_fence(tmp_path):
    """The gather must observe KV writes issued on the caller stream before
    async_store (hipEventRecord / StreamWaitEvent fence)."""
    group = make_group(num_layers=2)
    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BPF),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="fence"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])

    # big async fill on the current stream, then store without syncing
    for t in group:
        t.fill_(7)
    store.transfer_async([0xF1], {0: list(range(BPF))})
    assert wait_finished(store)[0].success
    for t in group:
        t.zero_()
    torch.cuda.synchronize()
    load.transfer_async([0xF1], {0: list(range(BPF))})
    assert wait_finished(load)[0].success
    torch.cuda.synchronize()
    for t in group:
        assert (t[:BPF] == 7).all()


def test_gpu_partial_and_slot_offset(tmp_path):
    group = make_group(num_layers=2)
    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BPF),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="partial"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])
    ids = list(range(24))  # 1 full file + 8-block partial
    store.transfer_async([0xB1, 0xB2], {0: ids})
    assert wait_finished(store)[0].success
    golden = [t[:24].cpu().clone() for t in group]
    for t in group:
        t.zero_()
    torch.cuda.synchronize()
    load.transfer_async([0xB1, 0xB2], {0: list(range(20, 24))},
                        skip_leading_blocks=20)
    assert wait_finished(load)[0].success
    torch.cuda.synchronize()
    for t, g in zip(group, golden):
        assert torch.equal(t[20:24].cpu(), g[20:24])
        assert (t[:20] == 0).all()


def test_native_extension_is_loaded():
    """Fail loudly if the GPU path would silently run without the HIP .so."""
    import llm_d_kv_cache_amd._kvoffload as ko

    assert "_kvoffload" in ko.__file__
    assert os.path.dirname(ko.__file__).endswith("llm_d_kv_cache_amd")


def test_prefix_hash_kernel_matches_cpu():
    from llm_d_kv_cache_amd import ensure_native, _kvoffload as ko

    k = ensure_native()
    tp = k.TokenProcessor(16, "")
    n_seq, toks_per_seq = 64, 256
    tokens = torch.randint(0, 120000, (n_seq * toks_per_seq,), dtype=torch.int32,
                           device="cuda")
    seq_off = torch.arange(0, (n_seq + 1) * toks_per_seq, toks_per_seq,
                           dtype=torch.int64, device="cuda")
    n_chunks = toks_per_seq // 16
    key_off = torch.arange(0, (n_seq + 1) * n_chunks, n_chunks, dtype=torch.int64,
                           device="cuda")
    # per-sequence chain seed: model init hash (computed on CPU)
    seed = tp.tokens_to_block_keys(list(range(16)), "m")  # force tp init
    import reference_impl as ref

    seed_val = ref.fnv64a_chain_seed("m")
    seed_i64 = seed_val - (1 << 64) if seed_val >= (1 << 63) else seed_val
    seeds = torch.full((n_seq,), seed_i64, dtype=torch.int64, device="cuda")
    keys = torch.zeros(n_seq * n_chunks, dtype=torch.int64, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    ko.prefix_hash(tokens.data_ptr(), seq_off.data_ptr(), seeds.data_ptr(),
                   keys.data_ptr(), key_off.data_ptr(), 16, n_seq, stream)
    torch.cuda.synchronize()
    cpu_tokens = tokens.cpu().numpy().astype("uint32")
    got = keys.cpu().numpy().astype("uint64")
    for s in range(0, n_seq, 7):
        want = tp.tokens_to_block_keys(cpu_tokens[s * toks_per_seq:(s + 1) * toks_per_seq], "m")
        assert list(got[s * n_chunks:(s + 1) * n_chunks]) == want


@pytest.mark.parametrize("copy_path", ["staged", "zero_copy"])
def test_gpu_fp8_serialize_roundtrip(tmp_path, copy_path):
    """CDNA4 fused gather+quantize / dequantize+scatter kernels: round-trip
    within e4m3 tolerance; compare against plain PyTorch fp32 reference of
    the same quantization."""
    torch.manual_seed(5)
    nl = 4
    group = [
        (torch.randn(NUM_BLOCKS, BLOCK_BYTES // 2, device="cuda") * 3)
        .to(torch.bfloat16)
        for _ in range(nl)
    ]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BPF,
                            copy_path=copy_path, serialize="fp8_e4m3"),
    )
    mapper = FileMapper(str(tmp_path),
                        KVCacheLayoutConfig(model=f"gpu-fp8-{copy_path}"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])
    ids = list(range(BPF))
    store.transfer_async([0xF9], {0: ids})
    assert wait_finished(store)[0].success
    assert os.path.getsize(mapper.file_name(0xF9, 0)) == \
        BPF * nl * (BLOCK_BYTES // 2 + 4)
    orig = [t[:BPF].float().cpu() for t in group]
    for t in group:
        t.zero_()
    torch.cuda.synchronize()
    load.transfer_async([0xF9], {0: ids})
    assert wait_finished(load)[0].success
    torch.cuda.synchronize()
    for t, want in zip(group, orig):
        got = t[:BPF].float().cpu()
        amax = want.abs().amax()
        assert (got - want).abs().max() <= 0.07 * amax, \
            f"fp8 error too large: {(got - want).abs().max()} vs amax {amax}"


def test_gpu_host_cache_tier(tmp_path):
    """DRAM tier on the GPU path: loads hit the pinned cache (file
    deleted to prove the filesystem was skipped), bit-exact."""
    group = make_group(num_layers=4)
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BPF,
                            host_cache_bytes=4 * BPF * 4 * BLOCK_BYTES),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="gpu-hc"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])
    ids = list(range(BPF))
    store.transfer_async([0xCA], {0: ids})
    assert wait_finished(store)[0].success
    assert eng.stats().host_cache_stores == 1
    os.unlink(mapper.file_name(0xCA, 0))
    golden = [t[:BPF].cpu().clone() for t in group]
    for t in group:
        t.zero_()
    torch.cuda.synchronize()
    load.transfer_async([0xCA], {0: ids})
    assert wait_finished(load)[0].success
    torch.cuda.synchronize()
    assert eng.stats().host_cache_hits == 1
    for t, g in zip(group, golden):
        assert torch.equal(t[:BPF].cpu(), g)


def test_gpu_reads_overtake_write_storm(tmp_path):
    """QoS on the real GPU path: HIGH-priority loads jump a deep store
    backlog."""
    import time

    group = make_group(num_layers=2)
    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=4),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="gpu-qos"))
    store = GPUToStorageHandler(eng, mapper, [4])
    load = StorageToGPUHandler(eng, mapper, [4])
    for i in range(4):
        store.transfer_async([i], {0: list(range(4))})
    assert len(wait_finished(store, n=4)) == 4
    for i in range(150):
        store.transfer_async([1000 + i], {0: list(range(4))})
    t0 = time.time()
    for i in range(4):
        load.transfer_async([i], {0: list(range(4))})
    got = 0
    while got < 4 and time.time() < t0 + 20:
        got += len(load.get_finished())
        time.sleep(0.002)
    latency = time.time() - t0
    assert got == 4
    assert eng.native.pending_writes > 0, "storm drained before reads measured"
    assert latency < 5.0
    # drain
    deadline = time.time() + 60
    while eng.native.pending_writes > 0 and time.time() < deadline:
        time.sleep(0.05)


def test_gpu_obj_fp8_roundtrip():
    """Object-tier fp8: BlockCopier.gather_fp8/scatter_fp8 on device via the
    S3 engine (embedded fake server), e4m3 tolerance vs fp32 original."""
    import sys
    sys.path.insert(0, os.path.dirname(__file__))
    from test_obj_backend import _FakeS3, ThreadingHTTPServer
    import threading

    from llm_d_kv_cache_amd.offload.obj_backend import (
        ObjKeyMapper, ObjStorageConfig, ObjStorageEngine)

    _FakeS3.store = {}
    srv = ThreadingHTTPServer(("127.0.0.1", 0), _FakeS3)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        torch.manual_seed(7)
        group = [(torch.randn(16, 2048, device="cuda") * 2).to(torch.bfloat16)
                 for _ in range(2)]
        eng = ObjStorageEngine(
            [group],
            ObjStorageConfig(endpoint=f"http://127.0.0.1:{srv.server_port}",
                             serialize="fp8_e4m3"))
        mapper = ObjKeyMapper(
            FileMapper("/kv", KVCacheLayoutConfig(model="gobj8")))
        store = GPUToStorageHandler(eng, mapper, [8])
        load = StorageToGPUHandler(eng, mapper, [8])
        store.transfer_async([0xE8], {0: list(range(8))})
        assert wait_finished(store)[0].success
        assert sum(len(v) for v in _FakeS3.store.values()) == \
            8 * 2 * (2048 * 2 // 2 + 4)
        orig = [t[:8].float().cpu() for t in group]
        for t in group:
            t.zero_()
        torch.cuda.synchronize()
        load.transfer_async([0xE8], {0: list(range(8))})
        assert wait_finished(load)[0].success
        torch.cuda.synchronize()
        for t, want in zip(group, orig):
            amax = want.abs().amax()
            got = t[:8].float().cpu()
            assert (got - want).abs().max() <= 0.07 * amax
    finally:
        srv.shutdown()


def test_gpu_fp8_dram_writeback_compose(tmp_path):
    """fp8 + pinned DRAM cache + write-back on the staged GPU path: flush
    writes the fp8 file, a DRAM hit (file deleted) dequantizes on device."""
    torch.manual_seed(12)
    nl = 4
    group = [
        (torch.randn(NUM_BLOCKS, BLOCK_BYTES // 2, device="cuda") * 2)
        .to(torch.bfloat16)
        for _ in range(nl)
    ]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BPF,
                            copy_path="staged", serialize="fp8_e4m3",
                            host_cache_bytes=256 << 20, write_policy="back"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="gfp8wb"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])
    ids = list(range(BPF))
    store.transfer_async([0xFC], {0: ids})
    assert wait_finished(store)[0].success
    deadline = time.time() + 15
    while eng.stats().writeback_flushes < 1 and time.time() < deadline:
        time.sleep(0.01)
    path = mapper.file_name(0xFC, 0)
    assert os.path.getsize(path) == BPF * nl * (BLOCK_BYTES // 2 + 4)
    os.remove(path)
    orig = [t[:BPF].float().cpu() for t in group]
    for t in group:
        t.zero_()
    torch.cuda.synchronize()
    load.transfer_async([0xFC], {0: ids})
    assert wait_finished(load)[0].success
    torch.cuda.synchronize()
    assert eng.stats().host_cache_hits >= 1
    for t, o in zip(group, orig):
        amax = o.abs().amax()
        assert (t[:BPF].float().cpu() - o).abs().max() <= 0.07 * amax


def test_gpu_small_tile_flat_kernels_bit_exact():
    """8 KiB tiles (70B-TP8 shard geometry) exercise the flat small-tile
    gather/scatter kernels (LDS tile table + magic division): the packed
    slab must equal a plain PyTorch gather of the same blocks."""
    from llm_d_kv_cache_amd import ensure_offload_native

    ko = ensure_offload_native()
    num_layers, block_bytes, nb = 80, 8 * 1024, 16
    group = [
        torch.randint(0, 255, (64, block_bytes), dtype=torch.uint8,
                      device="cuda")
        for _ in range(num_layers)
    ]
    copier = ko.BlockCopier(
        [([t.data_ptr() for t in group], [t.stride(0) for t in group],
          block_bytes)],
        gpu_mode=True,
    )
    ids = list(range(1, 1 + nb * 3, 3))
    slab = torch.empty(copier.packed_bytes(0, nb), dtype=torch.uint8,
                       device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    copier.gather(0, ids, slab.data_ptr(), stream)
    torch.cuda.synchronize()
    ref = torch.cat([group[l][i] for i in ids for l in range(num_layers)])
    assert torch.equal(slab, ref)
    # scatter back into zeroed pages and compare
    orig = [t.clone() for t in group]
    for t in group:
        t.zero_()
    copier.scatter(0, ids, slab.data_ptr(), stream)
    torch.cuda.synchronize()
    for t, o in zip(group, orig):
        assert torch.equal(t[ids], o[ids])


def test_gpu_dev_ids_path_bit_exact(tmp_path):
    """blocks-per-file above the 128-entry kernarg limit ride the worker's
    device id buffer; the full store/load path stays bit-exact."""
    nb = 256
    group = [
        torch.randint(0, 255, (512, 4096), dtype=torch.uint8, device="cuda")
        for _ in range(2)
    ]
    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=nb,
                                     copy_path="staged"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="devids"))
    store = GPUToStorageHandler(eng, mapper, [nb])
    load = StorageToGPUHandler(eng, mapper, [nb])
    ids = list(range(3, 3 + nb))
    store.transfer_async([0xB1], {0: ids})
    assert wait_finished(store)[0].success
    orig = [t.clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0xB1], {0: ids})
    assert wait_finished(load)[0].success
    for t, o in zip(group, orig):
        assert torch.equal(t[ids], o[ids])


def test_gpu_read_latency_percentiles_under_saturation(tmp_path):
    """GPU twin of the host-path percentile QoS test: reads submitted into
    a saturated write queue keep bounded p50/p99 and a <5x tail ratio
    (reference test_priority_queue.py:257)."""
    group = make_group()
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=2, gpu_blocks_per_file=BPF,
                            copy_path="staged"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="gpu-pct"))
    store = GPUToStorageHandler(eng, mapper, [BPF])
    load = StorageToGPUHandler(eng, mapper, [BPF])

    n_read_files = 5
    for i in range(n_read_files):
        store.transfer_async([i], {0: list(range(i * BPF, (i + 1) * BPF))})
    assert len(wait_finished(store, n_read_files)) == n_read_files

    n_storm = 200
    for i in range(n_storm):
        store.transfer_async([1000 + i], {0: list(range(BPF))})
    lats = []
    backlog = 0  # deepest backlog observed while reads were in flight
    for i in range(15):
        f = i % n_read_files
        t0 = time.time()
        load.transfer_async([f], {0: list(range(f * BPF, (f + 1) * BPF))})
        assert wait_finished(load, 1)[0].success
        lats.append(time.time() - t0)
        backlog = max(backlog, eng.native.pending_writes)
    wait_finished(store, n_storm, timeout=120)
    lats.sort()
    p50 = lats[len(lats) // 2]
    p99 = lats[-1]
    tail = p99 / max(p50, 0.005)
    assert backlog > 0, "write backlog drained before the reads"
    assert p50 < 0.5, f"GPU read p50 {p50:.3f}s under write saturation"
    assert p99 < 1.5, f"GPU read p99 {p99:.3f}s under write saturation"
    assert tail < 5.0, f"GPU tail ratio {tail:.1f}x under saturation"
