"""Object-store backend against an embedded S3-compatible fake."""
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest
import torch

from llm_d_kv_cache_amd.offload import FileMapper, KVCacheLayoutConfig
from llm_d_kv_cache_amd.offload.handlers import (
    GPUToStorageHandler,
    StorageToGPUHandler,
)
from llm_d_kv_cache_amd.offload.obj_backend import (
    ObjClient,
    ObjKeyMapper,
    ObjOffloadManager,
    ObjStorageConfig,
    ObjStorageEngine,
)


class _FakeS3(BaseHTTPRequestHandler):
    store = {}

    def log_message(self, *a):
        pass

    def do_PUT(self):
        n = int(self.headers.get("Content-Length", 0))
        _FakeS3.store[self.path] = self.rfile.read(n)
        self.send_response(200)
        self.end_headers()

    def do_GET(self):
        data = _FakeS3.store.get(self.path)
        if data is None:
            self.send_response(404)
            self.end_headers()
            return
        rng = self.headers.get("Range")
        status = 200
        if rng and rng.startswith("bytes="):
            lo, hi = rng[6:].split("-")
            lo = int(lo)
            hi = int(hi) if hi else len(data) - 1
            data = data[lo:hi + 1]
            status = 206
        self.send_response(status)
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def do_HEAD(self):
        self.send_response(200 if self.path in _FakeS3.store else 404)
        self.end_headers()

    def do_DELETE(self):
        _FakeS3.store.pop(self.path, None)
        self.send_response(204)
        self.end_headers()


@pytest.fixture
def s3(tmp_path):
    _FakeS3.store = {}
    srv = ThreadingHTTPServer(("127.0.0.1", 0), _FakeS3)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{srv.server_port}"
    srv.shutdown()


def wait_finished(handler, n=1, timeout=20.0):
    import time

    out = []
    deadline = time.time() + timeout
    while len(out) < n and time.time() < deadline:
        out.extend(handler.get_finished())
        time.sleep(0.01)
    assert len(out) >= n
    return out


def test_obj_roundtrip(s3, tmp_path):
    group = [torch.randint(0, 255, (32, 2048), dtype=torch.uint8)
             for _ in range(2)]
    cfg = ObjStorageConfig(endpoint=s3, io_threads=4)
    eng = ObjStorageEngine([group], cfg)
    mapper = ObjKeyMapper(FileMapper("/kv", KVCacheLayoutConfig(model="obj")))
    store = GPUToStorageHandler(eng, mapper, [4])
    load = StorageToGPUHandler(eng, mapper, [4])
    mgr = ObjOffloadManager(mapper, eng.client)

    store.transfer_async([0x0B1, 0x0B2], {0: list(range(8))})
    assert wait_finished(store)[0].success
    assert mgr.lookup([0x0B1, 0x0B2, 0x0B3]) == 2

    orig = [t.clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0x0B1, 0x0B2], {0: list(range(8))})
    assert wait_finished(load)[0].success
    for t, o in zip(group, orig):
        assert torch.equal(t[:8], o[:8])


def test_obj_partial_range_load(s3):
    group = [torch.randint(0, 255, (16, 1024), dtype=torch.uint8)]
    eng = ObjStorageEngine([group], ObjStorageConfig(endpoint=s3))
    mapper = ObjKeyMapper(FileMapper("/kv", KVCacheLayoutConfig(model="objp")))
    store = GPUToStorageHandler(eng, mapper, [8])
    load = StorageToGPUHandler(eng, mapper, [8])
    store.transfer_async([0xAA], {0: list(range(8))})
    assert wait_finished(store)[0].success
    orig = group[0].clone()
    group[0].zero_()
    load.transfer_async([0xAA], {0: [6, 7]}, skip_leading_blocks=6)
    assert wait_finished(load)[0].success
    assert torch.equal(group[0][6:8], orig[6:8])
    assert (group[0][:6] == 0).all()


def test_obj_dedupe_and_missing(s3):
    group = [torch.randint(0, 255, (8, 1024), dtype=torch.uint8)]
    eng = ObjStorageEngine([group], ObjStorageConfig(endpoint=s3))
    mapper = ObjKeyMapper(FileMapper("/kv", KVCacheLayoutConfig(model="objd")))
    store = GPUToStorageHandler(eng, mapper, [4])
    load = StorageToGPUHandler(eng, mapper, [4])
    store.transfer_async([0x1], {0: [0, 1, 2, 3]})
    assert wait_finished(store)[0].success
    n_objects = len(_FakeS3.store)
    store.transfer_async([0x1], {0: [0, 1, 2, 3]})  # dedupe via HEAD
    assert wait_finished(store)[0].success
    assert len(_FakeS3.store) == n_objects
    load.transfer_async([0x999], {0: [0, 1]})
    assert not wait_finished(load)[0].success


def test_obj_fp8_roundtrip(s3):
    torch.manual_seed(6)
    group = [(torch.randn(16, 1024) * 3).to(torch.bfloat16) for _ in range(2)]
    eng = ObjStorageEngine([group], ObjStorageConfig(endpoint=s3,
                                                     serialize="fp8_e4m3"))
    mapper = ObjKeyMapper(FileMapper("/kv", KVCacheLayoutConfig(model="obj8")))
    store = GPUToStorageHandler(eng, mapper, [4])
    load = StorageToGPUHandler(eng, mapper, [4])
    store.transfer_async([0xF0], {0: [0, 1, 2, 3]})
    assert wait_finished(store)[0].success
    # object is ~half the raw bytes
    raw = 4 * 2 * 2048
    assert sum(len(v) for v in _FakeS3.store.values()) < raw * 0.6
    orig = [t[:4].float().clone() for t in group]
    for t in group:
        t.zero_()
    load.transfer_async([0xF0], {0: [0, 1, 2, 3]})
    assert wait_finished(load)[0].success
    for t, o in zip(group, orig):
        amax = o.abs().amax()
        assert (t[:4].float() - o).abs().max() <= 0.07 * amax
