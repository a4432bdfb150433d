"""Object-store offload backend (S3-compatible HTTP).

Capability parity with the reference llmd_nixl OBJ backend
(llmd_nixl/nixl_offload.py, obj_backend.py, nixl_lookup.py): the same
handler protocol as the filesystem engine, but chunks live in an object
store. The data path is GPU pages -> CDNA4 gather into a packed slab ->
pinned staging -> HTTP PUT (and GET -> staging -> scatter on load); lookup
is a HEAD per chunk. Works against any S3-compatible endpoint (MinIO,
Ceph RGW, S3). Authenticated buckets are supported via AWS Signature V4
(set access_key/secret_key in ObjStorageConfig); anonymous/proxy-fronted
endpoints work with no credentials configured.
"""
from __future__ import annotations

import concurrent.futures
import hashlib
import hmac as hmaclib
import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple
from urllib import request as urlrequest
from urllib.error import HTTPError, URLError
from urllib.parse import parse_qsl, quote, urlparse


@dataclass
class ObjStorageConfig:
    endpoint: str = "http://127.0.0.1:9000"
    bucket: str = "kvcache"
    io_threads: int = 8
    timeout_s: float = 30.0
    serialize: str = "raw"  # raw | fp8_e4m3 (via the native codec kernels)
    # AWS SigV4 credentials; leave access_key empty for anonymous access.
    access_key: str = ""
    secret_key: str = ""
    session_token: str = ""
    region: str = "us-east-1"


_EMPTY_SHA256 = hashlib.sha256(b"").hexdigest()


def sigv4_headers(method: str, url: str, *, access_key: str, secret_key: str,
                  region: str, payload_hash: str = _EMPTY_SHA256,
                  service: str = "s3", amz_date: Optional[str] = None,
                  session_token: str = "",
                  extra_headers: Optional[Dict[str, str]] = None,
                  sign_content_sha256: bool = True) -> Dict[str, str]:
    """AWS Signature Version 4 headers for one request.

    Pure-Python implementation of the SigV4 canonicalization + HMAC chain
    (validated against the AWS SigV4 test-suite `get-vanilla` vector in
    tests/test_obj_backend.py). Signs host, x-amz-date,
    x-amz-content-sha256 (S3 requires it), the session token if present,
    and any extra headers passed in (e.g. Range is deliberately NOT
    signed — S3 treats non-amz headers as optional in the signature).
    """
    parsed = urlparse(url)
    amz_date = amz_date or time.strftime("%Y%m%dT%H%M%SZ", time.gmtime())
    datestamp = amz_date[:8]
    canonical_uri = quote(parsed.path or "/", safe="/-_.~")
    qs = sorted(parse_qsl(parsed.query, keep_blank_values=True))
    canonical_query = "&".join(
        f"{quote(k, safe='-_.~')}={quote(v, safe='-_.~')}" for k, v in qs)
    headers = {"host": parsed.netloc, "x-amz-date": amz_date}
    if sign_content_sha256:
        headers["x-amz-content-sha256"] = payload_hash
    if session_token:
        headers["x-amz-security-token"] = session_token
    for k, v in (extra_headers or {}).items():
        headers[k.lower()] = v.strip()
    names = sorted(headers)
    signed = ";".join(names)
    canonical_headers = "".join(f"{k}:{headers[k]}\n" for k in names)
    canonical_request = "\n".join(
        [method, canonical_uri, canonical_query, canonical_headers, signed,
         payload_hash])
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    string_to_sign = "\n".join(
        ["AWS4-HMAC-SHA256", amz_date, scope,
         hashlib.sha256(canonical_request.encode()).hexdigest()])
    key = f"AWS4{secret_key}".encode()
    for part in (datestamp, region, service, "aws4_request"):
        key = hmaclib.new(key, part.encode(), hashlib.sha256).digest()
    sig = hmaclib.new(key, string_to_sign.encode(), hashlib.sha256).hexdigest()
    out = dict(headers)
    out.pop("host")  # urllib sets Host itself
    out["Authorization"] = (
        f"AWS4-HMAC-SHA256 Credential={access_key}/{scope}, "
        f"SignedHeaders={signed}, Signature={sig}")
    return out


class ObjClient:
    """Minimal S3-compatible object client (PUT/GET/HEAD/DELETE)."""

    def __init__(self, cfg: ObjStorageConfig):
        self.cfg = cfg

    def _url(self, key: str) -> str:
        return f"{self.cfg.endpoint}/{self.cfg.bucket}/{key}"

    def _sign(self, req: urlrequest.Request, payload: bytes = b"") -> None:
        if not self.cfg.access_key:
            return
        for k, v in sigv4_headers(
                req.get_method(), req.full_url,
                access_key=self.cfg.access_key,
                secret_key=self.cfg.secret_key, region=self.cfg.region,
                payload_hash=hashlib.sha256(payload).hexdigest(),
                session_token=self.cfg.session_token).items():
            req.add_header(k, v)

    def put(self, key: str, data: bytes) -> None:
        req = urlrequest.Request(self._url(key), data=data, method="PUT")
        req.add_header("Content-Type", "application/octet-stream")
        self._sign(req, data)
        with urlrequest.urlopen(req, timeout=self.cfg.timeout_s) as resp:
            if resp.status not in (200, 201, 204):
                raise IOError(f"PUT {key}: HTTP {resp.status}")

    def get(self, key: str, offset: int = 0, length: Optional[int] = None) -> bytes:
        req = urlrequest.Request(self._url(key), method="GET")
        if offset or length is not None:
            end = "" if length is None else str(offset + length - 1)
            req.add_header("Range", f"bytes={offset}-{end}")
        self._sign(req)
        with urlrequest.urlopen(req, timeout=self.cfg.timeout_s) as resp:
            return resp.read()

    def head(self, key: str) -> bool:
        req = urlrequest.Request(self._url(key), method="HEAD")
        self._sign(req)
        try:
            with urlrequest.urlopen(req, timeout=self.cfg.timeout_s) as resp:
                return resp.status == 200
        except HTTPError:
            return False
        except URLError:
            return False

    def delete(self, key: str) -> None:
        req = urlrequest.Request(self._url(key), method="DELETE")
        self._sign(req)
        try:
            urlrequest.urlopen(req, timeout=self.cfg.timeout_s)
        except HTTPError:
            pass


class ObjStorageEngine:
    """Same surface as TorchOffloadEngine for the handler layer, but
    transfers target object keys instead of file paths."""

    def __init__(self, groups: Sequence[Sequence], config: ObjStorageConfig):
        import torch

        from .. import ensure_offload_native

        ko = ensure_offload_native()
        self._torch = torch
        self.config = config
        self.gpu_mode = groups[0][0].is_cuda
        native_groups = []
        self.group_geometry: List[dict] = []
        for g in groups:
            ptrs = [t.data_ptr() for t in g]
            strides = [t.stride(0) * t.element_size() for t in g]
            bb = strides[0]
            native_groups.append((ptrs, strides, bb, int(g[0].shape[0])))
            record = bb // 2 + 4 if config.serialize == "fp8_e4m3" else bb
            self.group_geometry.append(
                {"num_layers": len(g), "block_bytes": bb, "record_bytes": record,
                 "num_device_blocks": g[0].shape[0]}
            )
        self._fp8 = config.serialize == "fp8_e4m3"
        device = groups[0][0].device.index or 0 if self.gpu_mode else 0
        self._copier = ko.BlockCopier(native_groups, self.gpu_mode, device)
        self._tensors = [list(g) for g in groups]
        self.client = ObjClient(config)
        self._pool = concurrent.futures.ThreadPoolExecutor(
            max_workers=config.io_threads)
        self._jobs: Dict[int, List[concurrent.futures.Future]] = {}
        self._next_job = 1
        self._mu = threading.Lock()
        self._fin_lock = threading.Lock()
        self._fin_buffer: Dict[int, Tuple[bool, bool]] = {}

    # ---- engine surface -----------------------------------------------------

    def _packed(self, group: int, n_blocks: int) -> int:
        if self._fp8:
            return self._copier.packed_bytes_fp8(group, n_blocks)
        return self._copier.packed_bytes(group, n_blocks)

    def _slab(self, group: int, n_blocks: int):
        t = self._torch.empty(self._packed(group, n_blocks),
                              dtype=self._torch.uint8,
                              pin_memory=self.gpu_mode)
        return t

    def _store_one(self, group: int, key: str, ids: List[int]) -> None:
        if self.client.head(key):
            return  # dedupe
        nb = self._packed(group, len(ids))
        if self.gpu_mode:
            scratch_b = (self._copier.fp8_scratch_bytes(group, len(ids))
                         if self._fp8 else 0)
            dev = self._torch.empty(nb + scratch_b, dtype=self._torch.uint8,
                                    device="cuda")
            stream = self._torch.cuda.Stream()
            with self._torch.cuda.stream(stream):
                if self._fp8:
                    self._copier.gather_fp8(group, ids, dev.data_ptr(),
                                            dev.data_ptr() + nb,
                                            stream.cuda_stream)
                else:
                    self._copier.gather(group, ids, dev.data_ptr(),
                                        stream.cuda_stream)
                host = self._slab(group, len(ids))
                host.copy_(dev[:nb], non_blocking=True)
            stream.synchronize()
        else:
            host = self._slab(group, len(ids))
            if self._fp8:
                self._copier.gather_fp8(group, ids, host.data_ptr(), 0, 0)
            else:
                self._copier.gather(group, ids, host.data_ptr(), 0)
        self.client.put(key, host.numpy().tobytes())

    def _load_one(self, group: int, key: str, ids: List[int],
                  slot_offset: int) -> None:
        geo = self.group_geometry[group]
        rec = geo["record_bytes"] * geo["num_layers"]
        data = self.client.get(key, offset=slot_offset * rec,
                               length=len(ids) * rec)
        if len(data) != len(ids) * rec:
            raise IOError(f"short object read for {key}")
        host = self._torch.frombuffer(bytearray(data), dtype=self._torch.uint8)
        scatter = (self._copier.scatter_fp8 if self._fp8
                   else self._copier.scatter)
        if self.gpu_mode:
            dev = host.cuda()
            self._torch.cuda.synchronize()
            scatter(group, ids, dev.data_ptr(),
                    self._torch.cuda.current_stream().cuda_stream)
            self._torch.cuda.synchronize()
        else:
            scatter(group, ids, host.data_ptr(), 0)

    def async_store(self, files, stream: int = None) -> int:
        if self.gpu_mode:
            self._torch.cuda.synchronize()  # KV-ready fence (coarse)
        with self._mu:
            job = self._next_job
            self._next_job += 1
        futs = [self._pool.submit(self._store_one, g, key, ids)
                for (g, key, ids, _off) in files]
        with self._mu:
            self._jobs[job] = futs
        return job

    def async_load(self, files) -> int:
        with self._mu:
            job = self._next_job
            self._next_job += 1
        futs = [self._pool.submit(self._load_one, g, key, ids, off)
                for (g, key, ids, off) in files]
        with self._mu:
            self._jobs[job] = futs
        return job

    def get_finished(self):
        out = []
        with self._mu:
            for job, futs in list(self._jobs.items()):
                if all(f.done() for f in futs):
                    success = all(f.exception() is None for f in futs)
                    out.append((job, success, False))
                    del self._jobs[job]
        return out

    def poll_finished(self, job_ids):
        with self._fin_lock:
            for jid, success, dropped in self.get_finished():
                self._fin_buffer[jid] = (success, dropped)
            out = []
            for jid in list(job_ids):
                if jid in self._fin_buffer:
                    success, dropped = self._fin_buffer.pop(jid)
                    out.append((jid, success, dropped))
            return out

    def wait_job(self, job_id: int) -> bool:
        with self._mu:
            futs = self._jobs.pop(job_id, [])
        ok = True
        for f in futs:
            if not f.cancel():
                try:
                    f.result(timeout=60)
                except Exception:
                    ok = False
        return ok

    def current_stream_handle(self) -> int:
        return 0

    def close(self):
        self._pool.shutdown(wait=True)


class ObjKeyMapper:
    """FileMapper-compatible naming for object keys (no leading root)."""

    def __init__(self, mapper):
        self._mapper = mapper
        self.run_dir = mapper.run_dir

    def file_name(self, chunk_hash: int, group: int = 0) -> str:
        import os

        return os.path.relpath(self._mapper.file_name(chunk_hash, group),
                               self._mapper.root)


class ObjOffloadManager:
    """Scheduler-side lookup over HEAD (reference NixlLookup parity)."""

    def __init__(self, key_mapper: ObjKeyMapper, client: ObjClient,
                 num_groups: int = 1):
        self.mapper = key_mapper
        self.client = client
        self.num_groups = num_groups

    def lookup(self, chunk_hashes: Sequence[int]) -> int:
        hits = 0
        for h in chunk_hashes:
            if all(self.client.head(self.mapper.file_name(h, g))
                   for g in range(self.num_groups)):
                hits += 1
            else:
                break
        return hits
