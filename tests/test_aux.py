"""Auxiliary subsystems: Prometheus metrics, tracing, PVC evictor."""
import os
import time

import pytest

from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer

k = ensure_native()


# ---- metrics ----------------------------------------------------------------

def test_prometheus_collector():
    from prometheus_client import CollectorRegistry, generate_latest

    from llm_d_kv_cache_amd.utils.metrics import KVCacheMetricsCollector

    ix = KVCacheIndexer(IndexerConfig())
    tokens = list(range(32))
    keys = ix.compute_block_keys(tokens, "m")
    ix.index.add([], keys, [k.PodEntry("pod-a", "gpu")])
    ix.score_tokens(tokens, "m")

    reg = CollectorRegistry()
    reg.register(KVCacheMetricsCollector(indexer=ix))
    text = generate_latest(reg).decode()
    assert "kvcache_index_admissions_total 2.0" in text
    assert "kvcache_index_lookup_requests_total 1.0" in text
    assert "kvcache_index_lookup_hits_total 1.0" in text
    assert "kvcache_index_keys 2.0" in text


def test_metrics_beat_logs(caplog):
    import logging

    from llm_d_kv_cache_amd.utils.metrics import start_metrics_logging

    ix = KVCacheIndexer(IndexerConfig())
    with caplog.at_level(logging.INFO, logger="llm_d_kv_cache_amd.utils.metrics"):
        stop = start_metrics_logging(ix, interval_s=0.05)
        time.sleep(0.2)
        stop.set()
    assert any("metrics beat" in r.message for r in caplog.records)


# ---- tracing ----------------------------------------------------------------

def test_tracing_jsonl_export(tmp_path):
    import json

    from llm_d_kv_cache_amd.utils import tracing

    path = str(tmp_path / "trace.jsonl")
    t = tracing.Tracer("test", 1.0, tracing._JsonlExporter(path))
    with t.span("outer", foo=1) as outer:
        outer.set_attribute("bar", 2)
        with t.span("inner"):
            pass
    recs = [json.loads(line) for line in open(path)]
    assert len(recs) == 2
    inner, outer_rec = recs
    assert inner["name"] == "test.inner"
    assert inner["trace_id"] == outer_rec["trace_id"]
    assert inner["parent_id"] == outer_rec["span_id"]
    assert outer_rec["attributes"] == {"foo": 1, "bar": 2}


def test_tracing_sampling_zero_ratio():
    from llm_d_kv_cache_amd.utils import tracing

    calls = []

    class Exp:
        def export(self, span):
            calls.append(span)

    t = tracing.Tracer("test", 0.0, Exp())
    with t.span("never") as s:
        s.set_attribute("x", 1)  # noop span accepts attributes
    assert calls == []


def test_traced_indexer_score_path():
    from llm_d_kv_cache_amd.utils import tracing

    spans = []

    class Exp:
        def export(self, span):
            spans.append(span)

    ix = KVCacheIndexer(IndexerConfig())
    tokens = list(range(64))
    keys = ix.compute_block_keys(tokens, "m")
    ix.index.add([], keys[:2], [k.PodEntry("p", "gpu")])
    traced = tracing.TracedIndexer(ix, tracing.Tracer("kvc", 1.0, Exp()))
    scores = traced.score_tokens(tokens, "m")
    assert scores == {"p": 2.0}
    assert len(spans) == 1
    assert spans[0].attributes["total_blocks"] == 4
    assert spans[0].attributes["hit_blocks"] == 2
    assert abs(spans[0].attributes["block_hit_ratio"] - 0.5) < 1e-9


# ---- evictor ----------------------------------------------------------------

def util_always_high(root):
    return 0.99


def util_always_low(root):
    return 0.01


def _layout_with_files(tmp_path, n=8, age_s=7200, offset=0):
    """Build a mapper-style layout with aged files."""
    from llm_d_kv_cache_amd.offload import FileMapper, KVCacheLayoutConfig

    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="evict-test"))
    paths = []
    old = time.time() - age_s
    for h in range(offset + 1, offset + n + 1):
        p = mapper.file_name(h * 7919, 0)
        os.makedirs(os.path.dirname(p), exist_ok=True)
        with open(p, "wb") as f:
            f.write(b"x" * 128)
        os.utime(p, (old, old))
        paths.append(p)
    return paths


def test_evictor_deletes_cold_files_under_pressure(tmp_path):
    from llm_d_kv_cache_amd.evictor import EvictorConfig, PvcEvictor

    paths = _layout_with_files(tmp_path, n=8)
    cfg = EvictorConfig(root=str(tmp_path), crawlers=2, atime_threshold_s=60,
                        crawl_interval_s=0.1, check_interval_s=0.05)
    ev = PvcEvictor(cfg, utilization=util_always_high)
    ev.start()
    try:
        deadline = time.time() + 30
        while any(os.path.exists(p) for p in paths) and time.time() < deadline:
            time.sleep(0.1)
        assert not any(os.path.exists(p) for p in paths)
        assert ev.deleted.value >= 8
    finally:
        ev.shutdown()


def test_evictor_idle_below_threshold(tmp_path):
    from llm_d_kv_cache_amd.evictor import EvictorConfig, PvcEvictor

    paths = _layout_with_files(tmp_path, n=4)
    cfg = EvictorConfig(root=str(tmp_path), crawlers=1, atime_threshold_s=60,
                        crawl_interval_s=0.1, check_interval_s=0.05)
    ev = PvcEvictor(cfg, utilization=util_always_low)
    ev.start()
    try:
        time.sleep(1.0)
        assert all(os.path.exists(p) for p in paths)
    finally:
        ev.shutdown()


def test_evictor_spares_warm_files(tmp_path):
    from llm_d_kv_cache_amd.evictor import EvictorConfig, PvcEvictor

    cold = _layout_with_files(tmp_path, n=4, age_s=7200)
    warm = _layout_with_files(tmp_path, n=2, age_s=0, offset=100)
    # same run dir; warm files have fresh atime
    cfg = EvictorConfig(root=str(tmp_path), crawlers=1, atime_threshold_s=3600,
                        crawl_interval_s=0.1, check_interval_s=0.05)
    ev = PvcEvictor(cfg, utilization=util_always_high)
    ev.start()
    try:
        deadline = time.time() + 30
        while any(os.path.exists(p) for p in cold) and time.time() < deadline:
            time.sleep(0.1)
        assert not any(os.path.exists(p) for p in cold)
        assert all(os.path.exists(p) for p in warm)
    finally:
        ev.shutdown()


def test_evictor_restarts_dead_children(tmp_path):
    from llm_d_kv_cache_amd.evictor import EvictorConfig, PvcEvictor

    cfg = EvictorConfig(root=str(tmp_path), crawlers=1, crawl_interval_s=0.1,
                        check_interval_s=0.05)
    ev = PvcEvictor(cfg, utilization=util_always_low)
    ev.start()
    try:
        victim = ev._procs["activator"]
        victim.terminate()
        victim.join(timeout=10)
        ev.supervise_once()
        assert ev._procs["activator"].is_alive()
    finally:
        ev.shutdown()


def test_prometheus_offload_engine_metrics(tmp_path):
    import torch
    from prometheus_client import CollectorRegistry, generate_latest

    from llm_d_kv_cache_amd.offload import (
        FileMapper,
        GPUToStorageHandler,
        KVCacheLayoutConfig,
        OffloadEngineConfig,
        TorchOffloadEngine,
    )
    from llm_d_kv_cache_amd.utils.metrics import KVCacheMetricsCollector

    group = [torch.zeros(16, 1024, dtype=torch.uint8)]
    eng = TorchOffloadEngine(
        [group], OffloadEngineConfig(io_threads=1, gpu_blocks_per_file=4,
                                     copy_path="host"),
    )
    mapper = FileMapper(str(tmp_path), KVCacheLayoutConfig(model="metrics"))
    store = GPUToStorageHandler(eng, mapper, [4])
    store.transfer_async([1], {0: [0, 1, 2, 3]})
    deadline = time.time() + 10
    while not store.get_finished() and time.time() < deadline:
        time.sleep(0.01)
    reg = CollectorRegistry()
    reg.register(KVCacheMetricsCollector(offload_engine=eng))
    text = generate_latest(reg).decode()
    assert "kv_offload_files_written_total 1.0" in text
    assert "kv_offload_bytes_stored_total 4096.0" in text


def test_evictor_folder_cleaner(tmp_path):
    from llm_d_kv_cache_amd.evictor import clean_empty_dirs

    run = tmp_path / "m_abc_r0"
    (run / "0aa" / "bb_g0").mkdir(parents=True)      # empty chain -> pruned
    (run / "0cc" / "dd_g0").mkdir(parents=True)
    (run / "0cc" / "dd_g0" / "x.bin").write_bytes(b"1")  # non-empty -> kept
    (run / "config.json").write_text("{}")
    removed = clean_empty_dirs(str(tmp_path))
    assert removed == 2  # 0aa/bb_g0 and then 0aa itself
    assert not (run / "0aa").exists()
    assert (run / "0cc" / "dd_g0" / "x.bin").exists()
    assert (run / "config.json").exists()


def test_routing_sim_precise_beats_baselines():
    """The routing-value experiment (reference benchmarking shape) driven
    by the real native scoring + event-ingestion paths: precise scheduling
    must cut TTFT and raise the cached-prefix fraction vs load/random."""
    import sys
    from pathlib import Path

    sys.path.insert(0, str(Path(__file__).resolve().parents[1] / "tools"))
    import routing_sim

    out = routing_sim.main([
        "--requests", "300", "--qps", "18", "--groups", "12",
        "--prefix-tokens", "1024", "--question-tokens", "128",
        "--output-tokens", "30", "--prefill-tok-s", "3000",
        "--capacity-blocks", "512",
    ])
    res = {r["scheduler"]: r for r in out["results"]}
    assert res["precise"]["cached_prefix_frac"] > \
        1.5 * res["random"]["cached_prefix_frac"]
    assert res["precise"]["ttft_mean_s"] < res["load"]["ttft_mean_s"] / 1.5
    assert res["precise"]["ttft_mean_s"] < res["random"]["ttft_mean_s"] / 3


def test_cli_serve_and_score(tmp_path):
    """python -m llm_d_kv_cache_amd: serve + score end to end over a real
    gRPC port, with a snapshot restored at startup."""
    import json
    import signal
    import socket
    import subprocess
    import sys
    import time

    from llm_d_kv_cache_amd import ensure_native
    from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer

    k = ensure_native()
    ix = KVCacheIndexer(IndexerConfig())
    keys = ix.compute_block_keys(list(range(32)), "cli-m")
    ix.index.add([], keys, [k.PodEntry("pod-cli", "gpu")])
    snap = str(tmp_path / "cli.snap")
    ix.save_index(snap)

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    srv = subprocess.Popen(
        [sys.executable, "-m", "llm_d_kv_cache_amd", "serve",
         "--grpc-port", str(port), "--zmq-endpoint", "tcp://127.0.0.1:0",
         "--snapshot-path", snap],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        deadline = time.time() + 30
        out = None
        while time.time() < deadline:
            try:
                out = subprocess.run(
                    [sys.executable, "-m", "llm_d_kv_cache_amd", "score",
                     "--target", f"127.0.0.1:{port}", "--model", "cli-m",
                     "--tokens", ",".join(str(t) for t in range(32)),
                     "--pods", "pod-cli"],
                    capture_output=True, text=True, timeout=15)
                if out.returncode == 0 and "pod-cli" in out.stdout:
                    break
            except subprocess.TimeoutExpired:
                pass
            time.sleep(0.5)
        assert out is not None and "pod-cli\t2.0" in out.stdout, \
            (out.stdout if out else "no output")
    finally:
        srv.send_signal(signal.SIGINT)
        try:
            srv.wait(timeout=10)
        except subprocess.TimeoutExpired:
            srv.kill()


def test_metered_indexer_metrics():
    from prometheus_client import CollectorRegistry, generate_latest

    from llm_d_kv_cache_amd import ensure_native
    from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
    from llm_d_kv_cache_amd.utils.metrics import MeteredIndexer

    k = ensure_native()
    reg = CollectorRegistry()
    ix = MeteredIndexer(KVCacheIndexer(IndexerConfig()), registry=reg)
    keys = ix.compute_block_keys(list(range(48)), "m")
    ix.index.add([], keys, [k.PodEntry("p", "gpu")])
    assert ix.score_tokens(list(range(48)), "m") == {"p": 3.0}
    ix.score_tokens(list(range(48)), "m")
    text = generate_latest(reg).decode()
    assert "kvcache_lookup_latency_seconds_count 2.0" in text
    assert "kvcache_max_pod_hit_count_total 6.0" in text
