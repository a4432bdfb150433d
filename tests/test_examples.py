"""Examples + connector wiring stay runnable."""
import json
import subprocess
import sys
import time
from pathlib import Path

import pytest
import torch

REPO = Path(__file__).resolve().parent.parent


def test_offline_example_runs():
    out = subprocess.run(
        [sys.executable, str(REPO / "examples" / "kv_cache_index.py")],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr
    assert "pod scores" in out.stdout


def test_pod_reconciler_logic(tmp_path):
    sys.path.insert(0, str(REPO / "examples"))
    import pod_reconciler

    from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
    from llm_d_kv_cache_amd.events import (
        EventPoolConfig,
        KVEventsPool,
        SubscriberManager,
    )
    from llm_d_kv_cache_amd.events.publisher import EventPublisher

    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(discover_pods=True), ix)
    pool.start()
    mgr = SubscriberManager(pool, reconnect_ms=200)
    pub = EventPublisher("tcp://127.0.0.1:0", "pod-0", "m", bind=True)
    try:
        pods_file = tmp_path / "pods.json"
        pods_file.write_text(json.dumps(
            {"pod-0": f"tcp://127.0.0.1:{pub.port}"}))
        pod_reconciler.reconcile(mgr, pod_reconciler.load_pods(str(pods_file)))
        assert mgr.pods() == ["pod-0"]
        # pod removed -> subscriber torn down
        pods_file.write_text("{}")
        pod_reconciler.reconcile(mgr, pod_reconciler.load_pods(str(pods_file)))
        assert mgr.pods() == []
    finally:
        mgr.shutdown()
        pub.close()
        pool.shutdown()


def test_online_service_http(tmp_path):
    """The online scoring app scores over HTTP and exposes /metrics."""
    from starlette.testclient import TestClient

    sys.path.insert(0, str(REPO / "examples"))
    import online_scoring_service

    from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
    from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
    from llm_d_kv_cache_amd.events.publisher import (
        block_stored_payload,
        encode_batch,
    )

    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), ix)
    tokens = list(range(32))
    pool.process("kv@pod-h@m", 0,
                 encode_batch([block_stored_payload([1, 2], None, tokens, 16)]))

    app = online_scoring_service.build_app(ix, pool)
    client = TestClient(app)
    r = client.post("/score_completions", json={"tokens": tokens, "model": "m"})
    assert r.status_code == 200
    assert r.json()["scores"] == {"pod-h": 2.0}
    r = client.post("/score_completions", json={"model": "m"})
    assert r.status_code == 400
    r = client.get("/metrics")
    assert r.status_code == 200


def test_offload_connector_wiring(tmp_path):
    from llm_d_kv_cache_amd.offload.engine import OffloadEngineConfig
    from llm_d_kv_cache_amd.offload.file_mapper import KVCacheLayoutConfig
    from llm_d_kv_cache_amd.offload.spec import (
        OffloadConnector,
        OffloadConnectorConfig,
    )

    group = [torch.randint(0, 255, (64, 4096), dtype=torch.uint8)
             for _ in range(2)]
    conn = OffloadConnector(
        [group],
        OffloadConnectorConfig(
            root=str(tmp_path),
            layout=KVCacheLayoutConfig(model="conn-test"),
            engine=OffloadEngineConfig(io_threads=2, copy_path="host"),
            offloaded_block_tokens=128,
            group_block_tokens=(16,),
        ),
    )
    assert conn.blocks_per_file == [8]
    store, load = conn.get_handlers()
    mgr = conn.get_manager()
    store.transfer_async([9], {0: list(range(8))})
    deadline = time.time() + 10
    done = []
    while not done and time.time() < deadline:
        done = store.get_finished()
        time.sleep(0.01)
    assert done and done[0].success
    assert mgr.lookup([9, 10]) == 1
    # run manifest written
    import os

    assert os.path.exists(os.path.join(conn.mapper.run_dir, "config.json"))
    conn.close()


def test_connector_rejects_bad_geometry(tmp_path):
    from llm_d_kv_cache_amd.offload.engine import OffloadEngineConfig
    from llm_d_kv_cache_amd.offload.spec import (
        OffloadConnector,
        OffloadConnectorConfig,
    )

    group = [torch.zeros(8, 1024, dtype=torch.uint8)]
    with pytest.raises(ValueError):
        OffloadConnector(
            [group],
            OffloadConnectorConfig(
                root=str(tmp_path),
                engine=OffloadEngineConfig(copy_path="host"),
                offloaded_block_tokens=100,  # not a multiple of 16
                group_block_tokens=(16,),
            ),
        )


def test_epp_scorer_demo_runs():
    out = subprocess.run(
        [sys.executable, str(REPO / "examples" / "epp_scorer.py")],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr
    assert "ok" in out.stdout


def test_valkey_example_runs():
    out = subprocess.run(
        [sys.executable, str(REPO / "examples" / "valkey_example.py")],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr


def test_kv_cache_manager_facade():
    from llm_d_kv_cache_amd.events.publisher import EventPublisher, block_stored_payload
    from llm_d_kv_cache_amd.manager import KVCacheManager, KVCacheManagerConfig
    from llm_d_kv_cache_amd.events import EventPoolConfig

    mgr = KVCacheManager(KVCacheManagerConfig(
        events=EventPoolConfig(zmq_endpoint="tcp://127.0.0.1:0"),
        register_metrics=False,
    )).start()
    try:
        pub = EventPublisher(f"tcp://127.0.0.1:{mgr.events_port}", "pod-f", "m",
                             bind=False)
        deadline = time.time() + 10
        while pub._pub.peer_count < 1 and time.time() < deadline:
            time.sleep(0.01)
        time.sleep(0.2)
        pub.publish_events([block_stored_payload([1, 2], None, list(range(32)), 16)])
        deadline = time.time() + 10
        while mgr.events.stats().processed < 1 and time.time() < deadline:
            time.sleep(0.01)
        assert mgr.score_tokens(list(range(32)), "m") == {"pod-f": 2.0}
        pub.close()
    finally:
        mgr.shutdown()


def test_kv_events_offline_example():
    out = subprocess.run(
        [sys.executable, str(REPO / "examples" / "kv_events_offline.py")],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr
    assert out.stdout.strip().endswith("ok")


def test_indexer_service_client_example():
    """End to end: start the real gRPC service on a free port, score with
    the example client."""
    import time

    from llm_d_kv_cache_amd import ensure_native
    from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
    from llm_d_kv_cache_amd.services.indexer_service import create_server

    k = ensure_native()
    ix = KVCacheIndexer(IndexerConfig())
    keys = ix.compute_block_keys(list(range(48)), "demo-model")
    ix.index.add([], keys, [k.PodEntry("pod-a", "gpu")])
    server, port = create_server(ix, "127.0.0.1:0")
    server.start()
    try:
        sys.path.insert(0, str(REPO / "examples"))
        import indexer_service_client

        resp = indexer_service_client.main([
            "--target", f"127.0.0.1:{port}", "--model", "demo-model",
            "--tokens", ",".join(str(t) for t in range(48)),
            "--pods", "pod-a,pod-b"])
        scores = {s.pod_identifier: s.score for s in resp.scores}
        assert scores == {"pod-a": 3.0}
        assert resp.total_blocks == 3
    finally:
        server.stop(0)
