"""bench.py multi-rank path (the driver's 8-GPU scaling contract),
exercised here with world_size=2 on gloo + the host copy path
(KVC_BENCH_CPU_FULL=1). Covers rank coordination, barriers, the peer
phase, max-over-ranks reduction and the single-JSON-line contract."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.timeout(420)
def test_bench_world2_cpu_full(tmp_path):
    env = dict(os.environ)
    env.update({
        "KVC_BENCH_CPU_FULL": "1",
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": "29571",
    })
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29571", str(REPO / "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--root", str(tmp_path), "--device-blocks", "1536",
         "--io-threads", "2"],
        capture_output=True, text=True, timeout=400, env=env, cwd=str(REPO),
    )
    assert out.returncode == 0, f"stdout:\n{out.stdout}\nstderr:\n{out.stderr}"
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["metric"] == "kv_block_offload_GBps"
    assert d["value"] is not None and d["value"] > 0
    assert d["scaling"] == "weak"
    peer = d["aux"]["peer_xgmi"]
    assert peer["ok"] is True
    assert peer["n_pulls"] == 32
    assert peer["pull_GBps_aggregate"] > 0


@pytest.mark.timeout(240)
def test_bench_single_rank_cpu_control_plane():
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=220, cwd=str(REPO),
    )
    assert out.returncode == 0, out.stderr
    d = json.loads([ln for ln in out.stdout.splitlines() if ln.startswith("{")][0])
    assert d["aux"]["score_req_s"] > 100
    assert d["aux"]["ingest_batches_s"] > 100


@pytest.mark.timeout(600)
def test_bench_world8_tiny_cpu(tmp_path):
    """World-8 CPU sim of the driver's 8-GPU scaling run:
    the full bench path — rank coordination, barriers, the peer-mesh
    phase, max-over-ranks — at the scaling run's world size, tiny
    geometry."""
    env = dict(os.environ)
    env.update({
        "KVC_BENCH_CPU_FULL": "1",
        "KVC_BENCH_TINY": "1",
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": "29573",
    })
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29573", str(REPO / "bench.py"),
         "--gpus", "8", "--steps", "1", "--warmup", "0",
         "--root", str(tmp_path), "--device-blocks", "512",
         "--io-threads", "1"],
        capture_output=True, text=True, timeout=580, env=env, cwd=str(REPO),
    )
    assert out.returncode == 0, f"stdout:\n{out.stdout}\nstderr:\n{out.stderr}"
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["config"]["parallelism"] == "dp8"
    assert d["value"] is not None and d["value"] > 0
    peer = d["aux"]["peer_xgmi"]
    assert peer["ok"] is True, peer
    assert peer["pull_GBps_aggregate"] > 0
