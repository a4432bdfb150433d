#!/usr/bin/env python3
"""Flagship benchmark: KV-block offload GB/s on Llama-3-8B KV geometry.

One step = one steady-state offload cycle per GPU: store a fresh 16-block
set of KV files (GPU -> pinned host -> filesystem via the CDNA4 gather
kernel + SDMA), load them back (file -> host -> GPU scatter), and unlink
the previous generation. The headline value is the whole-job aggregate
offload throughput in GB/s across all ranks (weak scaling: per-GPU work is
fixed).

Score() and KVEvents-ingest rates — the control-plane north-star metrics —
are measured in separate bracketed phases and reported as aux fields.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N ...
"""
from __future__ import annotations

import argparse
import json
import os
import shutil
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

# KV geometry presets (--model). Values are the models' published configs;
# 70B is the per-rank TP=8 shard (1 of 8 KV heads), the deployment
# BASELINE.json names for the 128k-context fp8 config.
MODEL_PRESETS = {
    "llama-3-8b": {"model": "meta-llama/Llama-3-8B", "layers": 32, "kv_heads": 8},
    "llama-3-70b-tp8": {"model": "meta-llama/Llama-3-70B", "layers": 80,
                        "kv_heads": 1},
    # MLA: one compressed latent per token (kv_lora_rank 512 + rope 64),
    # not K+V heads — block_bytes computed from the latent width
    "deepseek-v3-mla": {"model": "deepseek-ai/DeepSeek-V3", "layers": 61,
                        "latent": 576},
}
MODEL = MODEL_PRESETS["llama-3-8b"]["model"]
NUM_LAYERS = 32
if os.environ.get("KVC_BENCH_TINY"):  # CPU rank-coordination shakeout only
    NUM_LAYERS = 4
KV_HEADS = 8
HEAD_SIZE = 128
BLOCK_TOKENS = 16
BLOCK_BYTES = 2 * BLOCK_TOKENS * KV_HEADS * HEAD_SIZE * 2  # K+V, bf16 = 64 KiB
FILES_PER_STEP = 64
if os.environ.get("KVC_BENCH_TINY"):
    FILES_PER_STEP = 16
BLOCKS_PER_FILE = 16  # 256-token offload chunks


def apply_model_preset(name):
    global MODEL, NUM_LAYERS, KV_HEADS, BLOCK_BYTES, FILES_PER_STEP
    p = MODEL_PRESETS[name]
    MODEL = p["model"]
    NUM_LAYERS = p["layers"] if not os.environ.get("KVC_BENCH_TINY") else 4
    if "latent" in p:  # MLA: latent vector per token, bf16
        KV_HEADS = 0
        BLOCK_BYTES = BLOCK_TOKENS * p["latent"] * 2
    else:
        KV_HEADS = p["kv_heads"]
        BLOCK_BYTES = 2 * BLOCK_TOKENS * KV_HEADS * HEAD_SIZE * 2
    if not os.environ.get("KVC_BENCH_TINY"):
        # keep ~2 GB of KV per step regardless of shard geometry so small-
        # file presets (70B-TP8: 10 MB files) don't just measure per-step
        # fixed overhead; the step is still whole files of the preset's
        # real size
        file_bytes = BLOCKS_PER_FILE * NUM_LAYERS * BLOCK_BYTES
        FILES_PER_STEP = max(16, min(256, (2 << 30) // file_bytes))


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(f"[bench] {msg}", file=sys.stderr, flush=True)


def pick_root():
    for cand in ("/dev/shm", "/tmp"):
        if os.path.isdir(cand):
            free = shutil.disk_usage(cand).free
            if free > 40 * 1024**3:
                return os.path.join(cand, "kvcache_bench")
    return os.path.join("/tmp", "kvcache_bench")


def find_nvme_root(min_free_gb=20):
    """A writable filesystem backed by a real block device (preferring
    NVMe), for the storage-tier-in-the-loop aux bench. Returns (path,
    device, is_nvme) or None — tmpfs/overlay mounts never qualify."""
    best = None
    try:
        with open("/proc/mounts") as f:
            for line in f:
                parts = line.split()
                if len(parts) < 3:
                    continue
                dev, mnt, fstype = parts[0], parts[1], parts[2]
                if fstype not in ("ext4", "xfs", "btrfs", "f2fs"):
                    continue
                if not os.access(mnt, os.W_OK):
                    continue
                try:
                    free = shutil.disk_usage(mnt).free
                except OSError:
                    continue
                if free < min_free_gb * 1024**3:
                    continue
                score = ("nvme" in dev, free)
                if best is None or score > best[0]:
                    best = (score, mnt, dev)
    except OSError:
        return None
    if best is None:
        return None
    return os.path.join(best[1], "kvcache_bench_nvme"), best[2], best[0][0]


def bench_offload_variant(group, root, host_cache_gb, direct_io, steps,
                          local_rank, io_threads, gpu):
    """Short sequential store->load cycle used for the aux variants (cache
    disabled / real NVMe): every timed byte crosses the storage tier the
    variant names. Returns GB/s and the phase breakdown."""
    from llm_d_kv_cache_amd.offload import (
        FileMapper,
        GPUToStorageHandler,
        KVCacheLayoutConfig,
        OffloadEngineConfig,
        StorageToGPUHandler,
        TorchOffloadEngine,
    )

    shutil.rmtree(root, ignore_errors=True)
    os.makedirs(root, exist_ok=True)
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=io_threads,
                            gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path="staged" if gpu else "host",
                            host_cache_bytes=int(host_cache_gb * 1024**3),
                            direct_io=direct_io, device=local_rank),
    )
    mapper = FileMapper(root, KVCacheLayoutConfig(model=MODEL))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])
    blocks_per_step = FILES_PER_STEP * BLOCKS_PER_FILE
    step_bytes = blocks_per_step * len(group) * BLOCK_BYTES

    def one_step(sid, timed):
        base = sid * FILES_PER_STEP + 1
        hashes = list(range(base, base + FILES_PER_STEP))
        ids = list(range(blocks_per_step))
        n = 0
        for i in range(0, FILES_PER_STEP, 8):
            store.transfer_async(
                hashes[i:i + 8],
                {0: ids[i * BLOCKS_PER_FILE:(i + 8) * BLOCKS_PER_FILE]})
            n += 1
        done = 0
        while done < n:
            done += len(store.get_finished())
            time.sleep(0.0002)
        for i in range(0, FILES_PER_STEP, 8):
            load.transfer_async(
                hashes[i:i + 8],
                {0: ids[i * BLOCKS_PER_FILE:(i + 8) * BLOCKS_PER_FILE]})
        done = 0
        while done < n:
            done += len(load.get_finished())
            time.sleep(0.0002)
        # delete the older generation so disk usage stays bounded
        prev = (sid - 1) * FILES_PER_STEP + 1
        if sid > -1:
            for h in range(prev, prev + FILES_PER_STEP):
                try:
                    os.unlink(mapper.file_name(h, 0))
                except OSError:
                    pass

    try:
        one_step(-1, timed=False)  # warmup
        import torch
        if gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for s in range(steps):
            one_step(s, timed=True)
        if gpu:
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        stats = eng.stats()
        return {
            "GBps": round(2 * step_bytes * steps / dt / 1e9, 2),
            "ms_per_step": round(dt / steps * 1e3, 1),
            "steps": steps,
            "host_cache_hits": stats.host_cache_hits,
            "files_written": stats.files_written,
            "files_read": stats.files_read,
        }
    finally:
        shutil.rmtree(root, ignore_errors=True)


def bench_control_plane():
    """Score req/s + p50, ingest events/s (CPU-side, per rank)."""
    import numpy as np

    from llm_d_kv_cache_amd import ensure_native
    from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
    from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
    from llm_d_kv_cache_amd.events.publisher import (
        block_stored_payload,
        encode_batch,
    )

    k = ensure_native()
    ix = KVCacheIndexer(IndexerConfig())
    tokens = np.arange(4096, dtype=np.uint32)
    keys = ix.compute_block_keys(tokens, MODEL)
    for p in range(64):
        ix.index.add([], keys[: 4 * (p % 64 + 1)], [k.PodEntry(f"pod-{p}", "gpu")])
    lat = []
    n = 3000
    t0 = time.perf_counter()
    for _ in range(n):
        t1 = time.perf_counter()
        ix.score_tokens(tokens, MODEL)
        lat.append(time.perf_counter() - t1)
    dt = time.perf_counter() - t0
    lat.sort()
    score_rps = n / dt
    score_p50_us = lat[n // 2] * 1e6

    pool = KVEventsPool(EventPoolConfig(concurrency=4), ix)
    pool._pool.start()
    payloads = []
    for p in range(8):
        topic = f"kv@pod-{p}@{MODEL}"
        for j in range(1000):
            payloads.append(
                (topic,
                 encode_batch([
                     block_stored_payload([j * 16 + x for x in range(8)], None,
                                          list(range(128)), 16)
                 ]))
            )
    t0 = time.perf_counter()
    for topic, pl in payloads:
        pool.add_task(topic, 0, pl)
    pool.drain()
    dt = time.perf_counter() - t0
    ingest_eps = len(payloads) / dt
    pool.shutdown()
    return {
        "score_req_s": round(score_rps, 1),
        "score_p50_us": round(score_p50_us, 1),
        "ingest_batches_s": round(ingest_eps, 1),
        "ingest_blocks_s": round(ingest_eps * 8, 1),
    }


def _trace(rank, msg):
    if os.environ.get("KVC_BENCH_TRACE"):
        print(f"[trace r{rank}] {msg}", file=sys.stderr, flush=True)


def _destroy_peer_groups(dist, *groups):
    # gloo subgroups left to interpreter-exit GC abort the process
    # sporadically ("terminate called without an active exception",
    # ~1/6 world-8 runs): destroy them in order while everything is alive
    for g in groups:
        try:
            dist.destroy_process_group(g)
        except Exception:
            pass


def bench_peer_phase(dist, torch, group, rank, world, local_rank):
    """Measure cross-GPU block-pull bandwidth over RCCL/xGMI (or gloo on
    CPU). Each rank pulls 16-block chunks from its ring neighbor.

    Hang-proof by construction: every rank executes the SAME collective
    sequence regardless of local failures — local errors are caught and
    agreed on via all-reduce(MIN) sync points, never by skipping a
    barrier (a skipped barrier would stall the whole scaling run until
    the NCCL watchdog)."""
    from llm_d_kv_cache_amd.peer import PeerMigrationService

    gpu = torch.cuda.is_available()
    dev = "cuda" if gpu else "cpu"
    ctrl_pg = dist.new_group(backend="gloo")
    data_pg = dist.new_group(backend="nccl" if gpu else "gloo")

    def agree(ok_local):
        t = torch.tensor([1 if ok_local else 0], device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        return bool(t.item())

    def bail(result):
        _destroy_peer_groups(dist, ctrl_pg, data_pg)
        return result

    _trace(rank, "peer: groups created")
    svc = None
    err = None
    try:
        svc = PeerMigrationService([group], data_group=data_pg,
                                   control_group=ctrl_pg,
                                   device=local_rank if gpu else 0)
    except Exception as e:
        err = f"service init: {e}"
    _trace(rank, "peer: service up")
    if not agree(svc is not None):
        if svc is not None:
            svc.close()
        return bail({"ok": False, "error": err or "peer rank failed init"})

    bpf = BLOCKS_PER_FILE
    nb = int(group[0].shape[0])
    # src chunks live in the head of the page pool, pull destinations in
    # the tail — sized to the ACTUAL pool so small --device-blocks runs
    # (CPU sims, tiny boxes) stay in range
    n_chunks = max(1, min(32, nb // (2 * bpf)))
    dst_base = nb - n_chunks * bpf
    src = (rank + 1) % world
    ok_local = True
    try:
        for c in range(n_chunks):
            svc.register_blocks(0xE000 + rank * 1000 + c, 0,
                                list(range(c * bpf, (c + 1) * bpf)))
    except Exception as e:
        ok_local, err = False, f"register: {e}"
    if not agree(ok_local):
        svc.close()
        return bail({"ok": False, "error": err or "peer rank failed register"})
    _trace(rank, "peer: registered")

    try:
        svc.pull(0xE000 + src * 1000, 0,
                 list(range(dst_base, dst_base + bpf)),
                 src_rank=src).result(timeout=120)  # warmup
    except Exception as e:
        ok_local, err = False, f"warmup pull: {e}"
    if not agree(ok_local):
        svc.close()
        return bail({"ok": False, "error": err or "peer rank failed warmup"})
    _trace(rank, "peer: warmup pull done")

    if gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    try:
        futs = [
            svc.pull(0xE000 + src * 1000 + c, 0,
                     list(range(dst_base + c * bpf, dst_base + (c + 1) * bpf)),
                     src_rank=src, timeout=120)
            for c in range(n_chunks)
        ]
        ok_local = all(f.result(timeout=150) for f in futs)
    except Exception as e:
        ok_local, err = False, f"pull: {e}"
    if gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    ok = agree(ok_local)
    _trace(rank, "peer: timed pulls done")
    pulled = svc.stats().bytes_received
    t = torch.tensor([dt], device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    if not ok:
        svc.close()
        return bail({"ok": False, "error": err or "peer pull failed on a rank"})

    # second round: fp8 on the wire (peer quantizes during the gather) —
    # same collective structure, half the data-plane bytes per pull.
    # The CPU-sim software codec is slow, so that path runs fewer pulls.
    n8 = n_chunks if gpu else 4
    raw_bytes = svc._copier.packed_bytes(0, bpf) * n8
    t1 = time.perf_counter()
    try:
        futs = [
            svc.pull(0xE000 + src * 1000 + c, 0,
                     list(range(dst_base + c * bpf, dst_base + (c + 1) * bpf)),
                     src_rank=src, timeout=120, fp8=True)
            for c in range(n8)
        ]
        ok_local = all(f.result(timeout=150) for f in futs)
    except Exception as e:
        ok_local, err = False, f"fp8 pull: {e}"
    if gpu:
        torch.cuda.synchronize()
    dt8 = time.perf_counter() - t1
    ok8 = agree(ok_local)
    wire8 = svc.stats().bytes_received - pulled
    t8 = torch.tensor([dt8], device=dev)
    dist.all_reduce(t8, op=dist.ReduceOp.MAX)
    _trace(rank, "peer: fp8 round done")
    svc.close()
    _trace(rank, "peer: closed")
    _destroy_peer_groups(dist, ctrl_pg, data_pg)
    _trace(rank, "peer: groups destroyed")
    out = {
        "ok": True,
        "pull_GBps_per_gpu": round(pulled / dt / 1e9, 2),
        "pull_GBps_aggregate": round(pulled * world / float(t.item()) / 1e9, 2),
        "chunk_bytes": pulled // n_chunks,
        "n_pulls": n_chunks,
    }
    if ok8:
        out["fp8_wire_GBps_per_gpu"] = round(wire8 / dt8 / 1e9, 2)
        out["fp8_logical_GBps_per_gpu"] = round(raw_bytes / dt8 / 1e9, 2)
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--root", type=str, default=None)
    ap.add_argument("--copy-path", type=str, default="staged",
                    choices=["staged", "zero_copy"])
    ap.add_argument("--serialize", type=str, default="raw",
                    choices=["raw", "fp8_e4m3"])
    ap.add_argument("--io-threads", type=int, default=32)
    ap.add_argument("--read-ratio", type=float, default=0.25,
                    help="read-preferring worker fraction (0.25 measured "
                         "best for duplex throughput: most workers pump "
                         "store D2Hs; the ENGINE default stays 0.75 for "
                         "read-latency QoS — profiles/r02_offload.md)")
    ap.add_argument("--device-blocks", type=int, default=2048)
    ap.add_argument("--model", type=str, default="llama-3-8b",
                    choices=sorted(MODEL_PRESETS))
    ap.add_argument("--host-cache-gb", type=float, default=None,
                    help="pinned-DRAM cache tier size (0 disables; default "
                         "sized to hold the generations the step pipeline "
                         "keeps live)")
    ap.add_argument("--write-policy", type=str, default="through",
                    choices=["through", "back"])
    ap.add_argument("--overlap", dest="overlap", action="store_true",
                    default=True,
                    help="full-duplex steady state (default): stores of "
                         "generation N run concurrently with loads of "
                         "generation N-1 (both PCIe directions busy; "
                         "disjoint GPU pages)")
    ap.add_argument("--no-overlap", dest="overlap", action="store_false",
                    help="sequential store-then-load phases")
    ap.add_argument("--pipeline", type=int, default=1,
                    help="steps kept in flight (overlap mode). 1 (default) "
                         "= step-synchronous; 2 = the next step's stores "
                         "start while the previous step's tail writes "
                         "drain (within noise of 1 when the DRAM tier "
                         "covers the window — profiles/r02_offload.md). "
                         "All work completes inside the timed region.")
    args = ap.parse_args()
    apply_model_preset(args.model)
    if args.host_cache_gb is None:
        # the tier must hold every generation still loadable inside the
        # pipeline window (store gen N while loading N-2, deletions lag
        # further) plus slack — one generation short and every load falls
        # back to file reads (measured: hits 1536 -> 106, 55 -> 24 GB/s)
        gens = 3 + 2 * (max(1, args.pipeline) - 1)
        gen_gb = FILES_PER_STEP * BLOCKS_PER_FILE * NUM_LAYERS * BLOCK_BYTES / 1e9
        args.host_cache_gb = round(gens * gen_gb, 1)

    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    dist = None
    if world > 1:
        # Fail FAST before rendezvous when this box cannot host the rank:
        # a rank that dies inside init_process_group leaves the others
        # hanging until the NCCL watchdog; exiting pre-rendezvous makes
        # torchrun kill the job immediately.
        if torch.cuda.is_available() and torch.cuda.device_count() <= local_rank:
            print(f"[bench] rank {rank}: LOCAL_RANK {local_rank} >= "
                  f"{torch.cuda.device_count()} visible GPUs — aborting "
                  f"before rendezvous", file=sys.stderr, flush=True)
            sys.exit(3)
        from datetime import timedelta

        import torch.distributed as tdist

        dist = tdist
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        # bounded: a wedged peer turns into a clean error, not a hang
        dist.init_process_group(backend=backend,
                                timeout=timedelta(seconds=300))

    gpu = torch.cuda.is_available()
    cpu_full = bool(os.environ.get("KVC_BENCH_CPU_FULL"))
    if not gpu and not cpu_full:
        log("no GPU: control-plane phases only; offload GB/s unmeasured")
        aux = bench_control_plane()
        if rank == 0:
            print(json.dumps({
                "metric": "kv_block_offload_GBps", "value": None, "unit": "GB/s",
                "n_gpus": 0, "steps": args.steps, "warmup": args.warmup,
                "ms_per_step": None, "higher_is_better": True, "scaling": "weak",
                "vs_baseline": None, "dtype": "bf16", "data": "synthetic",
                "aux": aux,
                "config": {"model": MODEL, "note": "cpu-only container"},
            }))
        return

    if gpu:
        torch.cuda.set_device(local_rank)
    from llm_d_kv_cache_amd.offload import (
        FileMapper,
        GPUToStorageHandler,
        KVCacheLayoutConfig,
        OffloadEngineConfig,
        StorageToGPUHandler,
        TorchOffloadEngine,
    )

    root = args.root or pick_root()
    rank_root = os.path.join(root, f"rank{rank}")
    shutil.rmtree(rank_root, ignore_errors=True)
    os.makedirs(rank_root, exist_ok=True)

    page_sets = 1
    if args.overlap:
        # store pages + one load window, or two alternating load windows
        # when steps pipeline (concurrent generations never scatter into
        # the same pages)
        page_sets = 2 if args.pipeline <= 1 else 3
    need = FILES_PER_STEP * BLOCKS_PER_FILE * page_sets
    if args.device_blocks < need:
        log(f"raising --device-blocks {args.device_blocks} -> {need} "
            f"(disjoint store/load page windows)")
        args.device_blocks = need

    # Llama-3-8B canonical KV: one group, 32 layers, (num_blocks, 128 KiB)
    log(f"allocating KV cache: {args.device_blocks} blocks x {NUM_LAYERS} layers "
        f"x {BLOCK_BYTES // 1024} KiB = "
        f"{args.device_blocks * NUM_LAYERS * BLOCK_BYTES / 1e9:.1f} GB")
    group = [
        torch.randint(0, 255, (args.device_blocks, BLOCK_BYTES), dtype=torch.uint8,
                      device="cuda" if gpu else "cpu")
        for _ in range(NUM_LAYERS)
    ]
    eng = TorchOffloadEngine(
        [group],
        OffloadEngineConfig(io_threads=args.io_threads,
                            gpu_blocks_per_file=BLOCKS_PER_FILE,
                            copy_path=args.copy_path if gpu else "host",
                            serialize=args.serialize,
                            host_cache_bytes=int(args.host_cache_gb * 1024**3),
                            write_policy=args.write_policy,
                            read_preferring_ratio=args.read_ratio,
                            device=local_rank),
    )
    mapper = FileMapper(rank_root, KVCacheLayoutConfig(
        model=MODEL, tp_size=1, kv_cache_groups=(("full_attention", 16, BLOCK_BYTES),),
    ))
    store = GPUToStorageHandler(eng, mapper, [BLOCKS_PER_FILE])
    load = StorageToGPUHandler(eng, mapper, [BLOCKS_PER_FILE])

    blocks_per_step = FILES_PER_STEP * BLOCKS_PER_FILE
    assert blocks_per_step <= args.device_blocks
    step_bytes = FILES_PER_STEP * BLOCKS_PER_FILE * NUM_LAYERS * BLOCK_BYTES

    phase_wall = {"store": 0.0, "load": 0.0, "unlink": 0.0}
    moved = {"bytes": 0}  # actual store+load volume issued in timed steps

    # Background deleters: steady-state disk management runs concurrently
    # with serving (the evictor's job in production), not on the hot path.
    import queue as _queue
    import threading as _threading

    del_q = _queue.Queue()

    def _deleter():
        while True:
            path = del_q.get()
            if path is None:
                return
            try:
                os.unlink(path)
            except OSError:
                pass

    deleters = [_threading.Thread(target=_deleter, daemon=True) for _ in range(4)]
    for t in deleters:
        t.start()

    pending = {"n": 0}
    store_jobs = {}        # generation -> store job id
    done_stores = set()    # completed store job ids

    def poll_done():
        fin_s = store.get_finished()
        done_stores.update(r.job_id for r in fin_s)
        pending["n"] -= len(fin_s) + len(load.get_finished())

    def drain_to(limit):
        poll_done()
        while pending["n"] > limit:
            time.sleep(0.0002)
            poll_done()

    def await_store(gen):
        # a generation's load must not race its own store (pipelined steps
        # drop the old per-step barrier that used to guarantee this)
        jid = store_jobs.get(gen)
        while jid is not None and jid not in done_stores:
            time.sleep(0.0002)
            poll_done()

    def run_step_overlap(step_id):
        # store gen step_id from pages [0, B); load gen step_id-1 into a
        # load window disjoint from the stores (and, when steps pipeline,
        # alternating by parity so concurrent generations never scatter
        # into the same pages)
        base = step_id * FILES_PER_STEP + 1
        hashes = list(range(base, base + FILES_PER_STEP))
        # pipelined steps load TWO generations back: gen N-1's store still
        # overlaps into step N, and gating the load on it would serialize
        # the two directions (measured 3x worse); gen N-2's store is
        # already done, so await_store is a no-op stall in steady state
        load_gen = step_id - (1 if args.pipeline <= 1 else 2)
        prev_base = load_gen * FILES_PER_STEP + 1
        prev_hashes = list(range(prev_base, prev_base + FILES_PER_STEP))
        do_load = load_gen >= -args.warmup
        p0 = time.perf_counter()
        # one submission per direction: tasks are per-FILE inside the
        # engine regardless of job granularity, and 2 python calls beat 16
        # (~100 us each) on the submission path
        all_ids = list(range(blocks_per_step))
        store_jobs[step_id] = store.transfer_async(hashes, {0: all_ids})
        pending["n"] += 1
        if do_load:
            await_store(load_gen)
            off = blocks_per_step
            if args.pipeline > 1:
                off = blocks_per_step * (1 + (step_id % 2))
            lids = [b + off for b in all_ids]
            load.transfer_async(prev_hashes, {0: lids})
            pending["n"] += 1
        # steady-state window: at most `pipeline` steps of jobs in flight
        drain_to(2 * (args.pipeline - 1))
        p1 = time.perf_counter()
        if step_id >= 0:
            moved["bytes"] += step_bytes + (step_bytes if do_load else 0)
        # deletion lags the pipeline window so a generation's files are
        # never unlinked while its load can still be in flight
        old_gen = step_id - 3 - max(0, args.pipeline - 1) - 1
        old = old_gen * FILES_PER_STEP + 1
        if old_gen >= -args.warmup:
            for h in range(old, old + FILES_PER_STEP):
                del_q.put(mapper.file_name(h, 0))
        phase_wall["store"] += p1 - p0

    def run_step(step_id):
        if args.overlap:
            return run_step_overlap(step_id)
        base = step_id * FILES_PER_STEP + 1
        hashes = list(range(base, base + FILES_PER_STEP))
        ids = list(range(blocks_per_step))
        n_jobs = 0
        p0 = time.perf_counter()
        # store in 8-file jobs to pipeline the I/O pool
        for i in range(0, FILES_PER_STEP, 8):
            store.transfer_async(hashes[i:i + 8], {0: ids[i * BLOCKS_PER_FILE:(i + 8) * BLOCKS_PER_FILE]})
            n_jobs += 1
        done = 0
        while done < n_jobs:
            done += len(store.get_finished())
            if done < n_jobs:
                time.sleep(0.0002)
        p1 = time.perf_counter()
        for i in range(0, FILES_PER_STEP, 8):
            load.transfer_async(hashes[i:i + 8], {0: ids[i * BLOCKS_PER_FILE:(i + 8) * BLOCKS_PER_FILE]})
        done = 0
        while done < n_jobs:
            done += len(load.get_finished())
            if done < n_jobs:
                time.sleep(0.0002)
        p2 = time.perf_counter()
        if step_id >= 0:
            moved["bytes"] += 2 * step_bytes
        # drop the previous generation in the background (evictor's role)
        prev = (step_id - 1) * FILES_PER_STEP + 1
        if step_id > 0:
            for h in range(prev, prev + FILES_PER_STEP):
                del_q.put(mapper.file_name(h, 0))
        p3 = time.perf_counter()
        phase_wall["store"] += p1 - p0
        phase_wall["load"] += p2 - p1
        phase_wall["unlink"] += p3 - p2

    def barrier():
        if dist is not None:
            dist.barrier()
        if gpu:
            torch.cuda.synchronize()

    for sid in range(-args.warmup, 0):  # ascending: overlap loads gen-1
        run_step(sid)
    if args.overlap:
        drain_to(0)  # flush before warmup-generation cleanup
    # drop warmup generations BEFORE the timed region starts (pre-existing
    # state, not steady-state work; one generation per step remains inside).
    # Overlap mode keeps the newest warmup generation: timed step 0 loads it.
    keep = set()
    if args.overlap:
        # the first timed steps still load these warmup generations
        keep = {-1} if args.pipeline <= 1 else {-1, -2}
    for gen in range(-args.warmup, 0):
        if gen in keep:
            continue
        base = gen * FILES_PER_STEP + 1
        for h in range(base, base + FILES_PER_STEP):
            del_q.put(mapper.file_name(h, 0))
    while not del_q.empty():
        time.sleep(0.005)
    log("warmup done")

    barrier()
    t0 = time.perf_counter()
    for s in range(args.steps):
        run_step(s)
    if args.overlap:
        drain_to(0)  # the pipeline tail is part of the timed work
    # the deletion backlog is part of the steady-state work: drain it
    # inside the timed region
    while not del_q.empty():
        time.sleep(0.001)
    barrier()
    elapsed = time.perf_counter() - t0
    for t in deleters:
        del_q.put(None)

    # max over ranks
    if dist is not None:
        t = torch.tensor([elapsed],
                         device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # actual store + load volume per rank (overlap mode's first step has
    # no generation to load when warmup == 0)
    moved_bytes = moved["bytes"]
    total_gbps = moved_bytes * world / elapsed / 1e9
    stats = eng.stats()

    # xGMI peer-migration phase (aux, outside the headline timed region):
    # each rank pulls its neighbor's cached chunks over RCCL send/recv.
    _trace(rank, "main: timed steps done")
    peer_aux = None
    if world > 1 and dist is not None:
        try:
            peer_aux = bench_peer_phase(dist, torch, group, rank, world,
                                        local_rank)
        except Exception as e:  # aux phase must never sink the headline run
            log(f"peer phase failed: {e}")
            peer_aux = {"ok": False, "error": str(e)}

    _trace(rank, "main: peer phase done")
    aux = bench_control_plane() if rank == 0 else None
    _trace(rank, "main: control plane done")
    if aux is not None and peer_aux is not None:
        aux["peer_xgmi"] = peer_aux

    # Storage-tier-in-the-loop variants: the headline's
    # DRAM-tier cache serves most loads, so report (a) a cache-disabled run
    # where every load reads the file and (b) a real-NVMe O_DIRECT run when
    # the box has one — probed honestly, skipped honestly.
    if aux is not None and world == 1 and gpu:
        try:
            aux["no_cache"] = bench_offload_variant(
                group, os.path.join(root, "nocache"), host_cache_gb=0,
                direct_io=False, steps=4, local_rank=local_rank,
                io_threads=args.io_threads, gpu=gpu)
        except Exception as e:
            aux["no_cache"] = {"error": str(e)}
        nvme = find_nvme_root()
        if nvme is None:
            aux["nvme"] = {"skipped": "no NVMe/block-device filesystem "
                                      "with >=20 GB free on this box"}
        else:
            nvme_root, nvme_dev, is_nvme = nvme
            try:
                aux["nvme"] = bench_offload_variant(
                    group, nvme_root, host_cache_gb=0, direct_io=True,
                    steps=3, local_rank=local_rank,
                    io_threads=args.io_threads, gpu=gpu)
                aux["nvme"]["device"] = nvme_dev
                aux["nvme"]["is_nvme"] = is_nvme
            except Exception as e:
                aux["nvme"] = {"error": str(e), "device": nvme_dev}

    shutil.rmtree(rank_root, ignore_errors=True)
    if rank == 0:
        print(json.dumps({
            "metric": "kv_block_offload_GBps",
            "value": round(total_gbps, 2),
            "unit": "GB/s",
            "n_gpus": world if gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "aux": aux,
            "config": {
                "model": MODEL,
                "global_batch": FILES_PER_STEP * world,
                "seq_len": FILES_PER_STEP * BLOCKS_PER_FILE * BLOCK_TOKENS,
                "parallelism": f"dp{world}",
                "kv_geometry": f"{NUM_LAYERS}L x {KV_HEADS}H x {HEAD_SIZE} bf16, "
                               f"{BLOCK_TOKENS}-token blocks",
                "bytes_per_step_per_gpu": step_bytes * 2,
                "copy_path": args.copy_path,
                "serialize": args.serialize,
                "io_threads": args.io_threads,
                "host_cache_gb": args.host_cache_gb,
                "write_policy": args.write_policy,
                "overlap": args.overlap,
                "pipeline": args.pipeline,
                "writeback_flushes": stats.writeback_flushes,
                "host_cache_hits": stats.host_cache_hits,
                "host_cache_stores": stats.host_cache_stores,
                "root": root,
                "files_written": stats.files_written,
                "engine_avg_write_ms": round(stats.avg_write_seconds * 1e3, 3),
                "phase_wall_ms_per_step": {
                    k: round(v / (args.steps + args.warmup) * 1e3, 1)
                    for k, v in phase_wall.items()
                },
                "phase_ms": {
                    "gather": round(stats.t_gather_ms, 1),
                    "d2h": round(stats.t_d2h_ms, 1),
                    "write": round(stats.t_write_ms, 1),
                    "read": round(stats.t_read_ms, 1),
                    "h2d": round(stats.t_h2d_ms, 1),
                    "scatter": round(stats.t_scatter_ms, 1),
                },
                # PCIe lane truth: busy wall + bytes per direction — the
                # direct measure of wire feeding vs the ~104 GB/s duplex
                # ceiling (profiles/r02_kernels.md §4)
                "lanes": {
                    "d2h_busy_ms": round(stats.d2h_lane_busy_ms, 1),
                    "h2d_busy_ms": round(stats.h2d_lane_busy_ms, 1),
                    "d2h_GB": round(stats.d2h_lane_bytes / 1e9, 2),
                    "h2d_GB": round(stats.h2d_lane_bytes / 1e9, 2),
                },
            },
        }))
    if dist is not None:
        dist.destroy_process_group()
    _trace(rank, "main: exit")


if __name__ == "__main__":
    main()
