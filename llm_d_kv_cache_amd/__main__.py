"""Operational entry points: ``python -m llm_d_kv_cache_amd <cmd>``.

Counterpart of the reference's shipped binaries/images (Dockerfile
kv-cache-manager image + pvc_evictor + example service wiring): one front
door for the service, the evictor, and quick diagnostics.
"""
from __future__ import annotations

import argparse
import sys


def _cmd_serve(argv):
    """gRPC indexer service + ZMQ event ingestion (see
    examples/indexer_service_main.py for the library-level wiring)."""
    import os
    import time

    from .core import IndexerConfig, KVCacheIndexer
    from .events import EventPoolConfig, KVEventsPool
    from .services.indexer_service import create_server

    ap = argparse.ArgumentParser(prog="kvcache-amd serve")
    ap.add_argument("--grpc-port", type=int, default=50051)
    ap.add_argument("--zmq-endpoint", default="tcp://0.0.0.0:5557")
    ap.add_argument("--zmq-username", default="")
    ap.add_argument("--zmq-password", default="")
    ap.add_argument("--snapshot-path", default=None)
    ap.add_argument("--snapshot-interval", type=float, default=300.0)
    args = ap.parse_args(argv)

    indexer = KVCacheIndexer(IndexerConfig())
    if args.snapshot_path and os.path.exists(args.snapshot_path):
        indexer.load_index(args.snapshot_path)
        print(f"restored snapshot: {indexer.stats().keys} keys")
    pool = KVEventsPool(
        EventPoolConfig(zmq_endpoint=args.zmq_endpoint,
                        zmq_username=args.zmq_username,
                        zmq_password=args.zmq_password), indexer)
    pool.start()
    server, port = create_server(indexer, f"0.0.0.0:{args.grpc_port}")
    server.start()
    print(f"indexer service on :{port}, events on {args.zmq_endpoint}")
    try:
        while True:
            time.sleep(args.snapshot_interval if args.snapshot_path else 3600)
            if args.snapshot_path:
                indexer.save_index(args.snapshot_path)
    except KeyboardInterrupt:
        if args.snapshot_path:
            indexer.save_index(args.snapshot_path)
        server.stop(1.0)
        pool.shutdown()


def _cmd_score(argv):
    """One-shot ScoreTokens / GetPodScores against a running service."""
    import grpc

    from .services import proto

    ap = argparse.ArgumentParser(prog="kvcache-amd score")
    ap.add_argument("--target", default="127.0.0.1:50051")
    ap.add_argument("--model", default="demo-model")
    ap.add_argument("--tokens", default="", help="comma-separated token ids")
    ap.add_argument("--prompt", default="")
    ap.add_argument("--pods", default="")
    args = ap.parse_args(argv)
    chan = grpc.insecure_channel(args.target)
    g = proto.get
    pods = [p for p in args.pods.split(",") if p]
    if args.tokens:
        call = chan.unary_unary(
            "/indexerpb.IndexerService/ScoreTokens",
            request_serializer=g(
                "indexerpb.ScoreTokensRequest").SerializeToString,
            response_deserializer=g("indexerpb.ScoreResponse").FromString)
        req = g("indexerpb.ScoreTokensRequest")(
            tokens=[int(t) for t in args.tokens.split(",")],
            model_name=args.model, pod_identifiers=pods)
    else:
        call = chan.unary_unary(
            "/indexerpb.IndexerService/GetPodScores",
            request_serializer=g("indexerpb.ScoreRequest").SerializeToString,
            response_deserializer=g("indexerpb.ScoreResponse").FromString)
        req = g("indexerpb.ScoreRequest")(
            prompt=args.prompt, model_name=args.model, pod_identifiers=pods)
    resp = call(req, timeout=10.0)
    for sc in resp.scores:
        print(f"{sc.pod_identifier}\t{sc.score}")
    print(f"# blocks total={resp.total_blocks} hit={resp.hit_blocks}")
    chan.close()


def _cmd_evict(argv):
    """Storage-tier disk-space manager (PVC evictor)."""
    import logging

    from .evictor import EvictorConfig, PvcEvictor

    logging.basicConfig(level=logging.INFO)
    ap = argparse.ArgumentParser(prog="kvcache-amd evict")
    ap.add_argument("--root", default=None)
    args = ap.parse_args(argv)
    cfg = EvictorConfig.from_env()
    if args.root:
        cfg.root = args.root
    PvcEvictor(cfg).run_forever()


def _cmd_build(argv):
    """Compile the native extensions in place (gfx950)."""
    from ._build import build_all

    build_all()
    print("native extensions built")


def main():
    cmds = {"serve": _cmd_serve, "score": _cmd_score, "evict": _cmd_evict,
            "build": _cmd_build}
    if len(sys.argv) < 2 or sys.argv[1] in ("-h", "--help"):
        print("usage: python -m llm_d_kv_cache_amd "
              f"{{{','.join(cmds)}}} [args]\n")
        for name, fn in cmds.items():
            print(f"  {name:8s} {fn.__doc__.splitlines()[0]}")
        return 0
    cmd = sys.argv[1]
    if cmd not in cmds:
        print(f"unknown command: {cmd}", file=sys.stderr)
        return 2
    return cmds[cmd](sys.argv[2:])


if __name__ == "__main__":
    sys.exit(main() or 0)
