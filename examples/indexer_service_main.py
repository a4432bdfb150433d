#!/usr/bin/env python3
"""gRPC indexer service launcher (parity with the reference
examples/kv_cache_index_service/server): serves indexerpb.IndexerService
over TCP, with ZMQ event ingestion and an optional UDS tokenizer pool for
the prompt-scoring path.

Run: python examples/indexer_service_main.py --grpc-port 50051
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.services.indexer_service import create_server


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--grpc-port", type=int, default=50051)
    ap.add_argument("--zmq-endpoint", default="tcp://0.0.0.0:5557")
    ap.add_argument("--tokenizer-uds", default=None)
    ap.add_argument("--snapshot-path", default=None,
                    help="warm-restart snapshot: loaded at startup if "
                         "present, saved every --snapshot-interval seconds")
    ap.add_argument("--snapshot-interval", type=float, default=300.0)
    args = ap.parse_args()

    indexer = KVCacheIndexer(IndexerConfig())
    if args.snapshot_path and os.path.exists(args.snapshot_path):
        indexer.load_index(args.snapshot_path)
        print(f"restored index snapshot: {indexer.stats().keys} keys")
    pool = KVEventsPool(EventPoolConfig(zmq_endpoint=args.zmq_endpoint), indexer)
    pool.start()

    tok_pool = None
    if args.tokenizer_uds:
        from llm_d_kv_cache_amd.services.tokenizer_client import (
            TokenizationPool,
            UdsTokenizerClient,
        )

        tok_pool = TokenizationPool(UdsTokenizerClient(args.tokenizer_uds))

    server, port = create_server(indexer, f"0.0.0.0:{args.grpc_port}",
                                 tokenizer_pool=tok_pool)
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


if __name__ == "__main__":
    main()
