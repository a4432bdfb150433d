#!/usr/bin/env python3
"""EPP-hop benchmark: IndexerService.ScoreTokens over real gRPC/TCP.

A real llm-d deployment consumes this framework's scorer through the
IndexerService hop (the reference is consumed in-process by the Go EPP;
here the EPP-equivalent calls gRPC — examples/epp_scorer.py). This
measures that hop end to end: client marshal -> TCP -> server -> native
score_tokens -> response, single-threaded and at 8 concurrent callers.

CPU-only (the control plane never touches the GPU): numbers measured in
the build container are representative.
"""
import os
import sys
import time
from concurrent.futures import ThreadPoolExecutor

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from llm_d_kv_cache_amd import ensure_native  # noqa: E402
from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer  # noqa: E402
from llm_d_kv_cache_amd.services.indexer_service import (  # noqa: E402
    IndexerClient,
    create_server,
)

N_TOKENS = int(os.environ.get("BENCH_TOKENS", "4096"))
N_CALLS = int(os.environ.get("BENCH_CALLS", "2000"))


def main():
    k = ensure_native()
    ix = KVCacheIndexer(IndexerConfig())
    tokens = list(range(N_TOKENS))
    keys = ix.compute_block_keys(tokens, "m")
    for p in range(64):
        ix.index.add([], keys[: 4 * (p % 64 + 1)],
                     [k.PodEntry(f"pod-{p}", "gpu")])

    server, port = create_server(ix, "127.0.0.1:0")
    server.start()
    client = IndexerClient(f"127.0.0.1:{port}")
    try:
        # warmup + correctness
        scores = client.score_tokens(tokens, "m")
        assert scores, "no scores through the hop"

        lats = []
        t0 = time.perf_counter()
        for _ in range(N_CALLS):
            t1 = time.perf_counter()
            client.score_tokens(tokens, "m")
            lats.append(time.perf_counter() - t1)
        dt = time.perf_counter() - t0
        lats.sort()
        print(f"grpc ScoreTokens 1 caller : {N_CALLS / dt:8.1f} req/s, "
              f"p50 {lats[len(lats) // 2] * 1e6:.0f} us, "
              f"p99 {lats[int(len(lats) * 0.99)] * 1e6:.0f} us "
              f"({N_TOKENS} tokens/call)")

        # concurrent callers (the EPP scores many requests in flight)
        clients = [IndexerClient(f"127.0.0.1:{port}") for _ in range(8)]
        per = N_CALLS // 8

        def run(c):
            for _ in range(per):
                c.score_tokens(tokens, "m")

        t0 = time.perf_counter()
        with ThreadPoolExecutor(8) as ex:
            list(ex.map(run, clients))
        dt = time.perf_counter() - t0
        print(f"grpc ScoreTokens 8 callers: {per * 8 / dt:8.1f} req/s "
              f"aggregate")
        for c in clients:
            c.close()

        # the in-process rate for comparison (what the hop costs)
        t0 = time.perf_counter()
        for _ in range(N_CALLS):
            ix.score_tokens(tokens, "m")
        dt = time.perf_counter() - t0
        print(f"in-process score_tokens   : {N_CALLS / dt:8.1f} req/s")
    finally:
        client.close()
        server.stop(0.1)


if __name__ == "__main__":
    main()
