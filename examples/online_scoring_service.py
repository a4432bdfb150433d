#!/usr/bin/env python3
"""Online scoring service: HTTP + ZMQ event ingestion + Prometheus.

Parity with the reference examples/kv_events/online/main.go:
  POST /score_completions       {"prompt"|"tokens", "model", "pods"?}
  POST /score_chat_completions  {"messages", "model", "pods"?}
  GET  /metrics                 Prometheus exposition

Events arrive on the bound ZMTP SUB endpoint (engines publish to it).

Run: python examples/online_scoring_service.py --http-port 8080 \
        --zmq-endpoint tcp://0.0.0.0:5557 [--tokenizer-uds /tmp/tok.sock]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from fastapi import FastAPI, HTTPException
from prometheus_client import generate_latest
import uvicorn

from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.utils.metrics import register
from llm_d_kv_cache_amd.utils.tracing import TracedIndexer, init_tracing


def build_app(indexer, pool, tokenizer_pool=None):
    from starlette.responses import PlainTextResponse

    app = FastAPI(title="kv-cache scoring")
    traced = TracedIndexer(indexer, init_tracing())

    @app.post("/score_completions")
    async def score_completions(body: dict):
        model = body.get("model", "")
        pods = body.get("pods", [])
        if "tokens" in body:
            tokens = body["tokens"]
        elif "prompt" in body and tokenizer_pool is not None:
            tokens = tokenizer_pool.tokenize(model, body["prompt"])
        else:
            raise HTTPException(400, "need 'tokens' (or 'prompt' + tokenizer)")
        return {"scores": traced.score_tokens(tokens, model, pods)}

    @app.post("/score_chat_completions")
    async def score_chat(body: dict):
        if tokenizer_pool is None:
            raise HTTPException(400, "chat scoring requires the tokenizer sidecar")
        model = body.get("model", "")
        msgs = [(m["role"], m["content"]) for m in body.get("messages", [])]
        ids, _ = tokenizer_pool._client.render_chat(model, msgs)
        return {"scores": traced.score_tokens(ids, model, body.get("pods", []))}

    @app.get("/metrics")
    async def metrics():
        return PlainTextResponse(generate_latest().decode())

    return app


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--http-port", type=int, default=8080)
    ap.add_argument("--zmq-endpoint", default="tcp://0.0.0.0:5557")
    ap.add_argument("--tokenizer-uds", default=None)
    ap.add_argument("--block-size", type=int, default=16)
    args = ap.parse_args()

    from llm_d_kv_cache_amd.core import TokenProcessorConfig

    indexer = KVCacheIndexer(IndexerConfig(
        token_processor=TokenProcessorConfig(block_size_tokens=args.block_size)))
    pool = KVEventsPool(EventPoolConfig(zmq_endpoint=args.zmq_endpoint), indexer)
    pool.start()
    register(indexer=indexer, events_pool=pool)

    tok_pool = None
    if args.tokenizer_uds:
        from llm_d_kv_cache_amd.services.tokenizer_client import (
            TokenizationPool,
            UdsTokenizerClient,
        )

        tok_pool = TokenizationPool(UdsTokenizerClient(args.tokenizer_uds))

    app = build_app(indexer, pool, tok_pool)
    uvicorn.run(app, host="0.0.0.0", port=args.http_port)


if __name__ == "__main__":
    main()
