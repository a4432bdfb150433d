"""IndexerService: gRPC server wrapping the KVCacheIndexer.

Capability parity with the reference examples/kv_cache_index_service
(server.go) + api/indexerpb: GetPodScores (prompt path, tokenized via the
UDS sidecar pool) and ScoreTokens (token path). Runs over TCP or UDS.
"""
from __future__ import annotations

import logging
from concurrent import futures
from typing import Optional

import grpc

from ..core import KVCacheIndexer
from . import proto
from .tokenizer_client import TokenizationPool

log = logging.getLogger(__name__)


class IndexerServicer:
    def __init__(self, indexer: KVCacheIndexer,
                 tokenizer_pool: Optional[TokenizationPool] = None):
        self.indexer = indexer
        self.pool = tokenizer_pool

    def get_pod_scores(self, req, ctx):
        Resp = proto.get("indexerpb.ScoreResponse")
        Pod = proto.get("indexerpb.PodScore")
        if self.pool is None:
            ctx.abort(grpc.StatusCode.FAILED_PRECONDITION,
                      "prompt scoring requires a tokenizer pool; use ScoreTokens")
        try:
            tokens = self.pool.tokenize(req.model_name, req.prompt)
        except Exception as e:
            ctx.abort(grpc.StatusCode.UNAVAILABLE, f"tokenization failed: {e}")
        scores, total, hits = self.indexer.score_tokens_detailed(
            tokens, req.model_name, list(req.pod_identifiers))
        return Resp(
            scores=[Pod(pod_identifier=p, score=s) for p, s in sorted(scores.items())],
            total_blocks=total, hit_blocks=hits,
        )

    def score_tokens(self, req, ctx):
        Resp = proto.get("indexerpb.ScoreResponse")
        Pod = proto.get("indexerpb.PodScore")
        scores, total, hits = self.indexer.score_tokens_detailed(
            list(req.tokens), req.model_name, list(req.pod_identifiers))
        return Resp(
            scores=[Pod(pod_identifier=p, score=s) for p, s in sorted(scores.items())],
            total_blocks=total, hit_blocks=hits,
        )


def create_server(indexer: KVCacheIndexer, address: str,
                  tokenizer_pool: Optional[TokenizationPool] = None,
                  max_workers: int = 16):
    """address: 'host:port' or 'unix:///path'. Returns (server, bound_port)."""
    servicer = IndexerServicer(indexer, tokenizer_pool)
    g = proto.get
    handlers = {
        "GetPodScores": grpc.unary_unary_rpc_method_handler(
            servicer.get_pod_scores,
            request_deserializer=g("indexerpb.ScoreRequest").FromString,
            response_serializer=g("indexerpb.ScoreResponse").SerializeToString,
        ),
        "ScoreTokens": grpc.unary_unary_rpc_method_handler(
            servicer.score_tokens,
            request_deserializer=g("indexerpb.ScoreTokensRequest").FromString,
            response_serializer=g("indexerpb.ScoreResponse").SerializeToString,
        ),
    }
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    server.add_generic_rpc_handlers((
        grpc.method_handlers_generic_handler("indexerpb.IndexerService", handlers),
    ))
    port = server.add_insecure_port(address)
    return server, port


class IndexerClient:
    """Typed client for IndexerService."""

    def __init__(self, address: str):
        self._channel = grpc.insecure_channel(address)
        g = proto.get
        svc = "indexerpb.IndexerService"
        self._scores = self._channel.unary_unary(
            f"/{svc}/GetPodScores",
            request_serializer=g("indexerpb.ScoreRequest").SerializeToString,
            response_deserializer=g("indexerpb.ScoreResponse").FromString,
        )
        self._score_tokens = self._channel.unary_unary(
            f"/{svc}/ScoreTokens",
            request_serializer=g("indexerpb.ScoreTokensRequest").SerializeToString,
            response_deserializer=g("indexerpb.ScoreResponse").FromString,
        )

    def get_pod_scores(self, prompt: str, model_name: str, pods=(), timeout=10.0):
        Req = proto.get("indexerpb.ScoreRequest")
        resp = self._scores(Req(prompt=prompt, model_name=model_name,
                                pod_identifiers=list(pods)), timeout=timeout)
        return {p.pod_identifier: p.score for p in resp.scores}

    def score_tokens(self, tokens, model_name: str, pods=(), timeout=10.0):
        Req = proto.get("indexerpb.ScoreTokensRequest")
        resp = self._score_tokens(Req(tokens=list(tokens), model_name=model_name,
                                      pod_identifiers=list(pods)), timeout=timeout)
        return {p.pod_identifier: p.score for p in resp.scores}

    def close(self):
        self._channel.close()
