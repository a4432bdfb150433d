#!/usr/bin/env python3
"""gRPC indexer-service client (parity with the reference
examples/kv_cache_index_service/client): dials indexerpb.IndexerService
and calls ScoreTokens / GetPodScores.

Run: python examples/indexer_service_client.py --target 127.0.0.1:50051 \
         --tokens 1,2,3,4 --model m --pods pod-a,pod-b
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import grpc

from llm_d_kv_cache_amd.services import proto


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--target", default="127.0.0.1:50051")
    ap.add_argument("--model", default="demo-model")
    ap.add_argument("--tokens", default="", help="comma-separated token ids")
    ap.add_argument("--prompt", default="", help="prompt string (needs the "
                    "service to run with a tokenizer pool)")
    ap.add_argument("--pods", default="", help="comma-separated candidates")
    args = ap.parse_args(argv)

    chan = grpc.insecure_channel(args.target)
    g = proto.get
    pods = [p for p in args.pods.split(",") if p]
    if args.tokens:
        call = chan.unary_unary(
            "/indexerpb.IndexerService/ScoreTokens",
            request_serializer=g(
                "indexerpb.ScoreTokensRequest").SerializeToString,
            response_deserializer=g("indexerpb.ScoreResponse").FromString,
        )
        req = g("indexerpb.ScoreTokensRequest")(
            tokens=[int(t) for t in args.tokens.split(",")],
            model_name=args.model, pod_identifiers=pods)
    else:
        call = chan.unary_unary(
            "/indexerpb.IndexerService/GetPodScores",
            request_serializer=g("indexerpb.ScoreRequest").SerializeToString,
            response_deserializer=g("indexerpb.ScoreResponse").FromString,
        )
        req = g("indexerpb.ScoreRequest")(
            prompt=args.prompt, model_name=args.model, pod_identifiers=pods)
    resp = call(req, timeout=10.0)
    for s in resp.scores:
        print(f"{s.pod_identifier}\t{s.score}")
    print(f"# blocks total={resp.total_blocks} hit={resp.hit_blocks}")
    chan.close()
    return resp


if __name__ == "__main__":
    main()
