#!/usr/bin/env python3
"""Dummy engine-fleet publisher for load tests and by-hand e2e.

Parity with the reference examples/helper/publisher.go: emits a stream of
vLLM-wire-format BlockStored/BlockRemoved batches for N synthetic pods.

Run: python examples/fleet_publisher.py --endpoint tcp://127.0.0.1:5557 \
        --pods 4 --rate 100
"""
import argparse
import os
import random
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from llm_d_kv_cache_amd.events.publisher import (
    EventPublisher,
    block_removed_payload,
    block_stored_payload,
)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--endpoint", default="tcp://127.0.0.1:5557")
    ap.add_argument("--model", default="meta-llama/Llama-3.1-8B-Instruct")
    ap.add_argument("--pods", type=int, default=4)
    ap.add_argument("--rate", type=float, default=100.0, help="batches/s")
    ap.add_argument("--block-size", type=int, default=16)
    args = ap.parse_args()

    pubs = [EventPublisher(args.endpoint, f"pod-{i}", args.model, bind=False)
            for i in range(args.pods)]
    time.sleep(0.5)  # handshakes
    print(f"publishing to {args.endpoint} at {args.rate}/s")
    next_hash = 1
    chains = {i: (0, []) for i in range(args.pods)}  # pod -> (parent, tokens)
    while True:
        pod = random.randrange(args.pods)
        pub = pubs[pod]
        if random.random() < 0.9:
            n_blocks = random.randint(1, 8)
            tokens = [random.randrange(128000)
                      for _ in range(n_blocks * args.block_size)]
            hashes = list(range(next_hash, next_hash + n_blocks))
            next_hash += n_blocks
            parent, _ = chains[pod]
            pub.publish_events([
                block_stored_payload(hashes, parent or None, tokens,
                                     args.block_size)
            ])
            chains[pod] = (hashes[-1], tokens)
        else:
            parent, _ = chains[pod]
            if parent:
                pub.publish_events([block_removed_payload([parent])])
                chains[pod] = (0, [])
        time.sleep(1.0 / args.rate)


if __name__ == "__main__":
    main()
