#!/usr/bin/env python3
"""Pod-discovery reconciler driving the SubscriberManager.

Parity with the reference examples/kv_events/pod_reconciler: watch the
fleet's pod set and keep one ZMTP subscriber dialed per live engine pod
(active-active replicas each converge independently). In Kubernetes the
watch source is the API server; this example reconciles from a JSON file
(pod -> endpoint) so the loop is runnable and testable anywhere — swap
``load_pods`` for a k8s watch in production.

Run: python examples/pod_reconciler.py --pods-file pods.json
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import (
    EventPoolConfig,
    KVEventsPool,
    SubscriberManager,
)


def load_pods(path):
    try:
        with open(path) as f:
            return json.load(f)
    except (OSError, ValueError):
        return {}


def reconcile(manager: SubscriberManager, desired: dict) -> None:
    current = set(manager.pods())
    for pod, endpoint in desired.items():
        manager.ensure_subscriber(pod, endpoint)  # idempotent
    for pod in current - set(desired):
        manager.remove_subscriber(pod)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--pods-file", required=True)
    ap.add_argument("--interval", type=float, default=5.0)
    args = ap.parse_args()

    indexer = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(discover_pods=True), indexer)
    pool.start()
    manager = SubscriberManager(pool)
    print("reconciling from", args.pods_file)
    while True:
        reconcile(manager, load_pods(args.pods_file))
        time.sleep(args.interval)


if __name__ == "__main__":
    main()
