#!/usr/bin/env python3
"""KV-cache-aware scorer plugin for an inference scheduler (EPP).

Parity with the reference examples/kv_cache_aware_scorer
(kvcache_aware_scorer.go:192-253): a Score(request, candidate_pods) hook
an endpoint-picker calls per routing decision, with TTL-cached per-pod
event subscriptions — the scorer dials a pod's KVEvents publisher the
first time the pod shows up as a candidate and drops subscriptions for
pods that stop appearing.

Library-style: embed KVCacheAwareScorer into the scheduler process.
"""
import os
import sys
import time
from typing import Dict, List, Optional, Sequence

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import (
    EventPoolConfig,
    KVEventsPool,
    SubscriberManager,
)


class KVCacheAwareScorer:
    def __init__(self, indexer: Optional[KVCacheIndexer] = None,
                 subscription_ttl_s: float = 300.0,
                 events_port: int = 5557):
        self.indexer = indexer or KVCacheIndexer(IndexerConfig())
        self.pool = KVEventsPool(EventPoolConfig(discover_pods=True),
                                 self.indexer)
        self.pool.start()
        self.manager = SubscriberManager(self.pool)
        self.ttl = subscription_ttl_s
        self.events_port = events_port
        self._last_seen: Dict[str, float] = {}

    def _ensure_subscriptions(self, pods: Sequence[str],
                              endpoints: Optional[Dict[str, str]] = None):
        now = time.time()
        for pod in pods:
            endpoint = (endpoints or {}).get(
                pod, f"tcp://{pod}:{self.events_port}")
            self.manager.ensure_subscriber(pod, endpoint)
            self._last_seen[pod] = now
        # TTL expiry: pods that stopped appearing as candidates
        for pod, seen in list(self._last_seen.items()):
            if now - seen > self.ttl:
                self.manager.remove_subscriber(pod)
                del self._last_seen[pod]

    def score(self, tokens: Sequence[int], model: str,
              candidate_pods: Sequence[str],
              endpoints: Optional[Dict[str, str]] = None) -> Dict[str, float]:
        """Returns pod -> longest-cached-prefix score; pods without any
        cached prefix score 0 (every candidate appears in the result)."""
        self._ensure_subscriptions(candidate_pods, endpoints)
        scores = self.indexer.score_tokens(tokens, model, candidate_pods)
        return {p: scores.get(p, 0.0) for p in candidate_pods}

    def mark_scheduled(self, tokens: Sequence[int], model: str,
                       pod: str) -> None:
        """Speculative stickiness (reference scorer parity): after routing
        a request to `pod`, predictively index its prefix as speculative
        entries so follow-up requests with the same prefix route sticky
        before the engine's confirming KVEvents arrive. The real events
        later overwrite the speculative entries in place."""
        from llm_d_kv_cache_amd import ensure_native

        k = ensure_native()
        keys = self.indexer.compute_block_keys(tokens, model)
        if keys:
            self.indexer.index.add(
                [], keys, [k.PodEntry(pod, "gpu", speculative=True)])

    def shutdown(self):
        self.manager.shutdown()
        self.pool.shutdown()


if __name__ == "__main__":
    # smoke demo with an in-process fleet
    from llm_d_kv_cache_amd.events.publisher import (
        EventPublisher,
        block_stored_payload,
    )

    scorer = KVCacheAwareScorer(subscription_ttl_s=60)
    pub = EventPublisher("tcp://127.0.0.1:0", "127.0.0.1", "m", bind=True)
    scorer.score(list(range(32)), "m", ["127.0.0.1"],
                 endpoints={"127.0.0.1": f"tcp://127.0.0.1:{pub.port}"})
    time.sleep(0.3)
    pub.publish_events([block_stored_payload([1, 2], None, list(range(32)), 16)])
    time.sleep(0.3)
    out = scorer.score(list(range(32)), "m", ["127.0.0.1", "other-pod"],
                       endpoints={"127.0.0.1": f"tcp://127.0.0.1:{pub.port}",
                                  "other-pod": "tcp://127.0.0.1:1"})
    print("scores:", out)
    assert out["127.0.0.1"] == 2.0 and out["other-pod"] == 0.0
    # speculative stickiness: a fresh prefix routed to other-pod scores
    # there before any engine event confirms it
    fresh = list(range(500, 532))
    scorer.mark_scheduled(fresh, "m", "other-pod")
    out2 = scorer.score(fresh, "m", ["127.0.0.1", "other-pod"],
                        endpoints={"127.0.0.1": f"tcp://127.0.0.1:{pub.port}",
                                   "other-pod": "tcp://127.0.0.1:1"})
    assert out2["other-pod"] == 2.0
    print("speculative stickiness ok")
    scorer.shutdown()
    pub.close()
    print("ok")
