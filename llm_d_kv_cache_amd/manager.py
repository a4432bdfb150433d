"""One-stop KV-cache manager facade: indexer + event ingestion + metrics.

Library-level equivalent of the reference's online wiring
(examples/kv_events/online/main.go setup + Indexer.Run lifecycle): one
object owns the whole control plane with start()/shutdown() and the
scoring surface.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional, Sequence

from .core import IndexerConfig, KVCacheIndexer
from .events import EventPoolConfig, KVEventsPool, SubscriberManager


@dataclass
class KVCacheManagerConfig:
    indexer: IndexerConfig = field(default_factory=IndexerConfig)
    events: EventPoolConfig = field(default_factory=EventPoolConfig)
    register_metrics: bool = True
    enable_tracing: bool = False


class KVCacheManager:
    def __init__(self, config: Optional[KVCacheManagerConfig] = None):
        self.config = config or KVCacheManagerConfig()
        self.indexer = KVCacheIndexer(self.config.indexer)
        self.events = KVEventsPool(self.config.events, self.indexer)
        self.subscribers: Optional[SubscriberManager] = None
        if self.config.events.discover_pods:
            self.subscribers = SubscriberManager(
                self.events, topic_filter=self.config.events.topic_filter)
        self._scorer = self.indexer
        if self.config.enable_tracing:
            from .utils.tracing import TracedIndexer, init_tracing

            self._scorer = TracedIndexer(self.indexer, init_tracing())
        self._metrics_collector = None
        self._started = False

    def start(self) -> "KVCacheManager":
        if self._started:
            return self
        self.events.start()
        if self.config.register_metrics:
            try:
                from .utils.metrics import register

                self._metrics_collector = register(
                    indexer=self.indexer, events_pool=self.events)
            except (ImportError, ValueError):
                pass  # prometheus absent or collector already registered
        self._started = True
        return self

    # ---- scoring surface ----------------------------------------------------

    def score_tokens(self, tokens: Sequence[int], model_name: str,
                     pod_identifiers: Sequence[str] = (),
                     extra_features=None) -> Dict[str, float]:
        return self._scorer.score_tokens(tokens, model_name, pod_identifiers,
                                         extra_features)

    # ---- pod discovery ------------------------------------------------------

    def ensure_pod(self, pod: str, endpoint: str) -> None:
        if self.subscribers is None:
            raise RuntimeError("pod discovery is off (events.discover_pods)")
        self.subscribers.ensure_subscriber(pod, endpoint)

    def remove_pod(self, pod: str) -> None:
        if self.subscribers is not None:
            self.subscribers.remove_subscriber(pod)

    @property
    def events_port(self) -> Optional[int]:
        return self.events.port

    def shutdown(self) -> None:
        if self.subscribers is not None:
            self.subscribers.shutdown()
        self.events.shutdown()
        self._started = False
