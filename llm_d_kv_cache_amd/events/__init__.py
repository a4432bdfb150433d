"""KVEvents write path: ZMTP subscribers + sharded ingestion pool.

Python facade over the native event plane. Capability parity with the
reference ``pkg/kvevents`` (pool.go, zmq_subscriber.go,
subscriber_manager.go): centralized bind topology or per-pod dial topology
(pod discovery), ordered per-pod processing, vLLM/SGLang wire format.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Dict, Optional

from .. import ensure_native


@dataclass
class EventPoolConfig:
    zmq_endpoint: str = "tcp://0.0.0.0:5557"
    topic_filter: str = "kv@"
    concurrency: int = 4
    engine_type: str = "vllm"  # wire layouts of vllm and sglang both decode
    discover_pods: bool = False  # False: bind fan-in; True: dial per pod
    # route per DP rank ("<pod>-dp<r>") when batches carry DataParallelRank
    dp_rank_routing: bool = False
    # per-shard backlog bound (0 = unbounded): floods drop oldest messages
    max_queue_depth: int = 0
    # ZMTP PLAIN credentials (empty username = NULL mechanism)
    zmq_username: str = ""
    zmq_password: str = ""


class KVEventsPool:
    """Sharded event-ingestion pool fed by one bound ZMTP SUB socket."""

    def __init__(self, config: EventPoolConfig, indexer) -> None:
        self.config = config
        k = ensure_native()
        self._k = k
        self._pool = k.EventPool(
            indexer.token_processor, indexer.index, config.concurrency,
            config.dp_rank_routing, config.max_queue_depth,
        )
        attach = getattr(indexer, "_attach_pool", None)
        if attach is not None:
            attach(self)  # enables window-aware scoring hints
        self._subscriber = None

    @property
    def native(self):
        return self._pool

    def start(self) -> None:
        self._pool.start()
        if not self.config.discover_pods:
            self._subscriber = self._k.Subscriber(
                self.config.zmq_endpoint,
                self.config.topic_filter,
                pool=self._pool,
                bind=True,
                username=self.config.zmq_username,
                password=self.config.zmq_password,
            )

    @property
    def port(self) -> Optional[int]:
        return self._subscriber.port if self._subscriber is not None else None

    def add_task(self, topic: str, seq: int, payload: bytes) -> None:
        self._pool.add_task(topic, seq, payload)

    def process(self, topic: str, seq: int, payload: bytes) -> None:
        """Synchronous processing (offline/batch ingestion path)."""
        self._pool.process(topic, seq, payload)

    def drain(self) -> None:
        self._pool.drain()

    def stats(self):
        return self._pool.stats()

    def group_metadata(self, pod: str, group: int):
        return self._pool.group_metadata(pod, group)

    def shutdown(self) -> None:
        if self._subscriber is not None:
            self._subscriber.close()
            self._subscriber = None
        self._pool.shutdown()


class SubscriberManager:
    """Per-pod subscriber lifecycle for the pod-discovery topology.

    Each engine pod binds its own PUB socket; every indexer replica dials
    every pod and converges independently (active-active HA).
    """

    def __init__(self, pool: KVEventsPool, topic_filter: str = "kv@",
                 reconnect_ms: int = 5000, username: str = "",
                 password: str = "") -> None:
        self._pool = pool
        self._topic = topic_filter
        self._reconnect_ms = reconnect_ms
        self._user = username or getattr(pool.config, "zmq_username", "")
        self._pass = password or getattr(pool.config, "zmq_password", "")
        self._k = ensure_native()
        self._subs: Dict[str, tuple] = {}  # pod -> (endpoint, subscriber)
        self._mu = threading.Lock()

    def ensure_subscriber(self, pod: str, endpoint: str) -> None:
        """Idempotent: re-dials only when the endpoint changed."""
        with self._mu:
            cur = self._subs.get(pod)
            if cur is not None:
                if cur[0] == endpoint:
                    return
                cur[1].close()
            sub = self._k.Subscriber(
                endpoint,
                self._topic,
                pool=self._pool.native,
                bind=False,
                reconnect_ms=self._reconnect_ms,
                username=self._user,
                password=self._pass,
            )
            self._subs[pod] = (endpoint, sub)

    def remove_subscriber(self, pod: str) -> None:
        with self._mu:
            cur = self._subs.pop(pod, None)
        if cur is not None:
            cur[1].close()

    def pods(self):
        with self._mu:
            return sorted(self._subs)

    def shutdown(self) -> None:
        with self._mu:
            subs = list(self._subs.values())
            self._subs.clear()
        for _, sub in subs:
            sub.close()
