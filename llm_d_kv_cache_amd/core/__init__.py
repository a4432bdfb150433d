"""Core KV-block index + scoring API (read path).

Python facade over the native ``_kvcore`` module. Capability parity with
the reference ``pkg/kvcache`` (indexer.go, kvblock/): config-driven
construction, ``score_tokens`` = hash-chain -> index lookup -> longest
prefix scoring, with device-tier weights.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

from .. import ensure_native

DEFAULT_BLOCK_SIZE = 16

# Device tiers known to the scorer, with their score weights. "peer-gpu" is
# the MI355X-native addition: a block on a peer GPU of the same node is
# reachable over xGMI at near-HBM speed, so it scores just below local HBM.
DEFAULT_TIER_WEIGHTS = {
    "gpu": 1.0,
    "peer-gpu": 0.95,
    "cpu": 0.8,
    "shared_storage": 0.6,
    "object_store": 0.5,
}


@dataclass
class TokenProcessorConfig:
    block_size_tokens: int = DEFAULT_BLOCK_SIZE
    hash_seed: str = ""


@dataclass
class InMemoryIndexConfig:
    size: int = 100_000_000
    pod_cache_size: int = 10
    shards: int = 64


@dataclass
class CostAwareMemoryIndexConfig:
    """Byte-budget index (reference CostAwareMemoryIndex parity): evicts
    LRU keys until the approximate resident size fits max_bytes."""
    max_bytes: int = 2 * 1024**3  # reference default "2GiB"
    pod_cache_size: int = 10
    shards: int = 64


@dataclass
class RedisIndexConfig:
    """Network-backed shared index (Redis or Valkey — wire compatible)."""
    host: str = "127.0.0.1"
    port: int = 6379
    pool_size: int = 4
    key_prefix: str = "kv"
    # Placeholder mirroring the reference's experimental Valkey-over-RDMA
    # flag (redis.go:39-40): accepted, currently a no-op on this transport.
    enable_rdma: bool = False


@dataclass
class KVCacheBackendConfig:
    name: str
    weight: float


def default_backend_configs() -> List[KVCacheBackendConfig]:
    return [KVCacheBackendConfig(n, w) for n, w in DEFAULT_TIER_WEIGHTS.items()]


@dataclass
class IndexerConfig:
    token_processor: TokenProcessorConfig = field(default_factory=TokenProcessorConfig)
    # First-non-None backend wins (reference index.go:68-93 selection order:
    # cost-aware > valkey/redis > in-memory).
    cost_aware_index: Optional[CostAwareMemoryIndexConfig] = None
    redis_index: Optional[RedisIndexConfig] = None
    index: InMemoryIndexConfig = field(default_factory=InMemoryIndexConfig)
    backends: List[KVCacheBackendConfig] = field(default_factory=default_backend_configs)
    # Window-aware scoring (reference marks hybrid-aware scoring WIP): pods
    # whose learned KV-cache groups are all sliding-window get credit for a
    # cached tail covering the window even after out-of-window leading blocks
    # were evicted. Hints come from the attached events pool's GroupCatalog.
    window_aware_scoring: bool = True


class KVCacheIndexer:
    """Global KV-block index with prefix-aware pod scoring."""

    def __init__(self, config: Optional[IndexerConfig] = None):
        self.config = config or IndexerConfig()
        k = ensure_native()
        self._k = k
        self.token_processor = k.TokenProcessor(
            self.config.token_processor.block_size_tokens,
            self.config.token_processor.hash_seed,
        )
        if self.config.cost_aware_index is not None:
            c = self.config.cost_aware_index
            self.index = k.InMemoryIndex(
                pods_per_key=c.pod_cache_size, shards=c.shards,
                max_bytes=c.max_bytes,
            )
        elif self.config.redis_index is not None:
            c = self.config.redis_index
            self.index = k.RedisIndex(
                host=c.host, port=c.port, pool_size=c.pool_size,
                key_prefix=c.key_prefix,
            )
        else:
            self.index = k.InMemoryIndex(
                size=self.config.index.size,
                pods_per_key=self.config.index.pod_cache_size,
                shards=self.config.index.shards,
            )
        weights = {b.name: b.weight for b in self.config.backends}
        self._indexer = k.Indexer(self.token_processor, self.index, weights)
        self._events_pool = None  # set by KVEventsPool for window hints

    def _attach_pool(self, pool) -> None:
        self._events_pool = pool
        self._hint_cache: Dict[str, list] = {}
        self._hint_version = -1

    def _window_hints(self, pods: Sequence[str]):
        """Per-pod KV-cache group structure for hybrid-aware scoring:
        {pod: [(group_idx, window_blocks)]} with window_blocks 0 for
        full-attention groups; only pods whose structure the events pool
        has learned get hints (others keep the vanilla prefix walk)."""
        if not self.config.window_aware_scoring or self._events_pool is None:
            return {}
        native = self._events_pool.native
        if native.catalog_pods() == 0:
            return {}  # no HMA fields seen: zero per-request overhead
        ver = native.catalog_version()
        if ver != self._hint_version:
            bs = self.token_processor.block_size
            self._hint_cache = {
                pod: [(g, -(-w // bs) if w > 0 else 0)
                      for g, w in sorted(gw.items())]
                for pod, gw in native.catalog_snapshot().items()
            }
            self._hint_version = ver
        cache = self._hint_cache
        if not pods:  # unfiltered scoring: hint every cataloged pod
            return dict(cache)
        return {p: cache[p] for p in pods if p in cache}

    @property
    def block_size(self) -> int:
        return self.token_processor.block_size

    def score_tokens(
        self,
        tokens: Sequence[int],
        model_name: str,
        pod_identifiers: Sequence[str] = (),
        extra_features: Optional[list] = None,
    ) -> Dict[str, float]:
        """Score pods by longest cached prefix for this token stream."""
        res = self._indexer.score_tokens(
            tokens, model_name, list(pod_identifiers), extra_features,
            self._window_hints(pod_identifiers),
        )
        return dict(res.scores)

    def score_tokens_detailed(
        self,
        tokens: Sequence[int],
        model_name: str,
        pod_identifiers: Sequence[str] = (),
        extra_features: Optional[list] = None,
    ):
        """Like score_tokens but also returns (total_blocks, hit_blocks)."""
        res = self._indexer.score_tokens(
            tokens, model_name, list(pod_identifiers), extra_features,
            self._window_hints(pod_identifiers),
        )
        return dict(res.scores), res.total_blocks, res.hit_blocks

    def compute_block_keys(self, tokens: Sequence[int], model_name: str) -> List[int]:
        return list(self._indexer.compute_block_keys(tokens, model_name))

    def stats(self):
        return self.index.stats()

    def save_index(self, path: str) -> None:
        """Snapshot the in-memory index for warm restarts (the event
        stream still re-converges state; this just skips the cold-start
        window). Network-backed indexes are already durable."""
        save = getattr(self.index, "save", None)
        if save is None:
            raise RuntimeError(
                "index backend has no snapshot support (network-backed "
                "indexes are durable on the server side)")
        save(path)

    def load_index(self, path: str) -> None:
        """Merge a snapshot back in (see save_index)."""
        load = getattr(self.index, "load", None)
        if load is None:
            raise RuntimeError("index backend has no snapshot support")
        load(path)
