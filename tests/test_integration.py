"""Cross-feature integration: the full control plane with every major
feature engaged at once — Redis-backed shared index, ZMTP transport,
hybrid (sliding-window) group scoring, DP-rank routing, speculative
stickiness, eviction and pod clears — exercised through the real wire
formats end to end."""
import sys
import time

import pytest

sys.path.insert(0, __file__.rsplit("/", 1)[0])
from fake_redis import FakeRedis

from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import (
    IndexerConfig,
    KVCacheIndexer,
    RedisIndexConfig,
)
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.events.publisher import (
    EventPublisher,
    all_blocks_cleared_payload,
    block_removed_payload,
    block_stored_payload,
)

MODEL = "int-model"


@pytest.fixture
def redis_srv():
    srv = FakeRedis()
    yield srv
    srv.close()


def test_full_control_plane(redis_srv):
    k = ensure_native()
    cfg = IndexerConfig(
        redis_index=RedisIndexConfig(host="127.0.0.1", port=redis_srv.port))
    ix = KVCacheIndexer(cfg)
    pool = KVEventsPool(EventPoolConfig(dp_rank_routing=True), ix)
    pool.start()
    port = pool.port

    tokens = list(range(128))  # 8 blocks

    # pod-a: plain full-attention pod, publishes over real ZMTP
    pub_a = EventPublisher("tcp://127.0.0.1:0", "pod-a", MODEL, bind=True)
    # (dial the pool's bound SUB? pool binds; publishers dial it)
    pub_a.close()
    pub_a = EventPublisher(f"tcp://127.0.0.1:{port}", "pod-a", MODEL,
                           bind=False)
    # pod-b: pure sliding-window pod with DP rank 2 -> routed pod-b-dp2
    pub_b = EventPublisher(f"tcp://127.0.0.1:{port}", "pod-b", MODEL,
                           bind=False)
    # dialing PUBs drop messages until the SUB handshake lands: retry the
    # idempotent stores until the index reflects both pods
    expected = {"pod-a": 8.0, "pod-b-dp2": 8.0}
    deadline = time.time() + 30
    while time.time() < deadline:
        pub_a.publish_events([
            block_stored_payload(list(range(1, 9)), None, tokens, 16)])
        pub_b.publish_events(
            [block_stored_payload(list(range(11, 19)), None, tokens, 16,
                                  group_idx=0, spec_kind="sliding_window",
                                  sliding_window=32)],
            dp_rank=2)
        s = ix.score_tokens(tokens, MODEL)
        if s == expected:  # retry until the EXACT state lands (idempotent)
            break
        time.sleep(0.1)
    assert s == expected

    # sliding pod evicts its out-of-window leading blocks: score survives
    # (window hints) even though the entries live in Redis
    pub_b.publish_events([block_removed_payload(list(range(11, 17)))],
                         dp_rank=2)
    deadline = time.time() + 10
    while time.time() < deadline:
        _, total, hits = ix.score_tokens_detailed(tokens, MODEL,
                                                  ["pod-b-dp2"])
        if hits == 2:
            break
        time.sleep(0.05)
    assert ix.score_tokens(tokens, MODEL, ["pod-b-dp2"]) == {"pod-b-dp2": 8.0}

    # a scheduler speculatively pins a fresh prefix to pod-a before any
    # engine event confirms it
    fresh = list(range(700, 764))
    keys = ix.compute_block_keys(fresh, MODEL)
    ix.index.add([], keys, [k.PodEntry("pod-a", "gpu", speculative=True)])
    assert ix.score_tokens(fresh, MODEL)["pod-a"] == 4.0

    # pod-a restarts: AllBlocksCleared wipes it from the shared index
    pub_a.publish_events([all_blocks_cleared_payload()])
    deadline = time.time() + 10
    while time.time() < deadline:
        s = ix.score_tokens(tokens, MODEL)
        if "pod-a" not in s:
            break
        time.sleep(0.05)
    assert s == {"pod-b-dp2": 8.0}
    assert ix.score_tokens(fresh, MODEL) == {}

    # a SECOND indexer process sharing the backend sees everything,
    # including window-correct scoring via its own learned catalog? The
    # catalog is local state — a fresh replica without the events falls
    # back to the strict walk, which scores the evicted-sliding pod 0.
    # That is the documented convergence behavior: replicas that consume
    # the stream agree; a cold replica under-scores (never over-scores).
    ix2 = KVCacheIndexer(IndexerConfig(
        redis_index=RedisIndexConfig(host="127.0.0.1", port=redis_srv.port)))
    assert ix2.score_tokens(tokens, MODEL, ["pod-b-dp2"]) == {}

    pub_a.close()
    pub_b.close()
    pool.shutdown()
