"""Alternative index backends: byte-budget (cost-aware) mode and the
Redis/Valkey network backend against an embedded fake server."""
import pytest

from fake_redis import FakeRedis
from llm_d_kv_cache_amd import ensure_native

k = ensure_native()


def entry(pod, tier="gpu", group=None):
    return k.PodEntry(pod, tier, False, group)


# ---- cost-aware (byte budget) ----------------------------------------------

def test_cost_aware_evicts_to_budget():
    # budget for roughly a handful of keys per shard
    idx = k.InMemoryIndex(shards=1, max_bytes=10 * (96 + 16))
    for i in range(100):
        idx.add([], [i], [entry("a")])
    found = idx.lookup(list(range(100)))
    assert 0 < len(found) <= 10
    assert 99 in found  # newest survives
    assert idx.stats().evictions > 0


def test_cost_aware_unbounded_by_default():
    idx = k.InMemoryIndex(shards=1)
    for i in range(100):
        idx.add([], [i], [entry("a")])
    assert len(idx.lookup(list(range(100)))) == 100


# ---- redis / valkey ---------------------------------------------------------

@pytest.fixture
def redis_pair():
    srv = FakeRedis()
    idx = k.RedisIndex(host="127.0.0.1", port=srv.port)
    yield srv, idx
    srv.close()


def test_redis_add_lookup(redis_pair):
    srv, idx = redis_pair
    idx.add([], [1, 2], [entry("pod-a"), entry("pod-b", "cpu")])
    got = idx.lookup([1, 2, 3])
    assert set(got.keys()) == {1, 2}
    pods = {(e.pod, e.tier) for e in got[1]}
    assert pods == {("pod-a", "gpu"), ("pod-b", "cpu")}


def test_redis_pod_filter(redis_pair):
    srv, idx = redis_pair
    idx.add([], [5], [entry("pod-a"), entry("pod-b")])
    got = idx.lookup([5], ["pod-b"])
    assert [e.pod for e in got[5]] == ["pod-b"]


def test_redis_engine_bridge(redis_pair):
    srv, idx = redis_pair
    idx.add([10, 11], [1, 2, 3, 4], [entry("a")])  # 1:many
    assert idx.get_request_key(10) == 2
    assert idx.get_request_key(11) == 4
    assert idx.get_request_key(99) is None


def test_redis_evict_engine_key(redis_pair):
    srv, idx = redis_pair
    idx.add([10], [1, 2], [entry("a")])
    idx.evict(10, "engine", [entry("a")])
    assert idx.lookup([1, 2]) == {}
    assert idx.get_request_key(10) is None


def test_redis_evict_partial(redis_pair):
    srv, idx = redis_pair
    idx.add([10], [1], [entry("a"), entry("b")])
    idx.evict(10, "engine", [entry("a")])
    got = idx.lookup([1])
    assert [e.pod for e in got[1]] == ["b"]
    assert idx.get_request_key(10) == 1  # mapping survives: key non-empty


def test_redis_clear_pod(redis_pair):
    srv, idx = redis_pair
    idx.add([], [1, 2], [entry("a"), entry("b")])
    idx.clear("a")
    got = idx.lookup([1, 2])
    assert all(e.pod == "b" for kk in got for e in got[kk])


def test_redis_group_flags_roundtrip(redis_pair):
    srv, idx = redis_pair
    idx.add([], [7], [entry("a", group=3)])
    got = idx.lookup([7])
    assert got[7][0].group == 3


def test_redis_shared_state_between_instances(redis_pair):
    """Two index instances (two replicas) see each other's writes."""
    srv, idx = redis_pair
    idx2 = k.RedisIndex(host="127.0.0.1", port=srv.port)
    idx.add([], [42], [entry("pod-x")])
    got = idx2.lookup([42])
    assert [e.pod for e in got[42]] == ["pod-x"]


def test_redis_connection_refused():
    with pytest.raises(Exception):
        k.RedisIndex(host="127.0.0.1", port=1)  # nothing listens there


def test_redis_backend_drives_indexer_and_pool(redis_pair):
    """Full wiring: Indexer + EventPool on the Redis backend."""
    from llm_d_kv_cache_amd.events.publisher import (
        block_stored_payload,
        encode_batch,
    )

    srv, idx = redis_pair
    tp = k.TokenProcessor(16, "")
    ix = k.Indexer(tp, idx, {"gpu": 1.0, "cpu": 0.8})
    pool = k.EventPool(tp, idx, 2)
    tokens = list(range(32))
    pool.process("kv@pod-r@m", 0,
                 encode_batch([block_stored_payload([1, 2], None, tokens, 16)]))
    res = ix.score_tokens(tokens, "m", [])
    assert res.scores == {"pod-r": 2.0}


def test_redis_cross_process_pod_filter(redis_pair):
    """Multi-replica shape: one indexer process ingests into the shared
    backend, a DIFFERENT process (fresh string table) scores with an
    explicit candidate list — the filter must match entries it has never
    interned locally."""
    from llm_d_kv_cache_amd.core import (
        IndexerConfig,
        KVCacheIndexer,
        RedisIndexConfig,
    )

    srv, _ = redis_pair
    cfg = lambda: IndexerConfig(  # noqa: E731
        redis_index=RedisIndexConfig(host="127.0.0.1", port=srv.port))
    writer = KVCacheIndexer(cfg())
    keys = writer.compute_block_keys(list(range(64)), "m")
    writer.index.add([], keys, [entry("pod-x")])
    reader = KVCacheIndexer(cfg())
    assert reader.score_tokens(list(range(64)), "m",
                               ["pod-x", "pod-absent"]) == {"pod-x": 4.0}


def test_redis_atomic_prune_no_lost_entries(redis_pair):
    """A concurrent add landing at the prune's empty-check must not be
    lost: the prune-if-empty runs as ONE atomic server-side script
    (reference redis.go:160-169), so an entry written just before it
    executes survives, and the engine bridge stays intact."""
    srv, idx = redis_pair
    idx.add([10], [1], [entry("pod-a")])

    def concurrent_add(s):
        # simulates another replica's add() racing the evict: the request
        # hash regains a field right before the prune script runs
        rkey = b"kv:r:" + b"%016x" % 1
        s.hashes.setdefault(rkey, {})[b"pod-b\x1fgpu\x1f0\x1f0"] = b"1"

    srv.on_eval = concurrent_add
    idx.evict(10, "engine", [entry("pod-a")])
    got = idx.lookup([1])
    assert [e.pod for e in got[1]] == ["pod-b"], "concurrent add was lost"
    assert idx.get_request_key(10) == 1, "engine bridge pruned under live key"


def test_redis_malformed_fields_skipped(redis_pair):
    """Corrupt shared-state fields (other versions, operator edits) are
    skipped, never thrown out of lookup()/evict()."""
    srv, idx = redis_pair
    idx.add([10], [1], [entry("pod-a")])
    rkey = b"kv:r:" + b"%016x" % 1
    with srv.lock:
        h = srv.hashes[rkey]
        h[b"pod-x\x1fgpu\x1fnot-a-number\x1f0"] = b"1"   # bad flags
        h[b"pod-y\x1fgpu\x1f0\x1f99999999999999999999"] = b"1"  # group overflow
        h[b"no-separators-at-all"] = b"1"
        srv.strings[b"kv:e:" + b"%016x" % 11] = b"zzzz,0001"  # bad rk list
    got = idx.lookup([1])
    assert [e.pod for e in got[1]] == ["pod-a"]
    assert idx.get_request_key(11) == 1  # undecodable item skipped, good one used
    idx.evict(10, "engine", [entry("pod-a")])  # must not raise


def test_pool_survives_backend_outage(redis_pair):
    """A transient Redis outage mid-stream must not kill the worker thread
    (std::terminate would take down the whole indexer process): events
    during the outage are counted as handler_failures and dropped."""
    from llm_d_kv_cache_amd.events.publisher import (
        block_stored_payload,
        encode_batch,
    )

    srv, idx = redis_pair
    tp = k.TokenProcessor(16, "")
    pool = k.EventPool(tp, idx, 1)
    pool.start()
    payload = encode_batch([block_stored_payload([1], None, list(range(16)), 16)])
    pool.add_task("kv@pod-a@m", 0, payload)
    pool.drain()
    assert pool.stats().handler_failures == 0
    srv.close()  # outage
    for i in range(3):
        pool.add_task("kv@pod-a@m", i + 1, payload)
    pool.drain()  # workers must still be alive to drain
    st = pool.stats()
    assert st.handler_failures == 3
    assert st.processed == 4
    pool.shutdown()


def test_cost_aware_admission_protects_hot_keys():
    """TinyLFU-style admission (ristretto's role, reference
    cost_aware_memory.go:41-52): a one-shot scan flood cannot evict the
    keys lookups keep hot — cold newcomers are rejected, not admitted at
    the hot set's expense."""
    idx = k.InMemoryIndex(shards=1, max_bytes=50 * (96 + 16))
    hot = list(range(20))
    for i in hot:
        idx.add([], [i], [entry("a")])
    for _ in range(30):  # repeated lookups warm the frequency sketch
        idx.lookup(hot)
    for i in range(10_000, 12_000):  # scan flood of one-shot keys
        idx.add([], [i], [entry("b")])
    found = idx.lookup(hot)
    assert len(found) >= 15, f"hot set washed out: {len(found)}/20 survive"
    st = idx.stats()
    assert st.rejections > 0, "no admission rejects during the flood"
