"""In-memory index: dual-key model, LRU bounds, concurrency.

Mirrors the reference test strategy for pkg/kvcache/kvblock/index_test.go /
in_memory.go semantics on the sharded C++ implementation.
"""
import threading

import pytest

from llm_d_kv_cache_amd import ensure_native

k = ensure_native()


def entry(pod, tier="gpu", spec=False, group=None):
    return k.PodEntry(pod, tier, spec, group)


def test_add_lookup_roundtrip():
    idx = k.InMemoryIndex()
    idx.add([], [1, 2, 3], [entry("pod-a")])
    got = idx.lookup([1, 2, 3])
    assert set(got.keys()) == {1, 2, 3}
    assert got[1][0].pod == "pod-a"
    assert got[1][0].tier == "gpu"


def test_lookup_requires_keys():
    idx = k.InMemoryIndex()
    with pytest.raises(Exception):
        idx.lookup([])


def test_add_requires_keys_and_entries():
    idx = k.InMemoryIndex()
    with pytest.raises(Exception):
        idx.add([], [], [entry("a")])
    with pytest.raises(Exception):
        idx.add([], [1], [])


def test_lookup_filters_pods():
    idx = k.InMemoryIndex()
    idx.add([], [1], [entry("pod-a"), entry("pod-b")])
    got = idx.lookup([1], ["pod-b"])
    assert [e.pod for e in got[1]] == ["pod-b"]
    # Unknown pod filter matches nothing.
    assert idx.lookup([1], ["pod-zzz"]) == {}


def test_missing_key_skipped_not_fatal():
    idx = k.InMemoryIndex()
    idx.add([], [1, 3], [entry("a")])
    got = idx.lookup([1, 2, 3])
    assert set(got.keys()) == {1, 3}


def test_engine_key_mapping_1to1():
    idx = k.InMemoryIndex()
    idx.add([10, 11], [1, 2], [entry("a")])
    assert idx.get_request_key(10) == 1
    assert idx.get_request_key(11) == 2
    assert idx.get_request_key(99) is None


def test_engine_key_mapping_many_to_1():
    # engine block 16 tokens, canonical 64 -> 4 engine keys per request key
    idx = k.InMemoryIndex()
    idx.add([10, 11, 12, 13], [1], [entry("a")])
    for ek in (10, 11, 12, 13):
        assert idx.get_request_key(ek) == 1


def test_engine_key_mapping_1_to_many():
    # engine block 128 tokens, canonical 64 -> each engine key spans 2 rks;
    # get_request_key returns the LAST in the span (chain continuation).
    idx = k.InMemoryIndex()
    idx.add([10, 11], [1, 2, 3, 4], [entry("a")])
    assert idx.get_request_key(10) == 2
    assert idx.get_request_key(11) == 4


def test_evict_engine_key_removes_all_spanned():
    idx = k.InMemoryIndex()
    idx.add([10], [1, 2], [entry("a")])
    idx.evict(10, "engine", [entry("a")])
    assert idx.lookup([1, 2]) == {}
    # mapping dropped once all spanned keys are empty
    assert idx.get_request_key(10) is None


def test_evict_request_key():
    idx = k.InMemoryIndex()
    idx.add([], [1], [entry("a"), entry("b")])
    idx.evict(1, "request", [entry("a")])
    got = idx.lookup([1])
    assert [e.pod for e in got[1]] == ["b"]


def test_evict_only_named_entries():
    idx = k.InMemoryIndex()
    idx.add([10], [1], [entry("a", "gpu"), entry("a", "cpu")])
    idx.evict(10, "engine", [entry("a", "gpu")])
    got = idx.lookup([1])
    assert [(e.pod, e.tier) for e in got[1]] == [("a", "cpu")]
    # mapping survives: key not empty
    assert idx.get_request_key(10) == 1


def test_evict_unknown_engine_key_is_noop():
    idx = k.InMemoryIndex()
    idx.evict(123, "engine", [entry("a")])


def test_clear_pod():
    idx = k.InMemoryIndex()
    idx.add([], [1, 2], [entry("a", "gpu"), entry("b", "gpu")])
    idx.add([], [3], [entry("a", "cpu")])
    idx.clear("a")
    got = idx.lookup([1, 2, 3])
    assert set(got.keys()) == {1, 2}
    assert all(e.pod == "b" for key in got for e in got[key])


def test_pods_per_key_lru_bound():
    idx = k.InMemoryIndex(pods_per_key=3)
    for i in range(5):
        idx.add([], [1], [entry(f"pod-{i}")])
    got = idx.lookup([1])
    pods = [e.pod for e in got[1]]
    assert len(pods) == 3
    # most recent first, oldest dropped
    assert pods == ["pod-4", "pod-3", "pod-2"]


def test_re_add_moves_to_front():
    idx = k.InMemoryIndex(pods_per_key=3)
    for i in range(3):
        idx.add([], [1], [entry(f"pod-{i}")])
    idx.add([], [1], [entry("pod-0")])
    got = idx.lookup([1])
    assert [e.pod for e in got[1]] == ["pod-0", "pod-2", "pod-1"]


def test_key_capacity_eviction():
    idx = k.InMemoryIndex(size=64, shards=1)
    for i in range(200):
        idx.add([], [i], [entry("a")])
    found = idx.lookup(list(range(200)))
    assert len(found) <= 64
    # newest keys survive
    assert 199 in found


def test_group_entries_distinct():
    idx = k.InMemoryIndex()
    idx.add([], [1], [entry("a", group=0), entry("a", group=1)])
    got = idx.lookup([1])
    assert len(got[1]) == 2
    groups = {e.group for e in got[1]}
    assert groups == {0, 1}


def test_speculative_flag_roundtrip():
    idx = k.InMemoryIndex()
    idx.add([], [1], [entry("a", spec=True)])
    got = idx.lookup([1])
    assert got[1][0].speculative is True


def test_stats_counters():
    idx = k.InMemoryIndex()
    idx.add([], [1, 2], [entry("a")])
    idx.lookup([1, 2])
    s = idx.stats()
    assert s.admissions == 2
    assert s.lookups == 1
    assert s.hits == 1
    assert s.keys == 2


def test_concurrent_add_evict_lookup():
    idx = k.InMemoryIndex(shards=8)
    stop = threading.Event()
    errors = []

    def adder(pod):
        try:
            while not stop.is_set():
                idx.add([7], [1, 2], [entry(pod)])
        except Exception as e:  # pragma: no cover
            errors.append(e)

    def evicter(pod):
        try:
            while not stop.is_set():
                idx.evict(7, "engine", [entry(pod)])
        except Exception as e:  # pragma: no cover
            errors.append(e)

    def looker():
        try:
            while not stop.is_set():
                idx.lookup([1, 2])
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [
        threading.Thread(target=adder, args=("pod-a",)),
        threading.Thread(target=adder, args=("pod-b",)),
        threading.Thread(target=evicter, args=("pod-a",)),
        threading.Thread(target=looker),
        threading.Thread(target=looker),
    ]
    for t in threads:
        t.start()
    import time

    time.sleep(1.0)
    stop.set()
    for t in threads:
        t.join()
    assert not errors


def test_snapshot_save_load(tmp_path):
    """Warm-restart snapshot: a fresh index restored from the file serves
    the same lookups (entries, tiers, flags, groups, engine bridge)."""
    idx = k.InMemoryIndex(shards=4)
    idx.add([10, 11], [1, 2], [k.PodEntry("pod-a", "gpu"),
                               k.PodEntry("pod-b", "cpu", speculative=True)])
    idx.add([], [3], [k.PodEntry("pod-c", "gpu", group=2)])
    path = str(tmp_path / "idx.snap")
    idx.save(path)

    idx2 = k.InMemoryIndex(shards=8)  # different shard count is fine
    idx2.load(path)
    got = idx2.lookup([1, 2, 3])
    assert set(got.keys()) == {1, 2, 3}
    assert {(e.pod, e.tier) for e in got[1]} == {("pod-a", "gpu"),
                                                 ("pod-b", "cpu")}
    assert [e for e in got[1] if e.pod == "pod-b"][0].speculative
    e3 = got[3][0]
    assert e3.pod == "pod-c" and e3.group == 2
    assert idx2.get_request_key(10) == 1
    assert idx2.get_request_key(11) == 2
    assert idx2.stats().keys == 3


def test_snapshot_corrupt_file_raises(tmp_path):
    bad = tmp_path / "bad.snap"
    bad.write_bytes(b"KVIXSNP1" + b"\xff" * 40)
    idx = k.InMemoryIndex(shards=4)
    with pytest.raises(RuntimeError):
        idx.load(str(bad))
    bad.write_bytes(b"NOTASNAP")
    with pytest.raises(RuntimeError):
        idx.load(str(bad))
    # index stays usable after a failed load
    idx.add([], [5], [k.PodEntry("p", "gpu")])
    assert 5 in idx.lookup([5])
