"""Property-based tests (hypothesis): hash-chain contract vs the pure
oracle, wire decode against arbitrary msgpack-built events, and index
invariants under arbitrary op sequences."""
import msgpack
from hypothesis import given, settings, strategies as st

import reference_impl as ref
from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool

k = ensure_native()

tokens_st = st.lists(st.integers(0, 2**32 - 1), min_size=0, max_size=130)
model_st = st.text(min_size=0, max_size=24)


@settings(max_examples=200, deadline=None)
@given(tokens=tokens_st, model=model_st,
       block_size=st.sampled_from([1, 4, 16, 64]),
       seed=st.text(max_size=8))
def test_hash_chain_matches_oracle(tokens, model, block_size, seed):
    tp = k.TokenProcessor(block_size, seed)
    got = tp.tokens_to_block_keys(tokens, model)
    want = ref.block_keys(tokens, model, block_size=block_size, hash_seed=seed)
    assert got == want


@settings(max_examples=200, deadline=None)
@given(
    hashes=st.lists(st.one_of(st.integers(0, 2**64 - 1),
                              st.binary(min_size=1, max_size=16)),
                    min_size=0, max_size=8),
    tokens=st.lists(st.integers(0, 2**31 - 1), max_size=64),
    block_size=st.integers(1, 64),
    medium=st.one_of(st.none(), st.text(max_size=8)),
    lora=st.one_of(st.none(), st.text(min_size=1, max_size=8)),
    extra=st.one_of(
        st.none(),
        st.lists(st.one_of(st.none(), st.lists(st.text(max_size=6), max_size=3)),
                 max_size=6),
    ),
)
def test_arbitrary_block_stored_never_crashes(hashes, tokens, block_size,
                                              medium, lora, extra):
    """Any structurally-valid BlockStored the msgpack library can encode
    either applies or is skipped — never crashes, and the pool stays
    functional."""
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), ix)
    ev = ["BlockStored", hashes, None, tokens, block_size, None, medium,
          lora, extra]
    payload = msgpack.packb([0.0, [ev]], use_bin_type=True)
    pool.process("kv@pod@m", 0, payload)
    s = pool.stats()
    assert s.processed + s.parse_failures == 1


@settings(max_examples=100, deadline=None)
@given(ops=st.lists(
    st.tuples(st.sampled_from(["add", "evict", "lookup", "clear"]),
              st.integers(0, 15),  # key space
              st.integers(0, 3)),  # pod space
    min_size=1, max_size=60))
def test_index_invariants_under_arbitrary_ops(ops):
    """After any op sequence: every looked-up entry names a pod that was
    added and not since cleared/evicted past it; lookups never return
    empty lists for present keys."""
    idx = k.InMemoryIndex(shards=4, pods_per_key=4)
    for op, key, pod in ops:
        e = [k.PodEntry(f"p{pod}", "gpu")]
        if op == "add":
            idx.add([key + 100], [key], e)
        elif op == "evict":
            idx.evict(key + 100, "engine", e)
        elif op == "lookup":
            got = idx.lookup([key] if key else [1])
            for kk, entries in got.items():
                assert entries, "present key must never have an empty list"
        elif op == "clear":
            idx.clear(f"p{pod}")
    got = idx.lookup(list(range(16)))
    for kk, entries in got.items():
        assert entries
        for ent in entries:
            assert ent.pod.startswith("p")


@settings(max_examples=80, deadline=None)
@given(blob=st.binary(max_size=300), flip=st.integers(0, 10**6))
def test_snapshot_loader_rejects_garbage(tmp_path_factory, blob, flip):
    """Arbitrary bytes and bit-flipped valid snapshots must never crash the
    loader — they either raise or merge a harmless subset; the index stays
    usable either way."""
    import os
    import tempfile

    d = tempfile.mkdtemp(prefix="snapfuzz")
    garbage = os.path.join(d, "g.snap")
    with open(garbage, "wb") as f:
        f.write(blob)
    idx = k.InMemoryIndex(shards=4)
    try:
        idx.load(garbage)
    except Exception:
        pass
    # valid snapshot with one flipped byte
    src = k.InMemoryIndex(shards=4)
    src.add([7], [1, 2], [k.PodEntry("pod-a", "gpu"),
                          k.PodEntry("pod-b", "cpu")])
    valid = os.path.join(d, "v.snap")
    src.save(valid)
    data = bytearray(open(valid, "rb").read())
    pos = flip % len(data)
    data[pos] ^= 0xFF
    with open(valid, "wb") as f:
        f.write(bytes(data))
    try:
        idx.load(valid)
    except Exception:
        pass
    idx.add([], [9], [k.PodEntry("p", "gpu")])
    assert 9 in idx.lookup([9])


# ---- handler file-splitting math (property) --------------------------------

class _RecordingEngine:
    """Stub engine capturing the exact FileTransfer lists handlers build."""

    def __init__(self, num_layers=2, block_bytes=1024):
        self.group_geometry = [{"num_layers": num_layers,
                                "block_bytes": block_bytes}]
        self.calls = []
        self._next = 1

    def async_store(self, files, stream=None):
        self.calls.append(("store", files))
        self._next += 1
        return self._next - 1

    def async_load(self, files):
        self.calls.append(("load", files))
        self._next += 1
        return self._next - 1

    def poll_finished(self, ids):
        return []


@settings(max_examples=200, deadline=None)
@given(
    bpf=st.integers(min_value=1, max_value=16),
    n_chunks=st.integers(min_value=1, max_value=8),
    skip=st.integers(min_value=0, max_value=127),
    tail_drop=st.integers(min_value=0, max_value=15),
)
def test_load_split_covers_exactly_the_requested_blocks(
        bpf, n_chunks, skip, tail_drop):
    """For any (blocks/file, chunk count, leading skip, short tail): the
    generated per-file transfers cover exactly the requested block ids in
    order, the first file is tail-seeked by skip%bpf, later files start at
    slot 0, and no file exceeds its remaining capacity (reference
    worker.py head/tail-partial math)."""
    from llm_d_kv_cache_amd.offload.file_mapper import (
        FileMapper,
        KVCacheLayoutConfig,
    )
    from llm_d_kv_cache_amd.offload.handlers import StorageToGPUHandler

    total = n_chunks * bpf
    skip = min(skip, total)
    n_want = max(0, total - skip - min(tail_drop, total - skip))
    if n_want == 0:
        return
    ids = list(range(1000, 1000 + n_want))
    eng = _RecordingEngine()
    import tempfile

    mapper = FileMapper(tempfile.mkdtemp(), KVCacheLayoutConfig(model="prop"))
    h = StorageToGPUHandler(eng, mapper, [bpf])
    hashes = list(range(1, n_chunks + 1))
    h.transfer_async(hashes, {0: ids}, skip_leading_blocks=skip)
    (kind, files), = eng.calls
    assert kind == "load"
    got = []
    for fi, (group, path, f_ids, slot) in enumerate(files):
        assert group == 0
        ci = skip // bpf + fi
        assert mapper.file_name(hashes[ci], 0) == path
        expect_slot = skip % bpf if fi == 0 else 0
        assert slot == expect_slot
        assert 1 <= len(f_ids) <= bpf - expect_slot
        if fi not in (0, len(files) - 1):
            assert len(f_ids) == bpf  # middle files are always full spans
        got.extend(f_ids)
    assert got == ids
