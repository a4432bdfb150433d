"""Adversarial wire input: the event parser and ZMTP reader sit on network
boundaries and must survive arbitrary bytes (count a failure, never crash
or hang)."""
import random
import socket
import time

import msgpack
import pytest

from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer
from llm_d_kv_cache_amd.events import EventPoolConfig, KVEventsPool
from llm_d_kv_cache_amd.events.publisher import (
    block_stored_payload,
    encode_batch,
)

k = ensure_native()


def test_random_bytes_never_crash_parser():
    rng = random.Random(1234)
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), ix)
    for i in range(3000):
        n = rng.randrange(0, 200)
        payload = bytes(rng.randrange(256) for _ in range(n))
        pool.process(f"kv@pod-{i % 7}@m", i, payload)
    s = pool.stats()
    assert s.processed + s.parse_failures == 3000


def test_truncated_valid_payloads():
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), ix)
    full = encode_batch([
        block_stored_payload([1, 2, 3], None, list(range(48)), 16,
                             lora_name="l", extra_keys=[["a"], None, ["b"]],
                             group_idx=1, spec_kind="full_attention"),
    ])
    for cut in range(len(full)):
        pool.process("kv@pod-t@m", cut, full[:cut])
    pool.process("kv@pod-t@m", 0, full)  # the intact one still lands
    assert ix.score_tokens(list(range(48)), "l",
                           extra_features=[["a"], None, ["b"]]) == {"pod-t": 3.0}


def test_type_confused_fields():
    """Structurally valid msgpack with wrong field types in every slot."""
    ix = KVCacheIndexer(IndexerConfig())
    pool = KVEventsPool(EventPoolConfig(), ix)
    weird = [
        ["BlockStored", "not-a-list", None, [], 16],
        ["BlockStored", [{}], None, [1] * 16, 16],
        ["BlockStored", [1], "parent?", [1] * 16, "sixteen"],
        ["BlockRemoved", {"a": 1}],
        ["BlockRemoved"],
        [12345, [1]],
        {},
        "just a string",
        [],
    ]
    for i, ev in enumerate(weird):
        payload = msgpack.packb([0.0, [ev]], use_bin_type=True)
        pool.process("kv@pod-w@m", i, payload)  # must not raise
    # pool still functional afterwards
    pool.process("kv@pod-w@m", 99,
                 encode_batch([block_stored_payload([5], None, list(range(16)), 16)]))
    assert ix.score_tokens(list(range(16)), "m") == {"pod-w": 1.0}


def test_zmtp_garbage_connections():
    """Random bytes thrown at a bound SUB port must not kill the
    subscriber; a real publisher still works afterwards."""
    got = []
    sub = k.Subscriber("tcp://127.0.0.1:0", "",
                       callback=lambda t, s, p: got.append(p), bind=True)
    try:
        rng = random.Random(7)
        for _ in range(10):
            s = socket.create_connection(("127.0.0.1", sub.port))
            n = rng.randrange(1, 300)
            try:
                s.sendall(bytes(rng.randrange(256) for _ in range(n)))
            finally:
                s.close()
        time.sleep(0.2)
        pub = k.Publisher(f"tcp://127.0.0.1:{sub.port}", bind=False)
        deadline = time.time() + 10
        while pub.peer_count < 1 and time.time() < deadline:
            time.sleep(0.01)
        time.sleep(0.1)
        pub.publish("t", 1, b"alive")
        deadline = time.time() + 5
        while not got and time.time() < deadline:
            time.sleep(0.01)
        assert got == [b"alive"]
        pub.close()
    finally:
        sub.close()
