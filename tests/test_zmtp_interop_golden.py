"""Golden-bytes ZMTP interop: a simulated libzmq 4.x peer at the raw-TCP
level.

The engines publish KVEvents with pyzmq/libzmq; this image ships no libzmq,
so interop is asserted against hard-coded byte sequences — the greeting,
NULL-mechanism READY, subscription and message framing libzmq 4.3 emits are
deterministic and specified byte-for-byte in RFC 23 (ZMTP 3.0) / RFC 37
(ZMTP 3.1) (reference counterpart: the battle-tested go-zeromq/zmq4 wire
behavior relied on by pkg/kvevents/zmq_subscriber.go:29-31).

Every test drives one side with a plain socket speaking literal libzmq
bytes and asserts our side both ACCEPTS them and EMITS frames a libzmq
parser would accept.
"""
import socket
import struct
import threading
import time

import pytest

from llm_d_kv_cache_amd import ensure_native

k = ensure_native()

# ---- libzmq 4.x golden byte sequences (RFC 23 / RFC 37) ---------------------

# 64-byte greeting: signature FF + 8 padding + 7F, version, mechanism
# "NULL" null-padded to 20, as-server 0, 31 filler bytes.
def greeting(minor: int) -> bytes:
    g = bytearray(64)
    g[0] = 0xFF
    g[9] = 0x7F
    g[10] = 3          # version-major
    g[11] = minor      # libzmq 4.3 sends 3.1; older peers 3.0
    g[12:16] = b"NULL"
    return bytes(g)


def command_frame(name: bytes, payload: bytes) -> bytes:
    body = bytes([len(name)]) + name + payload
    assert len(body) <= 255
    return bytes([0x04, len(body)]) + body


def ready_frame(socket_type: bytes) -> bytes:
    # metadata: 1-byte name length + "Socket-Type" + 4-byte BE value length
    meta = bytes([11]) + b"Socket-Type" + struct.pack(">I", len(socket_type)) \
        + socket_type
    return command_frame(b"READY", meta)


def message_frames(topic: bytes, seq: int, payload: bytes) -> bytes:
    # the vLLM KVEvents 3-frame message: [topic | 8-byte BE seq | payload]
    out = bytes([0x01, len(topic)]) + topic          # flags: MORE
    out += bytes([0x01, 8]) + struct.pack(">Q", seq)
    out += bytes([0x00, len(payload)]) + payload     # flags: last
    return out


def recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        data = sock.recv(n - len(buf))
        if not data:
            raise ConnectionError(f"peer closed after {len(buf)}/{n} bytes")
        buf += data
    return buf


def recv_frame(sock: socket.socket):
    flags = recv_exact(sock, 1)[0]
    if flags & 0x02:
        n = struct.unpack(">Q", recv_exact(sock, 8))[0]
    else:
        n = recv_exact(sock, 1)[0]
    return flags, recv_exact(sock, n)


def assert_valid_greeting(g: bytes):
    """Verify OUR greeting byte-for-byte as libzmq's parser would."""
    assert len(g) == 64
    assert g[0] == 0xFF and g[9] == 0x7F, "bad signature"
    assert g[10] == 3, "version-major must be 3"
    mech = g[12:32].rstrip(b"\x00")
    assert mech == b"NULL"
    assert g[32] in (0, 1)  # as-server


def wait_for(cond, timeout=5.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if cond():
            return True
        time.sleep(0.01)
    return False


# ---- libzmq SUB peer -> our Publisher (bind) --------------------------------

@pytest.mark.parametrize("minor,sub_style", [(0, "message"), (1, "command")])
def test_libzmq_sub_peer_receives_published_batch(minor, sub_style):
    pub = k.Publisher("tcp://127.0.0.1:0", bind=True)
    s = socket.create_connection(("127.0.0.1", pub.port), timeout=5)
    s.settimeout(5)
    try:
        s.sendall(greeting(minor))
        assert_valid_greeting(recv_exact(s, 64))
        s.sendall(ready_frame(b"SUB"))
        flags, body = recv_frame(s)
        assert flags & 0x04 and body[1:1 + body[0]] == b"READY"
        if sub_style == "message":
            # ZMTP 3.0: subscription as a 0x01-prefixed message
            s.sendall(bytes([0x00, 1 + 3, 0x01]) + b"kv@")
        else:
            # ZMTP 3.1: SUBSCRIBE command
            s.sendall(command_frame(b"SUBSCRIBE", b"kv@"))
        assert wait_for(lambda: pub.peer_count >= 1)
        time.sleep(0.1)  # let the subscription land
        pub.publish("kv@pod-1@m", 42, b"\x92\x90\xa3abc")
        pub.publish("skip@x", 1, b"nope")  # filtered by prefix
        pub.publish("kv@pod-1@m", 43, b"tail")
        # exact 3-frame wire a libzmq SUB would deliver to the app
        f1, topic = recv_frame(s)
        assert f1 == 0x01 and topic == b"kv@pod-1@m"
        f2, seq = recv_frame(s)
        assert f2 == 0x01 and seq == struct.pack(">Q", 42)
        f3, payload = recv_frame(s)
        assert f3 == 0x00 and payload == b"\x92\x90\xa3abc"
        _, topic2 = recv_frame(s)
        assert topic2 == b"kv@pod-1@m"
        _, seq2 = recv_frame(s)
        assert seq2 == struct.pack(">Q", 43)
        recv_frame(s)
    finally:
        s.close()
        pub.close()


# ---- libzmq PUB peer -> our Subscriber (dial and bind) ----------------------

@pytest.mark.parametrize("topology", ["sub_dials", "sub_binds"])
def test_libzmq_pub_peer_feeds_subscriber(topology):
    got = []
    lock = threading.Lock()

    def on_msg(topic, seq, payload):
        with lock:
            got.append((topic, seq, payload))

    if topology == "sub_dials":
        # pod-discovery: libzmq PUB binds, our SUB dials it
        srv = socket.socket()
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind(("127.0.0.1", 0))
        srv.listen(1)
        sub = k.Subscriber(f"tcp://127.0.0.1:{srv.getsockname()[1]}", "kv@",
                           callback=on_msg)
        s, _ = srv.accept()
        srv.close()
    else:
        # centralized: our SUB binds, libzmq PUB dials in
        sub = k.Subscriber("tcp://127.0.0.1:0", "kv@", callback=on_msg,
                           bind=True)
        s = socket.create_connection(("127.0.0.1", sub.port), timeout=5)
    s.settimeout(5)
    try:
        s.sendall(greeting(1))
        assert_valid_greeting(recv_exact(s, 64))
        s.sendall(ready_frame(b"PUB"))
        flags, body = recv_frame(s)
        assert flags & 0x04 and body[1:1 + body[0]] == b"READY"
        # our SUB must announce its subscription in a form libzmq accepts:
        # a 0x01-prefixed message (3.0 style, valid for every peer version)
        sflags, sbody = recv_frame(s)
        assert (sflags & 0x04) == 0 and sbody[0] == 0x01
        assert sbody[1:] == b"kv@"
        # a libzmq heartbeat PING must come back as PONG with the context
        # echoed, or libzmq closes the connection after the TTL
        s.sendall(command_frame(b"PING", struct.pack(">H", 100) + b"ctx1"))
        pflags, pbody = recv_frame(s)
        assert pflags & 0x04
        assert pbody[1:1 + pbody[0]] == b"PONG" and pbody[1 + pbody[0]:] == b"ctx1"
        # exact libzmq 3-frame publish
        s.sendall(message_frames(b"kv@pod-9@m", 7, b"payload-bytes"))
        assert wait_for(lambda: len(got) == 1)
        with lock:
            assert got[0] == ("kv@pod-9@m", 7, b"payload-bytes")
    finally:
        s.close()
        sub.close()


def test_greeting_rejects_zmtp2_peer():
    """A ZMTP 1/2 peer (version-major < 3) must be refused, not garbled."""
    pub = k.Publisher("tcp://127.0.0.1:0", bind=True)
    s = socket.create_connection(("127.0.0.1", pub.port), timeout=5)
    s.settimeout(5)
    try:
        g = bytearray(greeting(0))
        g[10] = 2  # ZMTP 2
        s.sendall(bytes(g))
        recv_exact(s, 64)  # our greeting goes out first
        # our side must close rather than proceed
        s.settimeout(2)
        try:
            data = s.recv(64)
        except (socket.timeout, ConnectionError):
            data = b""
        assert data == b"", "ZMTP<3 peer was not refused"
        assert pub.peer_count == 0
    finally:
        s.close()
        pub.close()
