"""Embedded fake Redis/Valkey: a minimal RESP2 server for backend tests
(miniredis role, matching the reference test strategy of running index
backends against an in-process fake — SURVEY.md §4)."""
from __future__ import annotations

import socket
import threading
from typing import Dict


class FakeRedis:
    def __init__(self):
        self.hashes: Dict[bytes, Dict[bytes, bytes]] = {}
        self.strings: Dict[bytes, bytes] = {}
        # test hook: called (under the server lock) just before an EVAL
        # executes — lets tests inject a "concurrent" mutation at the exact
        # point the prune script's atomicity matters.
        self.on_eval = None
        self.lock = threading.Lock()
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind(("127.0.0.1", 0))
        self._srv.listen(16)
        self.port = self._srv.getsockname()[1]
        self._stop = False
        self._threads = []
        self._conns = []
        t = threading.Thread(target=self._accept_loop, daemon=True)
        t.start()
        self._threads.append(t)

    def close(self):
        self._stop = True
        try:
            # shutdown() wakes the blocked accept(); close() alone leaves the
            # listener alive inside the in-flight accept syscall
            self._srv.shutdown(socket.SHUT_RDWR)
        except OSError:
            pass
        try:
            self._srv.close()
        except OSError:
            pass
        # drop accepted connections too: a real outage severs live clients,
        # not just the listener
        for c in list(self._conns):
            try:
                c.shutdown(socket.SHUT_RDWR)
                c.close()
            except OSError:
                pass

    # ---- RESP plumbing ------------------------------------------------------

    def _accept_loop(self):
        while not self._stop:
            try:
                conn, _ = self._srv.accept()
            except OSError:
                return
            self._conns.append(conn)
            t = threading.Thread(target=self._serve, args=(conn,), daemon=True)
            t.start()
            self._threads.append(t)

    def _serve(self, conn):
        conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        buf = b""

        def read_line():
            nonlocal buf
            while b"\r\n" not in buf:
                data = conn.recv(4096)
                if not data:
                    raise ConnectionError
                buf += data
            line, buf = buf.split(b"\r\n", 1)
            return line

        def read_exact(n):
            nonlocal buf
            while len(buf) < n + 2:
                data = conn.recv(4096)
                if not data:
                    raise ConnectionError
                buf += data
            out, buf = buf[:n], buf[n + 2:]
            return out

        try:
            while True:
                line = read_line()
                if not line.startswith(b"*"):
                    conn.sendall(b"-ERR protocol\r\n")
                    continue
                nargs = int(line[1:])
                args = []
                for _ in range(nargs):
                    hdr = read_line()
                    assert hdr.startswith(b"$")
                    args.append(read_exact(int(hdr[1:])))
                conn.sendall(self._dispatch(args))
        except (ConnectionError, OSError):
            pass
        finally:
            conn.close()

    # ---- command handlers ---------------------------------------------------

    @staticmethod
    def _int(v):
        return b":" + str(v).encode() + b"\r\n"

    @staticmethod
    def _bulk(v):
        if v is None:
            return b"$-1\r\n"
        return b"$" + str(len(v)).encode() + b"\r\n" + v + b"\r\n"

    @staticmethod
    def _arr(items):
        return b"*" + str(len(items)).encode() + b"\r\n" + b"".join(items)

    def _dispatch(self, args):
        cmd = args[0].upper()
        with self.lock:
            if cmd == b"PING":
                return b"+PONG\r\n"
            if cmd == b"FLUSHALL":
                self.hashes.clear()
                self.strings.clear()
                return b"+OK\r\n"
            if cmd == b"HSET":
                h = self.hashes.setdefault(args[1], {})
                added = 0
                for i in range(2, len(args), 2):
                    if args[i] not in h:
                        added += 1
                    h[args[i]] = args[i + 1]
                return self._int(added)
            if cmd == b"HGETALL":
                h = self.hashes.get(args[1], {})
                items = []
                for f, v in h.items():
                    items.append(self._bulk(f))
                    items.append(self._bulk(v))
                return self._arr(items)
            if cmd == b"HKEYS":
                h = self.hashes.get(args[1], {})
                return self._arr([self._bulk(f) for f in h])
            if cmd == b"HDEL":
                h = self.hashes.get(args[1], {})
                n = 0
                for f in args[2:]:
                    if f in h:
                        del h[f]
                        n += 1
                if not h:
                    self.hashes.pop(args[1], None)
                return self._int(n)
            if cmd == b"HLEN":
                return self._int(len(self.hashes.get(args[1], {})))
            if cmd == b"DEL":
                n = 0
                for k in args[1:]:
                    n += int(self.hashes.pop(k, None) is not None)
                    n += int(self.strings.pop(k, None) is not None)
                return self._int(n)
            if cmd == b"SET":
                self.strings[args[1]] = args[2]
                return b"+OK\r\n"
            if cmd == b"GET":
                return self._bulk(self.strings.get(args[1]))
            if cmd == b"EVAL" and self.on_eval is not None:
                hook, self.on_eval = self.on_eval, None
                hook(self)
            if cmd == b"EVAL":
                # Atomic script execution (miniredis role): the only script
                # the client sends is the prune-if-empty one; interpret its
                # semantics under the single server lock so it is atomic
                # w.r.t. every other command, exactly like real Redis Lua.
                script = args[1]
                if b"HLEN" not in script or b"all_empty" not in script:
                    return b"-ERR unsupported script\r\n"
                numkeys = int(args[2])
                keys = args[3:3 + numkeys]
                all_empty = 1
                for k in keys[:-1]:
                    if len(self.hashes.get(k, {})) == 0:
                        self.hashes.pop(k, None)
                    else:
                        all_empty = 0
                if all_empty and keys:
                    ek = keys[-1]
                    self.hashes.pop(ek, None)
                    self.strings.pop(ek, None)
                return self._int(all_empty)
            if cmd == b"SCAN":
                # single-pass cursor: return everything matching on cursor 0
                pattern = b"*"
                for i, a in enumerate(args):
                    if a.upper() == b"MATCH":
                        pattern = args[i + 1]
                prefix = pattern.rstrip(b"*")
                keys = [k for k in list(self.hashes) + list(self.strings)
                        if k.startswith(prefix)]
                return self._arr([self._bulk(b"0"),
                                  self._arr([self._bulk(k) for k in keys])])
            return b"-ERR unknown command " + cmd + b"\r\n"
