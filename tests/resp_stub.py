"""Minimal RESP2 (Redis-protocol) server for integration tests.

Implements exactly the command surface the RedisCoordinatorStorage backend
uses — strings, hashes, sets, sorted sets, and WATCH/MULTI/EXEC with real
optimistic-locking semantics (per-key version counters; EXEC returns nil if
a watched key changed since WATCH). Single-threaded-per-connection, shared
store under one lock, binary-safe keys/values.

Stands in for a real redis-server (not shipped in this image); the C++
client speaks unmodified RESP2, so pointing it at a real Redis works the
same way (reference integration tests run against docker redis similarly,
rust/xaynet-server/src/storage/coordinator_storage/redis/mod.rs:558+).
"""
from __future__ import annotations

import socket
import threading


class RespStubServer(threading.Thread):
    def __init__(self, host="127.0.0.1", port=0):
        super().__init__(daemon=True)
        self.sock = socket.socket()
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.sock.bind((host, port))
        self.sock.listen(16)
        self.port = self.sock.getsockname()[1]
        self.lock = threading.Lock()
        self.strings: dict[bytes, bytes] = {}
        self.hashes: dict[bytes, dict[bytes, bytes]] = {}
        self.sets: dict[bytes, set[bytes]] = {}
        self.zsets: dict[bytes, dict[bytes, float]] = {}
        self.versions: dict[bytes, int] = {}
        self._stop = False
        # fault injection for reconnect tests: drop the next N connections
        self.drop_next = 0
        self.conns: list = []

    # ------------------------------------------------------------ protocol

    def run(self):
        while not self._stop:
            try:
                conn, _ = self.sock.accept()
            except OSError:
                break
            with self.lock:
                if self.drop_next > 0:
                    self.drop_next -= 1
                    conn.close()
                    continue
            self.conns.append(conn)
            threading.Thread(target=self._serve, args=(conn,), daemon=True).start()

    def stop(self):
        self._stop = True
        try:
            self.sock.close()
        except OSError:
            pass
        self.kill_connections()

    def kill_connections(self):
        """Sever every live client connection (reconnect fault injection)."""
        for c in self.conns:
            try:
                c.shutdown(socket.SHUT_RDWR)
                c.close()
            except OSError:
                pass
        self.conns = []

    def _serve(self, conn):
        buf = b""
        watched: dict[bytes, int] = {}
        multi: list[list[bytes]] | None = None
        try:
            while True:
                cmd, buf = self._read_command(conn, buf)
                if cmd is None:
                    return
                name = cmd[0].upper()
                if name == b"MULTI":
                    multi = []
                    conn.sendall(b"+OK\r\n")
                elif name == b"EXEC":
                    with self.lock:
                        dirty = any(self.versions.get(k, 0) != v for k, v in watched.items())
                        if dirty or multi is None:
                            conn.sendall(b"*-1\r\n")
                        else:
                            out = b"*%d\r\n" % len(multi)
                            for qc in multi:
                                out += self._execute(qc)
                            conn.sendall(out)
                    watched = {}
                    multi = None
                elif name == b"DISCARD":
                    multi = None
                    watched = {}
                    conn.sendall(b"+OK\r\n")
                elif name == b"WATCH":
                    with self.lock:
                        for k in cmd[1:]:
                            watched[k] = self.versions.get(k, 0)
                    conn.sendall(b"+OK\r\n")
                elif name == b"UNWATCH":
                    watched = {}
                    conn.sendall(b"+OK\r\n")
                elif multi is not None:
                    multi.append(cmd)
                    conn.sendall(b"+QUEUED\r\n")
                else:
                    with self.lock:
                        conn.sendall(self._execute(cmd))
        except (ConnectionError, OSError):
            pass
        finally:
            conn.close()

    def _read_command(self, conn, buf):
        def need(n):
            nonlocal buf
            while len(buf) < n:
                d = conn.recv(65536)
                if not d:
                    raise ConnectionError
                buf += d

        def line():
            nonlocal buf
            while b"\r\n" not in buf:
                d = conn.recv(65536)
                if not d:
                    raise ConnectionError
                buf += d
            ln, buf = buf.split(b"\r\n", 1)
            return ln

        try:
            first = line()
        except ConnectionError:
            return None, buf
        if not first.startswith(b"*"):
            return None, buf
        n = int(first[1:])
        parts = []
        for _ in range(n):
            hdr = line()
            assert hdr.startswith(b"$")
            ln = int(hdr[1:])
            need(ln + 2)
            parts.append(buf[:ln])
            buf = buf[ln + 2:]
        return parts, buf

    # ------------------------------------------------------------ commands

    def _touch(self, key):
        self.versions[key] = self.versions.get(key, 0) + 1

    def _execute(self, cmd) -> bytes:
        name = cmd[0].upper()
        try:
            fn = getattr(self, "_cmd_" + name.decode().lower())
        except (AttributeError, UnicodeDecodeError):
            return b"-ERR unknown command\r\n"
        return fn(cmd[1:])

    @staticmethod
    def _bulk(v: bytes | None) -> bytes:
        if v is None:
            return b"$-1\r\n"
        return b"$%d\r\n%s\r\n" % (len(v), v)

    def _cmd_ping(self, a):
        return b"+PONG\r\n"

    def _cmd_set(self, a):
        self.strings[a[0]] = a[1]
        self._touch(a[0])
        return b"+OK\r\n"

    def _cmd_setnx(self, a):
        if a[0] in self.strings:
            return b":0\r\n"
        self.strings[a[0]] = a[1]
        self._touch(a[0])
        return b":1\r\n"

    def _cmd_get(self, a):
        return self._bulk(self.strings.get(a[0]))

    def _cmd_del(self, a):
        n = 0
        for k in a:
            for store in (self.strings, self.hashes, self.sets, self.zsets):
                if k in store:
                    del store[k]
                    n += 1
                    self._touch(k)
        return b":%d\r\n" % n

    def _cmd_flushdb(self, a):
        for store in (self.strings, self.hashes, self.sets, self.zsets):
            for k in list(store):
                self._touch(k)
            store.clear()
        return b"+OK\r\n"

    def _cmd_hsetnx(self, a):
        h = self.hashes.setdefault(a[0], {})
        if a[1] in h:
            return b":0\r\n"
        h[a[1]] = a[2]
        self._touch(a[0])
        return b":1\r\n"

    def _cmd_hgetall(self, a):
        h = self.hashes.get(a[0], {})
        out = b"*%d\r\n" % (2 * len(h))
        for k, v in h.items():
            out += self._bulk(k) + self._bulk(v)
        return out

    def _cmd_hlen(self, a):
        return b":%d\r\n" % len(self.hashes.get(a[0], {}))

    def _cmd_hexists(self, a):
        return b":%d\r\n" % (1 if a[1] in self.hashes.get(a[0], {}) else 0)

    def _cmd_sadd(self, a):
        s = self.sets.setdefault(a[0], set())
        n = 0
        for m in a[1:]:
            if m not in s:
                s.add(m)
                n += 1
        if n:
            self._touch(a[0])
        return b":%d\r\n" % n

    def _cmd_sismember(self, a):
        return b":%d\r\n" % (1 if a[1] in self.sets.get(a[0], set()) else 0)

    def _cmd_smembers(self, a):
        s = self.sets.get(a[0], set())
        out = b"*%d\r\n" % len(s)
        for m in sorted(s):
            out += self._bulk(m)
        return out

    def _cmd_zincrby(self, a):
        z = self.zsets.setdefault(a[0], {})
        z[a[2]] = z.get(a[2], 0.0) + float(a[1])
        self._touch(a[0])
        v = z[a[2]]
        s = (b"%d" % int(v)) if v == int(v) else (repr(v).encode())
        return self._bulk(s)

    def _cmd_zcard(self, a):
        return b":%d\r\n" % len(self.zsets.get(a[0], {}))

    def _cmd_zrevrange(self, a):
        z = self.zsets.get(a[0], {})
        start, stop = int(a[1]), int(a[2])
        withscores = len(a) > 3 and a[3].upper() == b"WITHSCORES"
        items = sorted(z.items(), key=lambda kv: (-kv[1], kv[0]))
        if stop == -1:
            stop = len(items) - 1
        items = items[start: stop + 1]
        out = b"*%d\r\n" % ((2 if withscores else 1) * len(items))
        for m, sc in items:
            out += self._bulk(m)
            if withscores:
                s = (b"%d" % int(sc)) if sc == int(sc) else (repr(sc).encode())
                out += self._bulk(s)
        return out
