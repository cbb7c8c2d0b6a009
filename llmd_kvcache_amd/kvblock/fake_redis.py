"""In-process fake Redis server (RESP2 subset) for tests.

The role miniredis plays in the reference test suite
(pkg/kvcache/kvblock/redis_test.go:22-31): a real TCP server speaking
enough RESP to back RedisIndex/ValkeyIndex without a cluster.
Supported: PING, SET, GET, DEL, HSET, HDEL, HKEYS, HLEN, FLUSHALL.
"""

from __future__ import annotations

import socket
import threading
from typing import Dict


class FakeRedisServer:
    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self._listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._listener.bind((host, port))
        self._listener.listen(16)
        self._listener.settimeout(0.25)
        self.port = self._listener.getsockname()[1]
        self._strings: Dict[bytes, bytes] = {}
        self._hashes: Dict[bytes, Dict[bytes, bytes]] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._threads = []

    def start(self) -> None:
        t = threading.Thread(target=self._accept_loop, daemon=True,
                             name="fake-redis-accept")
        t.start()
        self._threads.append(t)

    def stop(self) -> None:
        self._stop.set()
        try:
            self._listener.close()
        except OSError:
            pass

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _ = self._listener.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            t = threading.Thread(target=self._serve, args=(conn,),
                                 daemon=True, name="fake-redis-conn")
            t.start()
            self._threads.append(t)

    def _serve(self, conn: socket.socket) -> None:
        rfile = conn.makefile("rb")
        try:
            while not self._stop.is_set():
                line = rfile.readline()
                if not line:
                    return
                if not line.startswith(b"*"):
                    conn.sendall(b"-ERR protocol error\r\n")
                    return
                n = int(line[1:-2])
                args = []
                for _ in range(n):
                    hdr = rfile.readline()
                    ln = int(hdr[1:-2])
                    args.append(rfile.read(ln + 2)[:-2])
                conn.sendall(self._dispatch(args))
        except (OSError, ValueError):
            pass
        finally:
            try:
                conn.close()
            except OSError:
                pass

    @staticmethod
    def _bulk(v) -> bytes:
        if v is None:
            return b"$-1\r\n"
        if isinstance(v, str):
            v = v.encode()
        return b"$%d\r\n%s\r\n" % (len(v), v)

    def _dispatch(self, args) -> bytes:
        cmd = args[0].upper()
        with self._lock:
            if cmd == b"PING":
                return b"+PONG\r\n"
            if cmd == b"SET":
                self._strings[args[1]] = args[2]
                return b"+OK\r\n"
            if cmd == b"GET":
                return self._bulk(self._strings.get(args[1]))
            if cmd == b"DEL":
                n = 0
                for k in args[1:]:
                    n += int(self._strings.pop(k, None) is not None)
                    n += int(self._hashes.pop(k, None) is not None)
                return b":%d\r\n" % n
            if cmd == b"HSET":
                h = self._hashes.setdefault(args[1], {})
                added = 0
                for i in range(2, len(args) - 1, 2):
                    added += int(args[i] not in h)
                    h[args[i]] = args[i + 1]
                return b":%d\r\n" % added
            if cmd == b"HDEL":
                h = self._hashes.get(args[1], {})
                n = 0
                for f in args[2:]:
                    n += int(h.pop(f, None) is not None)
                if not h:
                    self._hashes.pop(args[1], None)
                return b":%d\r\n" % n
            if cmd == b"HKEYS":
                h = self._hashes.get(args[1], {})
                out = b"*%d\r\n" % len(h)
                for f in h:
                    out += self._bulk(f)
                return out
            if cmd == b"HLEN":
                return b":%d\r\n" % len(self._hashes.get(args[1], {}))
            if cmd == b"FLUSHALL":
                self._strings.clear()
                self._hashes.clear()
                return b"+OK\r\n"
        return b"-ERR unknown command\r\n"
