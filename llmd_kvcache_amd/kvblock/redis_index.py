"""Redis/Valkey distributed index backend.

Parity with reference pkg/kvcache/kvblock/redis.go:
 - one Redis hash per request key; field = "pod@tier" (:222-238);
 - engine->request mapping via ``SET <engine-key-str> <request-hash>``
   (:227);
 - Lookup pipelines HKEYS for all keys in one RTT (:165-174) and applies
   the early-stop on the first present-but-empty key (:191-204);
 - Evict = HDEL + HLEN -> DEL when empty (:242-272);
 - valkey:// scheme is rewritten to redis:// (:78-89).

No Redis client library ships in this image, so this module includes a
minimal RESP2 client (pipelined) over stdlib sockets; tests use the
in-process FakeRedisServer (fake_redis.py) the way the reference uses
miniredis (redis_test.go:22-31).
"""

from __future__ import annotations

import socket
import threading
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Set, Tuple
from urllib.parse import urlparse

from .index import Index
from .keys import Key, PodEntry


class RespError(Exception):
    pass


class RespClient:
    """Minimal RESP2 client with pipelining."""

    def __init__(self, host: str, port: int, timeout_s: float = 5.0):
        self.sock = socket.create_connection((host, port), timeout=timeout_s)
        self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self._rfile = self.sock.makefile("rb")
        self._lock = threading.Lock()

    @staticmethod
    def _encode(args: Sequence) -> bytes:
        out = [f"*{len(args)}\r\n".encode()]
        for a in args:
            if isinstance(a, str):
                a = a.encode("utf-8")
            elif isinstance(a, int):
                a = str(a).encode()
            out.append(f"${len(a)}\r\n".encode() + a + b"\r\n")
        return b"".join(out)

    def _read_reply(self):
        line = self._rfile.readline()
        if not line:
            raise ConnectionError("redis connection closed")
        t, rest = line[:1], line[1:-2]
        if t == b"+":
            return rest.decode()
        if t == b"-":
            raise RespError(rest.decode())
        if t == b":":
            return int(rest)
        if t == b"$":
            n = int(rest)
            if n == -1:
                return None
            data = self._rfile.read(n + 2)
            return data[:-2]
        if t == b"*":
            n = int(rest)
            if n == -1:
                return None
            return [self._read_reply() for _ in range(n)]
        raise RespError(f"bad RESP type byte {t!r}")

    def execute(self, *args):
        with self._lock:
            self.sock.sendall(self._encode(args))
            return self._read_reply()

    def pipeline(self, commands: Sequence[Sequence]) -> List:
        if not commands:
            return []
        with self._lock:
            self.sock.sendall(b"".join(self._encode(c) for c in commands))
            return [self._read_reply() for _ in commands]

    def close(self) -> None:
        try:
            self.sock.close()
        except OSError:
            pass


@dataclass
class RedisIndexConfig:
    address: str = "redis://127.0.0.1:6379"
    # parity flag only; RDMA is a placeholder in the reference too
    # (redis.go:97-107)
    enable_rdma: bool = False


def _parse_address(address: str) -> Tuple[str, int]:
    # valkey:// -> redis:// rewrite (redis.go:78-89)
    if address.startswith("valkey://"):
        address = "redis://" + address[len("valkey://") :]
    if address.startswith("valkeys://") or address.startswith("rediss://"):
        raise ValueError("TLS redis/valkey endpoints are not supported")
    if not address.startswith("redis://"):
        address = "redis://" + address
    u = urlparse(address)
    return u.hostname or "127.0.0.1", u.port or 6379


class RedisIndex(Index):
    backend_name = "redis"

    def __init__(self, cfg: Optional[RedisIndexConfig] = None):
        cfg = cfg or RedisIndexConfig()
        host, port = _parse_address(cfg.address)
        self.client = RespClient(host, port)
        self.client.execute("PING")

    # key naming parity: Key.String() = "model@hash"
    @staticmethod
    def _key_str(key: Key) -> str:
        return f"{key.model_name}@{key.chunk_hash}"

    @staticmethod
    def _engine_key_str(key: Key) -> str:
        return f"engine:{key.model_name}@{key.chunk_hash}"

    def lookup(
        self, request_keys: Sequence[Key], pod_identifier_set: Set[str]
    ) -> Dict[Key, List[PodEntry]]:
        if not request_keys:
            raise ValueError("no request keys provided for lookup")
        replies = self.client.pipeline(
            [("HKEYS", self._key_str(k)) for k in request_keys]
        )
        result: Dict[Key, List[PodEntry]] = {}
        for key, fields in zip(request_keys, replies):
            if not fields:
                # absent (nil/empty hash): HKEYS can't distinguish an empty
                # from a missing hash; Redis removes empty hashes itself, so
                # treat as absent and keep scanning (redis.go:191-204).
                continue
            entries = []
            for f in fields:
                s = f.decode() if isinstance(f, bytes) else f
                pod, _, tier = s.rpartition("@")
                if not pod:
                    continue
                if pod_identifier_set and pod not in pod_identifier_set:
                    continue
                entries.append(PodEntry(pod, tier))
            if entries:
                result[key] = entries
        return result

    def add(
        self,
        engine_keys: Sequence[Key],
        request_keys: Sequence[Key],
        entries: Sequence[PodEntry],
    ) -> None:
        if not engine_keys or not request_keys or not entries:
            raise ValueError("no keys or entries provided for adding to index")
        if len(engine_keys) != len(request_keys):
            raise ValueError("mismatch between engine keys and request keys length")
        commands = []
        for engine_key, request_key in zip(engine_keys, request_keys):
            commands.append(
                ("SET", self._engine_key_str(engine_key),
                 str(request_key.chunk_hash))
            )
            for e in entries:
                commands.append(
                    ("HSET", self._key_str(request_key),
                     f"{e.pod_identifier}@{e.device_tier}", "1")
                )
        self.client.pipeline(commands)

    def evict(self, engine_key: Key, entries: Sequence[PodEntry]) -> None:
        if not entries:
            raise ValueError("no entries provided for eviction from index")
        request_key = self.get_request_key(engine_key)
        if request_key is None:
            return
        key_str = self._key_str(request_key)
        fields = [f"{e.pod_identifier}@{e.device_tier}" for e in entries]
        self.client.execute("HDEL", key_str, *fields)
        remaining = self.client.execute("HLEN", key_str)
        if not remaining:
            self.client.pipeline(
                [("DEL", key_str), ("DEL", self._engine_key_str(engine_key))]
            )

    def get_request_key(self, engine_key: Key) -> Optional[Key]:
        raw = self.client.execute("GET", self._engine_key_str(engine_key))
        if raw is None:
            return None
        s = raw.decode() if isinstance(raw, bytes) else raw
        try:
            return Key(engine_key.model_name, int(s))
        except ValueError:
            return None


class ValkeyIndex(RedisIndex):
    backend_name = "valkey"

    def __init__(self, cfg: Optional[RedisIndexConfig] = None):
        cfg = cfg or RedisIndexConfig(address="valkey://127.0.0.1:6379")
        super().__init__(cfg)
