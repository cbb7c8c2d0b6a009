"""Pure-Python ZMTP 3.0 PUB/SUB transport.

The reference consumes vLLM KV events over libzmq (pkg/kvcache/kvevents/
zmq_subscriber.go): a SUB socket that *binds* (vLLM publishers connect,
:90), subscribes to a topic prefix, and receives 3-part messages
``[topic, seq(BE u64), payload]`` (:124-132).  This image has no libzmq, so
this module speaks the ZMTP 3.0 wire protocol directly (RFC 23/ZMTP,
NULL mechanism), which every libzmq 4.x peer (pyzmq in vLLM) negotiates
down to:

 - greeting: 64 bytes = signature(10) 0xFF...0x7F, version 3.0,
   mechanism "NULL" (20 bytes, null padded), as-server 0, filler;
 - handshake: READY command with a Socket-Type metadata property;
 - SUB subscriptions are messages: 0x01+topic (subscribe) / 0x00+topic
   (cancel) - the 3.0 form, accepted by all libzmq versions;
 - frames: flags(1) [MORE=0x01, LONG=0x02, COMMAND=0x04], size (1 or 8
   bytes BE), body; multipart messages chain MORE frames.

Both socket types support bind and connect so tests and the offline
example (examples/offline_events.py) can run publisher and subscriber
in-process, byte-identical to what a vLLM pod would send.
"""

from __future__ import annotations

import logging
import socket
import struct
import threading
import time
from typing import Callable, Dict, List, Optional, Set, Tuple

logger = logging.getLogger("llmd_kvcache_amd.zmtp")

SIGNATURE = b"\xff" + b"\x00" * 8 + b"\x7f"
FLAG_MORE = 0x01
FLAG_LONG = 0x02
FLAG_COMMAND = 0x04


def _greeting(as_server: bool = False) -> bytes:
    mech = b"NULL" + b"\x00" * 16
    return (
        SIGNATURE
        + bytes([3, 0])  # version 3.0
        + mech
        + bytes([1 if as_server else 0])
        + b"\x00" * 31
    )


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("peer closed during read")
        buf += chunk
    return buf


def _send_frame(sock: socket.socket, body: bytes, more: bool = False,
                command: bool = False) -> None:
    flags = (FLAG_MORE if more else 0) | (FLAG_COMMAND if command else 0)
    if len(body) > 255:
        sock.sendall(bytes([flags | FLAG_LONG]) + struct.pack(">Q", len(body)) + body)
    else:
        sock.sendall(bytes([flags, len(body)]) + body)


def _recv_frame(sock: socket.socket) -> Tuple[int, bytes]:
    flags = _recv_exact(sock, 1)[0]
    if flags & FLAG_LONG:
        size = struct.unpack(">Q", _recv_exact(sock, 8))[0]
        if size > (1 << 28):  # 256 MiB cap: untrusted peers must not be
            raise ConnectionError(f"oversized ZMTP frame: {size}")  # able to force huge allocations
    else:
        size = _recv_exact(sock, 1)[0]
    body = _recv_exact(sock, size) if size else b""
    return flags, body


def _ready_command(socket_type: str) -> bytes:
    name = b"READY"
    prop_name = b"Socket-Type"
    prop_val = socket_type.encode()
    return (
        bytes([len(name)]) + name
        + bytes([len(prop_name)]) + prop_name
        + struct.pack(">I", len(prop_val)) + prop_val
    )


def _parse_command(body: bytes) -> Tuple[str, Dict[str, bytes]]:
    if not body:
        raise ConnectionError("empty command frame")
    name_len = body[0]
    name = body[1 : 1 + name_len].decode("ascii", "replace")
    props: Dict[str, bytes] = {}
    i = 1 + name_len
    while i < len(body):
        pn_len = body[i]
        pn = body[i + 1 : i + 1 + pn_len].decode("ascii", "replace")
        i += 1 + pn_len
        (pv_len,) = struct.unpack(">I", body[i : i + 4])
        i += 4
        props[pn] = body[i : i + pv_len]
        i += pv_len
    return name, props


def _handshake(sock: socket.socket, socket_type: str) -> str:
    """Exchange greeting + READY; returns the peer's socket type."""
    sock.sendall(_greeting())
    peer_greeting = _recv_exact(sock, 64)
    if peer_greeting[0] != 0xFF or peer_greeting[9] != 0x7F:
        raise ConnectionError("bad ZMTP signature from peer")
    mechanism = peer_greeting[12:32].rstrip(b"\x00")
    if mechanism != b"NULL":
        raise ConnectionError(f"unsupported ZMTP mechanism {mechanism!r}")
    _send_frame(sock, _ready_command(socket_type), command=True)
    flags, body = _recv_frame(sock)
    if not flags & FLAG_COMMAND:
        raise ConnectionError("expected READY command from peer")
    name, props = _parse_command(body)
    if name != "READY":
        raise ConnectionError(f"expected READY, got {name}")
    return props.get("Socket-Type", b"").decode("ascii", "replace")


def parse_endpoint(endpoint: str) -> Tuple[str, int]:
    """tcp://host:port; '*' binds all interfaces."""
    if not endpoint.startswith("tcp://"):
        raise ValueError(f"only tcp:// endpoints supported, got {endpoint}")
    hostport = endpoint[len("tcp://") :]
    host, _, port = hostport.rpartition(":")
    if host == "*":
        host = "0.0.0.0"
    return host, int(port)


class SubSocket:
    """ZMTP SUB socket. Supports bind (the reference topology: vLLM PUB
    peers connect to us) and connect.  Received multipart messages are
    delivered via the on_message callback as a list of byte frames."""

    def __init__(self, on_message: Callable[[List[bytes]], None]):
        self.on_message = on_message
        self._subscriptions: Set[bytes] = set()
        self._listener: Optional[socket.socket] = None
        self._conns: Set[socket.socket] = set()
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()
        self._lock = threading.Lock()
        self.port: Optional[int] = None  # resolved after bind

    def subscribe(self, topic: bytes) -> None:
        with self._lock:
            self._subscriptions.add(topic)
            for conn in list(self._conns):
                try:
                    _send_frame(conn, b"\x01" + topic)
                except OSError:
                    pass

    def bind(self, endpoint: str) -> None:
        host, port = parse_endpoint(endpoint)
        listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        listener.bind((host, port))
        listener.listen(64)
        listener.settimeout(0.25)
        self._listener = listener
        self.port = listener.getsockname()[1]
        self._accepting = True
        t = threading.Thread(target=self._accept_loop, daemon=True,
                             name="zmtp-sub-accept")
        t.start()
        self._threads.append(t)

    def connect(self, endpoint: str) -> None:
        host, port = parse_endpoint(endpoint)
        sock = socket.create_connection((host, port), timeout=5.0)
        self._accepting = True  # connected mode counts as alive
        self._setup_peer(sock)

    def _accept_loop(self) -> None:
        try:
            while not self._stop.is_set():
                try:
                    conn, _addr = self._listener.accept()
                except socket.timeout:
                    continue
                except OSError:
                    return
                t = threading.Thread(
                    target=self._setup_peer_safe, args=(conn,), daemon=True,
                    name="zmtp-sub-conn",
                )
                t.start()
                self._threads.append(t)
        finally:
            self._accepting = False

    @property
    def alive(self) -> bool:
        """True while the bound listener is still accepting peers."""
        return getattr(self, "_accepting", False) and not self._stop.is_set()

    def _setup_peer_safe(self, sock: socket.socket) -> None:
        try:
            self._setup_peer(sock)
        except Exception as e:
            logger.debug("ZMTP peer setup failed: %s", e)
            try:
                sock.close()
            except OSError:
                pass

    def _setup_peer(self, sock: socket.socket) -> None:
        sock.settimeout(10.0)
        sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        peer_type = _handshake(sock, "SUB")
        if peer_type not in ("PUB", "XPUB", ""):
            raise ConnectionError(f"SUB cannot talk to {peer_type}")
        with self._lock:
            for topic in self._subscriptions:
                _send_frame(sock, b"\x01" + topic)
            self._conns.add(sock)
        sock.settimeout(0.25)
        if threading.current_thread().name.startswith("zmtp-sub-conn"):
            self._read_loop(sock)
        else:
            t = threading.Thread(target=self._read_loop, args=(sock,),
                                 daemon=True, name="zmtp-sub-read")
            t.start()
            self._threads.append(t)

    def _read_loop(self, sock: socket.socket) -> None:
        parts: List[bytes] = []
        try:
            while not self._stop.is_set():
                try:
                    flags, body = _recv_frame(sock)
                except socket.timeout:
                    continue
                if flags & FLAG_COMMAND:
                    continue  # PING etc. - ignore
                parts.append(body)
                if not flags & FLAG_MORE:
                    try:
                        self.on_message(parts)
                    except Exception:
                        logger.exception("on_message callback failed")
                    parts = []
        except (ConnectionError, OSError):
            pass
        finally:
            with self._lock:
                self._conns.discard(sock)
            try:
                sock.close()
            except OSError:
                pass

    def close(self) -> None:
        self._stop.set()
        if self._listener is not None:
            try:
                self._listener.close()
            except OSError:
                pass
        with self._lock:
            for conn in list(self._conns):
                try:
                    conn.close()
                except OSError:
                    pass
            self._conns.clear()
        for t in self._threads:
            t.join(timeout=1.0)


class PubSocket:
    """ZMTP PUB socket (connect or bind).  Tracks peer subscriptions and
    prefix-filters outgoing multipart messages - the behavior a vLLM
    publisher (pyzmq PUB) exhibits toward our bound SUB."""

    def __init__(self) -> None:
        self._peers: Dict[socket.socket, Set[bytes]] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._listener: Optional[socket.socket] = None
        self.port: Optional[int] = None

    def connect(self, endpoint: str) -> None:
        host, port = parse_endpoint(endpoint)
        deadline = time.monotonic() + 10.0
        while True:
            try:
                sock = socket.create_connection((host, port), timeout=5.0)
                break
            except OSError:
                if time.monotonic() > deadline:
                    raise
                time.sleep(0.1)
        self._setup_peer(sock)

    def bind(self, endpoint: str) -> None:
        host, port = parse_endpoint(endpoint)
        listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        listener.bind((host, port))
        listener.listen(64)
        listener.settimeout(0.25)
        self._listener = listener
        self.port = listener.getsockname()[1]
        t = threading.Thread(target=self._accept_loop, daemon=True,
                             name="zmtp-pub-accept")
        t.start()
        self._threads.append(t)

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _addr = self._listener.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            try:
                self._setup_peer(conn)
            except Exception as e:
                logger.debug("ZMTP pub peer setup failed: %s", e)

    def _setup_peer(self, sock: socket.socket) -> None:
        sock.settimeout(10.0)
        sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        peer_type = _handshake(sock, "PUB")
        if peer_type not in ("SUB", "XSUB", ""):
            raise ConnectionError(f"PUB cannot talk to {peer_type}")
        with self._lock:
            self._peers[sock] = set()
        sock.settimeout(0.25)
        t = threading.Thread(target=self._sub_read_loop, args=(sock,),
                             daemon=True, name="zmtp-pub-read")
        t.start()
        self._threads.append(t)

    def _sub_read_loop(self, sock: socket.socket) -> None:
        """Reads subscription (0x01) / cancel (0x00) messages from the SUB."""
        try:
            while not self._stop.is_set():
                try:
                    flags, body = _recv_frame(sock)
                except socket.timeout:
                    continue
                if flags & FLAG_COMMAND:
                    try:
                        name, _props = _parse_command(body)
                    except Exception:
                        continue  # malformed command from peer: ignore
                    if name == "SUBSCRIBE":  # ZMTP 3.1 form
                        body = b"\x01" + body[1 + len(b"SUBSCRIBE") :]
                    elif name == "CANCEL":
                        body = b"\x00" + body[1 + len(b"CANCEL") :]
                    else:
                        continue
                if not body:
                    continue
                with self._lock:
                    subs = self._peers.get(sock)
                    if subs is None:
                        return
                    if body[0] == 0x01:
                        subs.add(body[1:])
                    elif body[0] == 0x00:
                        subs.discard(body[1:])
        except (ConnectionError, OSError):
            pass
        finally:
            with self._lock:
                self._peers.pop(sock, None)
            try:
                sock.close()
            except OSError:
                pass

    def wait_for_subscriber(self, timeout: float = 10.0) -> bool:
        """Blocks until at least one peer has at least one subscription."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            with self._lock:
                if any(self._peers.values()):
                    return True
            time.sleep(0.02)
        return False

    def send_multipart(self, parts: List[bytes]) -> None:
        """Sends to every peer whose subscription prefix-matches part 0."""
        topic = parts[0]
        with self._lock:
            targets = [
                s
                for s, subs in self._peers.items()
                if any(topic.startswith(p) for p in subs)
            ]
        for sock in targets:
            try:
                for i, part in enumerate(parts):
                    _send_frame(sock, part, more=(i < len(parts) - 1))
            except OSError:
                with self._lock:
                    self._peers.pop(sock, None)

    def close(self) -> None:
        self._stop.set()
        if self._listener is not None:
            try:
                self._listener.close()
            except OSError:
                pass
        with self._lock:
            for sock in list(self._peers):
                try:
                    sock.close()
                except OSError:
                    pass
            self._peers.clear()
        for t in self._threads:
            t.join(timeout=1.0)
