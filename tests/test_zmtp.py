"""ZMTP transport tests: frame codec bytes (spec golden values) and
PUB->SUB end-to-end over real TCP sockets, including the full subscriber ->
pool -> index path (the reference's offline example topology)."""

import socket
import struct
import threading
import time

import pytest

from llmd_kvcache_amd.kvevents import zmtp
from llmd_kvcache_amd.kvevents.zmtp import PubSocket, SubSocket


class TestFraming:
    def test_greeting_layout(self):
        g = zmtp._greeting()
        assert len(g) == 64
        assert g[0] == 0xFF and g[9] == 0x7F
        assert g[10:12] == bytes([3, 0])
        assert g[12:32].rstrip(b"\x00") == b"NULL"

    def test_short_frame_bytes(self):
        a, b = socket.socketpair()
        try:
            zmtp._send_frame(a, b"hello", more=True)
            data = b.recv(100)
            assert data == bytes([zmtp.FLAG_MORE, 5]) + b"hello"
        finally:
            a.close()
            b.close()

    def test_long_frame_bytes(self):
        a, b = socket.socketpair()
        try:
            body = b"x" * 300
            zmtp._send_frame(a, body)
            data = b""
            while len(data) < 309:
                data += b.recv(4096)
            assert data[0] == zmtp.FLAG_LONG
            assert struct.unpack(">Q", data[1:9])[0] == 300
            assert data[9:] == body
        finally:
            a.close()
            b.close()

    def test_ready_command_parse(self):
        body = zmtp._ready_command("PUB")
        name, props = zmtp._parse_command(body)
        assert name == "READY"
        assert props["Socket-Type"] == b"PUB"

    def test_parse_endpoint(self):
        assert zmtp.parse_endpoint("tcp://*:5557") == ("0.0.0.0", 5557)
        assert zmtp.parse_endpoint("tcp://127.0.0.1:1234") == ("127.0.0.1", 1234)
        with pytest.raises(ValueError):
            zmtp.parse_endpoint("ipc:///tmp/sock")


class TestPubSub:
    def test_pub_connect_sub_bind_roundtrip(self):
        received = []
        done = threading.Event()

        def on_message(parts):
            received.append(parts)
            done.set()

        sub = SubSocket(on_message)
        sub.subscribe(b"kv@")
        sub.bind("tcp://127.0.0.1:0")
        pub = PubSocket()
        try:
            pub.connect(f"tcp://127.0.0.1:{sub.port}")
            assert pub.wait_for_subscriber(5.0)
            pub.send_multipart(
                [b"kv@pod-a@model", struct.pack(">Q", 7), b"payload"]
            )
            assert done.wait(5.0)
            assert received[0] == [
                b"kv@pod-a@model",
                struct.pack(">Q", 7),
                b"payload",
            ]
        finally:
            pub.close()
            sub.close()

    def test_topic_prefix_filtering(self):
        received = []

        def on_message(parts):
            received.append(parts)

        sub = SubSocket(on_message)
        sub.subscribe(b"kv@")
        sub.bind("tcp://127.0.0.1:0")
        pub = PubSocket()
        try:
            pub.connect(f"tcp://127.0.0.1:{sub.port}")
            assert pub.wait_for_subscriber(5.0)
            pub.send_multipart([b"other@topic", b"x"])
            pub.send_multipart([b"kv@pod@m", b"y"])
            deadline = time.monotonic() + 5.0
            while not received and time.monotonic() < deadline:
                time.sleep(0.01)
            time.sleep(0.1)  # let any stray frame arrive
            assert len(received) == 1
            assert received[0][0] == b"kv@pod@m"
        finally:
            pub.close()
            sub.close()

    def test_multiple_publishers(self):
        received = []
        lock = threading.Lock()

        def on_message(parts):
            with lock:
                received.append(parts[0])

        sub = SubSocket(on_message)
        sub.subscribe(b"kv@")
        sub.bind("tcp://127.0.0.1:0")
        pubs = [PubSocket() for _ in range(3)]
        try:
            for i, pub in enumerate(pubs):
                pub.connect(f"tcp://127.0.0.1:{sub.port}")
                assert pub.wait_for_subscriber(5.0)
            for i, pub in enumerate(pubs):
                pub.send_multipart([f"kv@pod-{i}@m".encode(), b"p"])
            deadline = time.monotonic() + 5.0
            while time.monotonic() < deadline:
                with lock:
                    if len(received) == 3:
                        break
                time.sleep(0.01)
            with lock:
                assert sorted(received) == [
                    b"kv@pod-0@m",
                    b"kv@pod-1@m",
                    b"kv@pod-2@m",
                ]
        finally:
            for pub in pubs:
                pub.close()
            sub.close()


class TestSubscriberToIndex:
    def test_full_write_path_over_tcp(self):
        """PUB (vLLM-sim) -> ZMTP -> subscriber -> sharded pool -> index."""
        from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
        from llmd_kvcache_amd.kvblock.keys import PodEntry
        from llmd_kvcache_amd.kvblock.token_processor import (
            ChunkedTokenDatabase,
            TokenProcessorConfig,
        )
        from llmd_kvcache_amd.kvevents.events import BlockStored, EventBatch
        from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool

        index = InMemoryIndex(InMemoryIndexConfig(size=1000, pod_cache_size=10))
        tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
        pool = EventsPool(
            EventsConfig(zmq_endpoint="tcp://127.0.0.1:0", concurrency=2),
            index,
            tp,
        )
        pool.start(with_subscriber=True)
        pub = PubSocket()
        try:
            deadline = time.monotonic() + 5.0
            while pool._subscriber.port is None and time.monotonic() < deadline:
                time.sleep(0.02)
            assert pool._subscriber.port is not None
            pub.connect(f"tcp://127.0.0.1:{pool._subscriber.port}")
            assert pub.wait_for_subscriber(5.0)

            batch = EventBatch(
                ts=time.time(),
                events=[BlockStored([100], None, [1, 2, 3, 4], 4)],
            )
            pub.send_multipart(
                [b"kv@pod-a@m", struct.pack(">Q", 1), batch.encode()]
            )

            keys = tp.tokens_to_kv_block_keys(None, [1, 2, 3, 4], "m")
            deadline = time.monotonic() + 5.0
            result = {}
            while time.monotonic() < deadline:
                result = index.lookup(keys, set())
                if keys[0] in result:
                    break
                time.sleep(0.02)
            assert result.get(keys[0]) == [PodEntry("pod-a", "gpu")]
        finally:
            pub.close()
            pool.shutdown()


class TestReconnect:
    def test_publisher_reconnect_after_drop(self):
        """A dropped publisher connection must not poison the bound SUB;
        new publishers keep flowing (reference reconnect semantics,
        zmq_subscriber.go:55-77)."""
        import struct

        received = []
        lock = threading.Lock()

        def on_message(parts):
            with lock:
                received.append(parts[0])

        sub = SubSocket(on_message)
        sub.subscribe(b"kv@")
        sub.bind("tcp://127.0.0.1:0")
        try:
            pub1 = PubSocket()
            pub1.connect(f"tcp://127.0.0.1:{sub.port}")
            assert pub1.wait_for_subscriber(5.0)
            pub1.send_multipart([b"kv@p1@m", struct.pack(">Q", 1), b"x"])
            pub1.close()  # abrupt drop

            pub2 = PubSocket()
            pub2.connect(f"tcp://127.0.0.1:{sub.port}")
            assert pub2.wait_for_subscriber(5.0)
            pub2.send_multipart([b"kv@p2@m", struct.pack(">Q", 1), b"y"])
            deadline = time.monotonic() + 5.0
            while time.monotonic() < deadline:
                with lock:
                    if b"kv@p2@m" in received:
                        break
                time.sleep(0.01)
            with lock:
                assert b"kv@p2@m" in received
            pub2.close()
        finally:
            sub.close()

    def test_garbage_connection_rejected_cleanly(self):
        """A non-ZMTP client connecting to the bound SUB is dropped
        without affecting real peers."""
        import socket as socketlib
        import struct

        received = []
        done = threading.Event()

        def on_message(parts):
            received.append(parts)
            done.set()

        sub = SubSocket(on_message)
        sub.subscribe(b"kv@")
        sub.bind("tcp://127.0.0.1:0")
        try:
            garbage = socketlib.create_connection(("127.0.0.1", sub.port))
            garbage.sendall(b"GET / HTTP/1.1\r\n\r\n")
            time.sleep(0.1)
            garbage.close()

            pub = PubSocket()
            pub.connect(f"tcp://127.0.0.1:{sub.port}")
            assert pub.wait_for_subscriber(5.0)
            pub.send_multipart([b"kv@p@m", struct.pack(">Q", 1), b"z"])
            assert done.wait(5.0)
            pub.close()
        finally:
            sub.close()


class TestRebind:
    def test_subscriber_rebinds_after_listener_death(self):
        """Killing the bound listener must trigger the 5s-retry rebind
        (reference zmq_subscriber.go:55-77); publishers can then
        reconnect and flow resumes."""
        import llmd_kvcache_amd.kvevents.zmq_subscriber as zs

        received = []

        class PoolStub:
            def add_task(self, msg):
                received.append(msg)

        old_retry = zs.RETRY_INTERVAL_S
        zs.RETRY_INTERVAL_S = 0.2
        sub = zs.ZmqSubscriber(PoolStub(), "tcp://127.0.0.1:0", "kv@")
        try:
            sub.start()
            deadline = time.monotonic() + 5
            while sub.port is None and time.monotonic() < deadline:
                time.sleep(0.02)
            first_port = sub.port
            assert first_port
            old_sock = sub._sock

            # kill the listener out from under it; wait for a NEW socket
            # (the old one reports alive until its accept() times out)
            old_sock._listener.close()
            deadline = time.monotonic() + 10
            rebound = False
            while time.monotonic() < deadline:
                s = sub._sock
                if s is not None and s is not old_sock and s.alive:
                    rebound = True
                    break
                time.sleep(0.05)
            assert rebound

            pub = PubSocket()
            pub.connect(f"tcp://127.0.0.1:{sub.port}")
            assert pub.wait_for_subscriber(5.0)
            pub.send_multipart([b"kv@p@m", struct.pack(">Q", 1), b"x"])
            deadline = time.monotonic() + 5
            while not received and time.monotonic() < deadline:
                time.sleep(0.02)
            assert received
            pub.close()
        finally:
            zs.RETRY_INTERVAL_S = old_retry
            sub.stop()
