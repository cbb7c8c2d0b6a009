"""ZMQ subscriber feeding the events pool.

Parity with reference pkg/kvcache/kvevents/zmq_subscriber.go:
 - SUB socket that *binds*; vLLM publishers connect (:90);
 - topic filter, default "kv@" (pool.go:52);
 - reconnect/retry loop every 5 s on failure (:30-31,55-77);
 - messages are 3-part ``[topic, seq(BE u64), payload]`` (:124-132);
 - topic parsed as ``kv@<pod-id>@<model>`` (:136-144); messages with
   unparseable topics are dropped.

Transport is the in-repo pure-Python ZMTP 3.0 implementation (zmtp.py);
wire behavior toward pyzmq/libzmq peers is identical to the reference's
libzmq SUB socket.
"""

from __future__ import annotations

import logging
import struct
import threading
from typing import List, Optional

from .zmtp import SubSocket

logger = logging.getLogger("llmd_kvcache_amd.kvevents")

RETRY_INTERVAL_S = 5.0


class ZmqSubscriber:
    def __init__(self, pool, endpoint: str, topic_filter: str):
        self.pool = pool
        self.endpoint = endpoint
        self.topic_filter = topic_filter
        self._sock: Optional[SubSocket] = None
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.port: Optional[int] = None

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._run, name="zmq-subscriber", daemon=True
        )
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._sock is not None:
            self._sock.close()
            self._sock = None
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None

    def _run(self) -> None:
        while not self._stop.is_set():
            try:
                sock = SubSocket(self._on_message)
                sock.subscribe(self.topic_filter.encode("utf-8"))
                sock.bind(self.endpoint)
                self._sock = sock
                self.port = sock.port
                # bound; the socket's own threads handle accept/read.
                # Watch the listener: if it dies (OS error), fall through
                # to the retry loop and rebind - the reference's 5 s
                # reconnect semantics (zmq_subscriber.go:55-77).
                while not self._stop.is_set() and sock.alive:
                    self._stop.wait(0.25)
                if self._stop.is_set():
                    return
                raise ConnectionError("listener died; rebinding")
            except Exception as e:
                logger.warning(
                    "zmq-subscriber failed (%s); retrying in %.0fs",
                    e,
                    RETRY_INTERVAL_S,
                )
                if self._sock is not None:
                    self._sock.close()
                    self._sock = None
                self._stop.wait(RETRY_INTERVAL_S)

    def _on_message(self, parts: List[bytes]) -> None:
        from .pool import Message

        if len(parts) != 3:
            logger.debug("dropping %d-part message (want 3)", len(parts))
            return
        topic = parts[0].decode("utf-8", "replace")
        try:
            (seq,) = struct.unpack(">Q", parts[1])
        except struct.error:
            logger.debug("bad seq frame on topic %s", topic)
            return
        payload = parts[2]

        topic_parts = topic.split("@")
        if len(topic_parts) != 3:
            logger.debug(
                "cannot parse topic %r (want kv@<pod-id>@<model>)", topic
            )
            return
        _prefix, pod_identifier, model_name = topic_parts

        self.pool.add_task(
            Message(
                topic=topic,
                payload=payload,
                seq=seq,
                pod_identifier=pod_identifier,
                model_name=model_name,
            )
        )
