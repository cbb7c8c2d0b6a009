"""Sharded ordered event-processing pool (the write path).

Parity with reference pkg/kvcache/kvevents/pool.go:
 - FNV-1a(pod_identifier) % concurrency selects the worker queue, so events
   for one pod are always processed in order (:132-144); default 4 workers
   (:48-54);
 - processEvent decodes the msgpack batch; poison pills are dropped, not
   retried (:182-187);
 - digestEvents (:246-338): BlockStored -> engine keys from reported hashes,
   request keys recomputed locally from token_ids continuing the parent
   chain via Index.get_request_key, then Index.add; BlockRemoved ->
   Index.evict per hash; AllBlocksCleared -> no-op; Medium (lowercased)
   selects the device tier, defaulting to "gpu" (:35,255-259).

MI355X note: with a GpuIndex the workers burst-drain their queues and
apply up to 64 messages per k_apply_events kernel launch
(_process_burst_gpu), keeping per-pod ordering within each shard queue.
"""

from __future__ import annotations

import logging
import queue
import threading
from dataclasses import dataclass
from typing import List, Optional

from ..kvblock.index import Index
from ..kvblock.keys import DEFAULT_DEVICE_TIER, Key, PodEntry
from ..kvblock.token_processor import ChunkedTokenDatabase
from ..utils.hashing import fnv1a_32
from .events import (
    AllBlocksCleared,
    BlockRemoved,
    BlockStored,
    DecodeError,
    decode_event_batch,
    get_hash_as_uint64,
)

logger = logging.getLogger("llmd_kvcache_amd.kvevents")

DEFAULT_ZMQ_ENDPOINT = "tcp://*:5557"
DEFAULT_TOPIC_FILTER = "kv@"
DEFAULT_CONCURRENCY = 4


@dataclass
class EventsConfig:
    zmq_endpoint: str = DEFAULT_ZMQ_ENDPOINT
    topic_filter: str = DEFAULT_TOPIC_FILTER
    concurrency: int = DEFAULT_CONCURRENCY


@dataclass
class Message:
    topic: str
    payload: bytes
    seq: int
    pod_identifier: str
    model_name: str


class EventsPool:
    """Sharded worker pool; per-pod ordering guaranteed by queue selection."""

    def __init__(
        self,
        cfg: Optional[EventsConfig],
        index: Index,
        token_processor: Optional[ChunkedTokenDatabase] = None,
    ):
        self.cfg = cfg or EventsConfig()
        self.index = index
        self.token_processor = token_processor or ChunkedTokenDatabase()
        self.queues: List["queue.Queue[Optional[Message]]"] = [
            queue.Queue() for _ in range(self.cfg.concurrency)
        ]
        self._threads: List[threading.Thread] = []
        self._subscriber = None
        self._running = False

    # -- lifecycle -----------------------------------------------------
    def start(self, with_subscriber: bool = True) -> None:
        if self._running:
            return
        self._running = True
        for i in range(self.cfg.concurrency):
            t = threading.Thread(
                target=self._worker, args=(i,), name=f"kvevents-worker-{i}",
                daemon=True,
            )
            t.start()
            self._threads.append(t)
        if with_subscriber:
            from .zmq_subscriber import ZmqSubscriber

            self._subscriber = ZmqSubscriber(
                self, self.cfg.zmq_endpoint, self.cfg.topic_filter
            )
            self._subscriber.start()

    def shutdown(self) -> None:
        if not self._running:
            return
        self._running = False
        if self._subscriber is not None:
            self._subscriber.stop()
            self._subscriber = None
        for q in self.queues:
            q.put(None)
        for t in self._threads:
            t.join(timeout=2.0)
        self._threads.clear()

    def drain(self) -> None:
        """Block until all queues are empty (test/bench helper)."""
        for q in self.queues:
            q.join()

    # -- ingestion -----------------------------------------------------
    def add_task(self, msg: Message) -> None:
        shard = fnv1a_32(msg.pod_identifier.encode("utf-8")) % len(self.queues)
        self.queues[shard].put(msg)

    # -- workers -------------------------------------------------------
    def _worker(self, index: int) -> None:
        q = self.queues[index]
        fast = hasattr(self.index, "apply_event_batches") and getattr(
            getattr(self.index, "table", None), "is_cuda", False
        )
        while True:
            msg = q.get()
            if msg is None:
                q.task_done()
                return
            if not fast:
                try:
                    self.process_event(msg)
                finally:
                    q.task_done()
                continue
            # GPU fast path: drain a burst of queued messages and apply
            # them in ONE on-device kernel launch (per-pod ordering is
            # preserved: this queue owns its pods, and the kernel
            # processes each pod group's events serially in FIFO order).
            burst: List[Message] = [msg]
            stop = False
            while len(burst) < 64:
                try:
                    nxt = q.get_nowait()
                except queue.Empty:
                    break
                if nxt is None:
                    stop = True
                    break
                burst.append(nxt)
            try:
                self._process_burst_gpu(burst)
            finally:
                for _ in burst:
                    q.task_done()
                if stop:
                    q.task_done()
            if stop:
                return

    def _process_burst_gpu(self, burst: List[Message]) -> None:
        batches = []
        for msg in burst:
            try:
                batch = decode_event_batch(msg.payload)
            except DecodeError as e:
                logger.debug("dropping poison-pill message: %s", e)
                continue
            batches.append((msg.pod_identifier, msg.model_name, batch.events))
        if not batches:
            return
        # mixed-model bursts apply in ONE launch: the kernels take a
        # per-event model id (model_of), so no host-side model split
        try:
            self.index.apply_event_batches(batches, self.token_processor)
        except Exception:
            logger.exception("on-device event application failed; "
                             "falling back to per-event path")
            for pod, model_name, events in batches:
                self.digest_events(pod, model_name, events)

    def process_event(self, msg: Message) -> None:
        try:
            batch = decode_event_batch(msg.payload)
        except DecodeError as e:
            logger.debug("dropping poison-pill message: %s", e)
            return
        self.digest_events(msg.pod_identifier, msg.model_name, batch.events)

    def digest_events(self, pod_identifier: str, model_name: str, events) -> None:
        digest_events(self.index, self.token_processor, pod_identifier,
                      model_name, events)


def _tier(medium: Optional[str]) -> str:
    return medium.lower() if medium else DEFAULT_DEVICE_TIER


def digest_events(index: Index, token_processor: ChunkedTokenDatabase,
                  pod_identifier: str, model_name: str, events) -> None:
    """Apply decoded events to an index (pool.go:246-338 semantics);
    usable standalone by the tiered index and the sharded service."""
    logger.debug("digesting %d events from %s/%s", len(events),
                 pod_identifier, model_name)
    for ev in events:
        if isinstance(ev, BlockStored):
            _digest_block_stored(index, token_processor, pod_identifier,
                                 model_name, ev)
        elif isinstance(ev, BlockRemoved):
            _digest_block_removed(index, pod_identifier, model_name, ev)
        elif isinstance(ev, AllBlocksCleared):
            continue
        else:
            logger.debug("unknown event %r from %s", ev, pod_identifier)


def _digest_block_stored(index, token_processor, pod_identifier: str,
                         model_name: str, ev: BlockStored) -> None:
    entries = [PodEntry(pod_identifier, _tier(ev.medium))]

    engine_keys: List[Key] = []
    for raw in ev.block_hashes:
        try:
            engine_keys.append(Key(model_name, get_hash_as_uint64(raw)))
        except DecodeError as e:
            logger.debug("bad block hash %r: %s", raw, e)
            continue

    parent_request_key: Optional[Key] = None
    if ev.parent_block_hash is not None:
        try:
            parent_engine_key = Key(
                model_name, get_hash_as_uint64(ev.parent_block_hash)
            )
        except DecodeError as e:
            logger.debug("bad parent hash %r: %s", ev.parent_block_hash, e)
            return
        parent_request_key = index.get_request_key(parent_engine_key)

    request_keys = token_processor.tokens_to_kv_block_keys(
        parent_request_key, ev.token_ids, model_name
    )

    if engine_keys:
        try:
            index.add(engine_keys, request_keys, entries)
        except Exception as e:
            logger.debug("failed to add event to index: %s", e)


def _digest_block_removed(index, pod_identifier: str, model_name: str,
                          ev: BlockRemoved) -> None:
    entries = [PodEntry(pod_identifier, _tier(ev.medium))]
    for raw in ev.block_hashes:
        try:
            engine_key = Key(model_name, get_hash_as_uint64(raw))
        except DecodeError as e:
            logger.debug("bad block hash %r: %s", raw, e)
            continue
        try:
            index.evict(engine_key, entries)
        except Exception as e:
            logger.debug("failed to evict from index: %s", e)
