"""Metrics decorator around any Index backend.

Parity with reference pkg/kvcache/kvblock/instrumented_index.go:25-92:
records admissions, evictions, lookup requests/latency, per-lookup hits
and the max consecutive pod hit count (computed off the hot path).

The hit-metrics computation runs on ONE shared bounded worker (the
reference spawns a goroutine per lookup, which is cheap in Go; a Python
thread per lookup is not - under metrics-enabled load the create/join
cost dominates). When the queue is full the sample is dropped: metrics
are best-effort, the hot path never blocks.
"""

from __future__ import annotations

import queue
import threading
import time
from typing import Dict, List, Optional, Sequence, Set

from ..metrics import collector
from .index import Index
from .keys import Key, PodEntry

_HIT_QUEUE_MAX = 1024

_hit_queue: "queue.Queue" = queue.Queue(maxsize=_HIT_QUEUE_MAX)
_hit_worker_lock = threading.Lock()
_hit_worker: Optional[threading.Thread] = None


def _hit_worker_loop() -> None:
    while True:
        request_keys, result = _hit_queue.get()
        try:
            InstrumentedIndex._record_hit_metrics(request_keys, result)
        except Exception:
            pass


def _ensure_hit_worker() -> None:
    global _hit_worker
    if _hit_worker is not None and _hit_worker.is_alive():
        return
    with _hit_worker_lock:
        if _hit_worker is None or not _hit_worker.is_alive():
            _hit_worker = threading.Thread(
                target=_hit_worker_loop, name="kvidx-hit-metrics", daemon=True
            )
            _hit_worker.start()


class InstrumentedIndex(Index):
    def __init__(self, inner: Index):
        self.inner = inner

    def lookup(
        self, request_keys: Sequence[Key], pod_identifier_set: Set[str]
    ) -> Dict[Key, List[PodEntry]]:
        if collector.lookup_requests is not None:
            collector.lookup_requests.inc()
        t0 = time.monotonic()
        result = self.inner.lookup(request_keys, pod_identifier_set)
        if collector.lookup_latency is not None:
            collector.lookup_latency.observe(time.monotonic() - t0)
        if collector.lookup_hits is not None:
            collector.lookup_hits.inc(len(result))
        # max consecutive pod-hit count computed asynchronously on the
        # shared bounded worker (instrumented_index.go:62,71-92)
        _ensure_hit_worker()
        try:
            _hit_queue.put_nowait((list(request_keys), dict(result)))
        except queue.Full:
            pass  # drop the sample, never block the lookup
        return result

    @staticmethod
    def _record_hit_metrics(
        request_keys: List[Key], result: Dict[Key, List[PodEntry]]
    ) -> None:
        try:
            pod_hits: Dict[str, int] = {}
            active: Optional[Set[str]] = None
            for key in request_keys:
                pods = {e.pod_identifier for e in result.get(key, [])}
                active = pods if active is None else (active & pods)
                if not active:
                    break
                for p in active:
                    pod_hits[p] = pod_hits.get(p, 0) + 1
            max_hits = max(pod_hits.values()) if pod_hits else 0
            if collector.max_pod_hit_count is not None:
                collector.max_pod_hit_count.observe(max_hits)
        except Exception:
            pass

    def add(
        self,
        engine_keys: Sequence[Key],
        request_keys: Sequence[Key],
        entries: Sequence[PodEntry],
    ) -> None:
        self.inner.add(engine_keys, request_keys, entries)
        if collector.admissions is not None:
            collector.admissions.inc(len(request_keys) * len(entries))

    def evict(self, engine_key: Key, entries: Sequence[PodEntry]) -> None:
        self.inner.evict(engine_key, entries)
        if collector.evictions is not None:
            collector.evictions.inc(len(entries))

    def get_request_key(self, engine_key: Key) -> Optional[Key]:
        return self.inner.get_request_key(engine_key)

    def __getattr__(self, name):
        return getattr(self.inner, name)
