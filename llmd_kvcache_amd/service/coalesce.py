"""Request coalescing: many concurrent single-prompt Score() calls ->
one fused kernel launch.

The round-1 verdict's top item: the fused kernel scores ~21M prompts/s
but a single-prompt RPC pays a whole kernel launch + D2H by itself, so
the service layer was ~2000x slower than the kernel.  This module gives
every service front (gRPC threadpool, ASGI event loop, the C++
wirefront) one shared micro-batching core:

 - callers enqueue (tokens, model, pods) with a per-call future and
   block/await;
 - ONE dispatcher thread drains whatever is queued (up to max_batch),
   groups by (model, pod-filter), and runs each group through
   Indexer.score_tokens_batch - a single chain+probe+score pass;
 - under light load a request is dispatched immediately (the drain
   finds just it - no artificial delay); under heavy load batches grow
   naturally toward max_batch, amortizing the launch overhead.  This is
   the same self-adjusting batching vLLM's engine loop uses; there is
   deliberately no fixed "window" timer to add latency.

The reference has no analog: its Go read path is per-request CPU work
with no launch cost to amortize (pkg/kvcache/indexer.go:132-166).
"""

from __future__ import annotations

import queue
import threading
from typing import Dict, List, Optional, Sequence, Tuple

DEFAULT_MAX_BATCH = 4096


class _Pending:
    __slots__ = ("tokens", "model", "pods_key", "pods", "event", "result",
                 "error")

    def __init__(self, tokens, model, pods):
        self.tokens = tokens
        self.model = model
        self.pods = pods
        self.pods_key = (model, tuple(sorted(pods)) if pods else ())
        self.event = threading.Event()
        self.result: Optional[Dict[str, float]] = None
        self.error: Optional[BaseException] = None


class CoalescingScorer:
    """Thread-safe batching front over Indexer.score_tokens_batch."""

    def __init__(self, indexer, max_batch: int = DEFAULT_MAX_BATCH):
        self.indexer = indexer
        self.max_batch = max_batch
        self._queue: "queue.Queue[Optional[_Pending]]" = queue.Queue()
        self._thread: Optional[threading.Thread] = None
        self._running = False
        self._lock = threading.Lock()
        self.batches = 0
        self.requests = 0

    # -- lifecycle -----------------------------------------------------
    def start(self) -> None:
        with self._lock:
            if self._running:
                return
            self._running = True
            self._thread = threading.Thread(
                target=self._dispatch_loop, name="score-coalescer",
                daemon=True)
            self._thread.start()

    def stop(self) -> None:
        with self._lock:
            if not self._running:
                return
            self._running = False
            self._queue.put(None)
            t = self._thread
            self._thread = None
        if t is not None:
            t.join(timeout=2.0)
        # A caller that passed the _running check just before stop() may
        # have enqueued AFTER the sentinel; fail those explicitly so no
        # caller blocks forever on its event (score() re-checks after its
        # put and self-drains for the symmetric window).
        self._fail_drain()

    # -- API -----------------------------------------------------------
    def score(self, tokens: Sequence[int], model: str,
              pods: Sequence[str]) -> Dict[str, float]:
        """Blocking single-prompt score; coalesced with concurrent
        callers.  Falls back to a direct call when not started."""
        if not self._running:
            return self.indexer.score_tokens(tokens, model, pods)
        p = _Pending(tokens, model, pods)
        self._queue.put(p)
        if not self._running:
            # stop() may have finished its drain between our check and
            # the put; no dispatcher will ever serve the queue again, so
            # fail-drain whatever is left ourselves (possibly including
            # our own entry) - nobody may block forever
            self._fail_drain()
        p.event.wait()
        if p.error is not None:
            raise p.error
        return p.result

    def _fail_drain(self) -> None:
        while True:
            try:
                q = self._queue.get_nowait()
            except queue.Empty:
                return
            if q is not None:
                q.error = RuntimeError("coalescing scorer stopped")
                q.event.set()

    # -- dispatcher ----------------------------------------------------
    def _dispatch_loop(self) -> None:
        while True:
            first = self._queue.get()
            if first is None:
                return
            batch: List[_Pending] = [first]
            while len(batch) < self.max_batch:
                try:
                    nxt = self._queue.get_nowait()
                except queue.Empty:
                    break
                if nxt is None:
                    self._run_batch(batch)
                    return
                batch.append(nxt)
            self._run_batch(batch)

    def _run_batch(self, batch: List[_Pending]) -> None:
        self.batches += 1
        self.requests += len(batch)
        groups: Dict[Tuple, List[_Pending]] = {}
        for p in batch:
            groups.setdefault(p.pods_key, []).append(p)
        for (model, _pods), group in groups.items():
            try:
                results = self.indexer.score_tokens_batch(
                    [p.tokens for p in group], model, group[0].pods)
                for p, r in zip(group, results):
                    p.result = r
            except BaseException as e:  # noqa: BLE001 - delivered to caller
                for p in group:
                    p.error = e
            finally:
                for p in group:
                    p.event.set()
