"""Multi-rank service driver: one ingest/API rank, N table shards.

Completes BASELINE.json config 3 as a *service*: rank 0 owns the external
surfaces (ZMQ KVEvents subscriber, gRPC/HTTP scoring) and replicates work
to the other ranks of the node with torch.distributed object broadcasts;
every rank holds one shard of the table (parallel/sharded.py) and
participates in the mask-merge all_reduce.

Why broadcast instead of per-rank ZMQ subscribers: vLLM publishers
connect to ONE endpoint (the SUB binds, zmq_subscriber.go:90), so a
single rank must own the socket; event decode is cheap relative to the
collective, so raw payload bytes are shipped and decoded per rank.

Protocol (collective ops driven by rank 0):
  ("events", [(pod, model, payload_bytes), ...])  apply on every rank
  ("score", hashes_i64_list, offsets, model, pods) score collectively
  ("stop",)                                        exit serve loops
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Set, Tuple

import torch
import torch.distributed as dist

from ..kvblock.gpu_index import _to_i64
from ..kvblock.token_processor import ChunkedTokenDatabase
from ..kvevents.events import decode_event_batch
from .sharded import ShardedIndex, check_registry_sync

SYNC_CHECK_INTERVAL = 64  # apply_messages calls between registry audits


class ShardedIndexService:
    def __init__(
        self,
        sharded: ShardedIndex,
        token_processor: Optional[ChunkedTokenDatabase] = None,
    ):
        self.sharded = sharded
        self.token_processor = token_processor or ChunkedTokenDatabase()
        self.rank = sharded.rank
        self.group = sharded.group
        self._running = True
        self._applies = 0

    # -- collective plumbing -------------------------------------------
    def _broadcast(self, op):
        buf = [op]
        dist.broadcast_object_list(buf, src=0, group=self.group)
        return buf[0]

    def _dispatch(self, op) -> Optional[object]:
        kind = op[0]
        if kind == "events":
            batches = []
            for pod, model, payload in op[1]:
                try:
                    batch = decode_event_batch(payload)
                except Exception:
                    continue
                batches.append((pod, model, batch.events))
            self._apply(batches)
            return None
        if kind == "score":
            _, hashes, offsets, model, pods = op
            # Stage on CPU from the broadcast payload: this is symmetric
            # across ranks (same op object everywhere), so a failure here
            # fails every rank BEFORE any collective - no deadlock.  All
            # asymmetric work (device moves, registry, probe) happens
            # inside sharded_scores, which guarantees collective
            # participation on its own failure paths.
            h = torch.tensor(hashes, dtype=torch.int64)
            offs = torch.tensor(offsets, dtype=torch.int32)
            return self.sharded.sharded_scores(h, offs, model, set(pods))
        if kind == "sync_check":
            # ROADMAP #8: loud detection of registry divergence before it
            # can misattribute merged-mask scores
            check_registry_sync(self.sharded.local, self.group)
            return None
        if kind == "stop":
            self._running = False
            return None
        raise ValueError(f"unknown op {kind!r}")

    def _apply(self, batches: List[Tuple[str, str, list]]) -> None:
        local = self.sharded.local
        if local.table.is_cuda:
            from ..kvblock.gpu_index import GpuIndex

            GpuIndex.apply_event_batches.__get__(local)(
                batches, self.token_processor
            )
        else:
            from ..kvevents.pool import digest_events

            for pod, model, events in batches:
                digest_events(local, self.token_processor, pod, model,
                              events)

    # -- rank-0 API ----------------------------------------------------
    def apply_messages(
        self, messages: Sequence[Tuple[str, str, bytes]]
    ) -> None:
        """(pod_identifier, model_name, raw msgpack payload) triples -
        exactly what the ZMQ subscriber hands the events pool."""
        assert self.rank == 0
        op = ("events", list(messages))
        self._broadcast(op)
        self._dispatch(op)
        self._applies += 1
        if self._applies % SYNC_CHECK_INTERVAL == 0:
            self.check_sync()

    def check_sync(self) -> None:
        """Audit that pod/model/tier id spaces still agree on every rank
        (auto-run every SYNC_CHECK_INTERVAL applies; callable any time).
        Raises RuntimeError on divergence - on every rank."""
        assert self.rank == 0
        op = ("sync_check",)
        self._broadcast(op)
        self._dispatch(op)

    def score(
        self,
        request_keys: Sequence,
        pod_identifier_set: Set[str],
    ) -> Dict[str, float]:
        assert self.rank == 0
        if not request_keys:
            return {}
        hashes = [_to_i64(k.chunk_hash) for k in request_keys]
        offsets = [0, len(request_keys)]
        op = ("score", hashes, offsets, request_keys[0].model_name,
              sorted(pod_identifier_set))
        self._broadcast(op)
        scores = self._dispatch(op)
        return self.sharded.local.scores_to_map(scores)[0]

    def stop(self) -> None:
        assert self.rank == 0
        op = ("stop",)
        self._broadcast(op)
        self._dispatch(op)

    # -- non-rank-0 loop -----------------------------------------------
    def serve(self) -> None:
        """Follower ranks: execute broadcast ops until stop.  A failing op
        must not kill the follower - the next collective would then hang
        every rank - so failures are logged and the loop continues.  Score
        ops are guaranteed to participate in the mask all_reduce on EVERY
        failure path: sharded_scores all_reduces zero masks when the local
        probe throws, and _dispatch participates explicitly when tensor
        staging throws before sharded_scores is reached (covered by
        tests/test_sharded.py follower-failure test).  Event failures only
        lose the local application, mirroring the pool's poison-pill
        stance."""
        assert self.rank != 0
        import logging

        logger = logging.getLogger("llmd_kvcache_amd.parallel")
        while self._running:
            op = self._broadcast(None)
            try:
                self._dispatch(op)
            except Exception:
                logger.exception("follower dispatch failed for %r",
                                 op[0] if op else op)
