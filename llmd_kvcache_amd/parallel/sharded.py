"""Multi-GPU sharded index with RCCL mask merging over xGMI.

The reference has no collectives - its global
index is one shared store.  The MI355X-native design shards the
block->pod table across the 8 GPUs of one node and merges per-shard
lookup results with RCCL (torch.distributed backend "nccl" IS RCCL on
ROCm); CPU tests use gloo with the identical code path.

Design:
 - ownership: chunk_hash % world_size == rank; the engine->request map is
   REPLICATED on every rank (it is tiny - 24 B/entry - and replication
   makes parent-chain stitching (kvevents/pool.go:279-296) rank-local);
 - writes: every rank applies the full event stream; the insert kernels
   filter main-table stores by ownership (ops cpu_insert/gpu_insert
   shard args);
 - reads: every rank probes ALL keys of the batch against its shard
   (misses for unowned keys are ~1 HBM read), producing per-key per-tier
   pod bitmasks; one all_reduce(SUM) merges them (shards are disjoint,
   so SUM == OR); the longest-prefix walk then runs locally from merged
   masks (ops score_from_masks).  Mask vectors are small (K*4*W*8 bytes)
   so the collective is latency-bound - exactly the regime xGMI
   point-to-point links handle well; there is no ring-bandwidth concern.
"""

from __future__ import annotations

from typing import Dict, Optional, Sequence, Set

import torch
import torch.distributed as dist

from ..kvblock.gpu_index import TableIndex, TableIndexConfig, _to_i64


class ShardedIndex:
    """Data-parallel sharded table over an initialized process group."""

    def __init__(
        self,
        cfg: Optional[TableIndexConfig] = None,
        group: Optional[dist.ProcessGroup] = None,
    ):
        if not dist.is_initialized():
            raise RuntimeError(
                "ShardedIndex requires torch.distributed to be initialized"
            )
        self.group = group
        self.rank = dist.get_rank(group)
        self.world_size = dist.get_world_size(group)
        cfg = cfg or TableIndexConfig()
        cfg.shard_id = self.rank
        cfg.num_shards = self.world_size
        self.local = TableIndex(cfg)
        self.device = self.local.device
        self.registry = self.local.registry

    # -- write path ----------------------------------------------------
    def add(self, engine_keys, request_keys, entries) -> None:
        """Every rank calls this with the SAME arguments (replicated event
        stream); the shard filter in the insert op keeps only owned keys."""
        self.local.add(engine_keys, request_keys, entries)

    def evict(self, engine_key, entries) -> None:
        self.local.evict(engine_key, entries)

    def get_request_key(self, engine_key):
        return self.local.get_request_key(engine_key)

    def apply_event_batches(self, batches, token_processor=None) -> None:
        self.local.apply_event_batches(batches, token_processor)

    # -- read path -----------------------------------------------------
    def sharded_scores(
        self,
        hashes: torch.Tensor,
        offsets: torch.Tensor,
        model_name: str,
        pod_identifier_set: Set[str],
        weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Batched scores for B prompts: local shard probe -> all_reduce
        mask merge -> local longest-prefix walk.  All ranks return the
        same [B, num_pods] float tensor."""
        model_id = self.registry.model_id(model_name)
        num_pods = self.local._num_pods_padded()
        filt = self.local._filter_tensor(pod_identifier_set, num_pods)
        if weights is None:
            weights = self.local.tier_weights()

        found, masks = self.local.table.lookup(
            hashes, model_id, filt, num_pods,
            n_tiers=max(1, len(self.local.registry.id_to_tier)))
        del found  # chain-cut semantics handled by the scoring walk
        dist.all_reduce(masks, op=dist.ReduceOp.SUM, group=self.group)

        offs = offsets.to(dtype=torch.int32, device=self.device)
        if self.local.table.is_cuda:
            return self.local.table.ops.gpu_score_from_masks(
                masks.contiguous(), offs, weights, num_pods
            )
        return self.local.table.ops.cpu_score_from_masks(
            masks.contiguous(), offs.cpu(), weights.cpu(), num_pods
        )

    def score_keys(
        self, request_keys: Sequence, pod_identifier_set: Set[str]
    ) -> Dict[str, float]:
        """Single-prompt convenience wrapper (Key objects in, {pod: score}
        out) - the sharded analog of Indexer.get_pod_scores steps 3-4."""
        if not request_keys:
            return {}
        hashes = torch.tensor(
            [_to_i64(k.chunk_hash) for k in request_keys],
            dtype=torch.int64,
            device=self.device,
        )
        offsets = torch.tensor([0, len(request_keys)], dtype=torch.int32)
        scores = self.sharded_scores(
            hashes, offsets, request_keys[0].model_name, pod_identifier_set
        )
        return self.local.scores_to_map(scores)[0]


def registry_fingerprint(registry) -> int:
    """Order-sensitive FNV-1a fingerprint of the interned id spaces.

    Mask merging assumes pod/model/tier ids agree on every rank (the
    replicated event stream assigns them identically).  If a rank ever
    diverged - an error path that dropped one event, a locally-registered
    extra pod - merged masks would silently attribute hits to the wrong
    pods.  ROADMAP #8: detect loudly instead."""
    from ..utils.hashing import fnv1a_64

    acc = b"\x00".join(
        "\x1f".join(ns).encode("utf-8")
        for ns in (registry.id_to_pod, registry.id_to_model,
                   registry.id_to_tier)
    )
    return fnv1a_64(acc)


def check_registry_sync(index, group=None) -> None:
    """Raise on cross-rank registry divergence (call between batches -
    it is one tiny all_gather, latency-bound)."""
    fp = registry_fingerprint(index.registry)
    world = dist.get_world_size(group)
    fps = [None] * world
    dist.all_gather_object(fps, fp, group=group)
    if any(f != fps[0] for f in fps):
        raise RuntimeError(
            f"registry divergence across ranks (fingerprints {fps}): "
            "pod/model/tier id spaces no longer agree - scores from "
            "merged masks would be misattributed. A rank dropped or "
            "reordered events; rebuild the index from the stream."
        )
