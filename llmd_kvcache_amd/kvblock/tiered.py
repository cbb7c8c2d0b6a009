"""Two-tier GPU+CPU index (BASELINE.json config 4).

The HBM-resident table is the hot tier; a (much larger) host-memory table
of the same layout is the capacity tier.  The reference's closest analog
is the cost-aware index's byte-budget eviction (cost_aware_memory.go) -
here capacity pressure on the GPU tier resolves by approximate-LRU slot
stealing (kvidx table semantics) while the CPU tier retains the long
tail, so a hot-tier miss can still be served.

Write path: adds go to BOTH tiers (the CPU tier write is cheap relative
to event decode; this keeps the cold tier a superset without needing
eviction callbacks out of the GPU kernel).

Read path: one hot-tier lookup kernel yields per-key masks and found
flags; fully hot-resident batches (the common case) score straight from
the masks on-device. Any hot MISS routes through the generic merged
lookup - which serves the cold entries AND promotes them back into HBM -
so cold-tier data is never invisible to the fast path.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Set

from .gpu_index import GpuIndex, GpuIndexConfig, NativeIndex, TableIndexConfig
from .index import Index
from .keys import Key, PodEntry


class TieredIndexConfig:
    def __init__(
        self,
        hot: Optional[GpuIndexConfig] = None,
        cold: Optional[TableIndexConfig] = None,
    ):
        self.hot = hot or GpuIndexConfig()
        self.cold = cold or TableIndexConfig(
            capacity=max(self.hot.capacity * 8, 1 << 24)
        )
        self.cold.device = "cpu"


class TieredIndex(Index):
    def __init__(
        self,
        cfg: Optional[TieredIndexConfig] = None,
        hot: Optional[Index] = None,
        cold: Optional[Index] = None,
    ):
        """Tiers are injectable for CPU-only tests (hot=NativeIndex);
        production default is GpuIndex hot + NativeIndex cold."""
        cfg = cfg or TieredIndexConfig()
        self.hot = hot or GpuIndex(cfg.hot)
        # share one registry so pod/model/tier ids agree across tiers
        self.cold = cold or NativeIndex(cfg.cold, registry=self.hot.registry)
        self.registry = self.hot.registry

    # -- Index contract ------------------------------------------------
    def lookup(
        self, request_keys: Sequence[Key], pod_identifier_set: Set[str]
    ) -> Dict[Key, List[PodEntry]]:
        hot = self.hot.lookup(request_keys, pod_identifier_set)
        if len(hot) == len(request_keys):
            return hot
        missing = [k for k in request_keys if k not in hot]
        cold = self.cold.lookup(missing, pod_identifier_set)
        merged = dict(hot)
        for k, entries in cold.items():
            merged.setdefault(k, entries)
        self._promote(cold)
        return merged

    def _promote(self, cold_hits: Dict[Key, List[PodEntry]]) -> None:
        """Re-warm the HBM tier with cold-tier hits so the next request
        for this prefix is served by the single fused kernel again.
        write_emap=False: the engine map already carries these mappings
        (adds write both tiers) and a request-key self-mapping would
        corrupt get_request_key()."""
        from .gpu_index import TableIndex

        if not cold_hits or not isinstance(self.hot, TableIndex):
            return
        for k, entries in cold_hits.items():
            if entries:
                self.hot.add([k], [k], entries, write_emap=False)

    def add(
        self,
        engine_keys: Sequence[Key],
        request_keys: Sequence[Key],
        entries: Sequence[PodEntry],
    ) -> None:
        self.hot.add(engine_keys, request_keys, entries)
        self.cold.add(engine_keys, request_keys, entries)

    def evict(self, engine_key: Key, entries: Sequence[PodEntry]) -> None:
        self.hot.evict(engine_key, entries)
        self.cold.evict(engine_key, entries)

    def get_request_key(self, engine_key: Key) -> Optional[Key]:
        rk = self.hot.get_request_key(engine_key)
        if rk is None:
            rk = self.cold.get_request_key(engine_key)
        return rk

    # -- fast paths ----------------------------------------------------
    @property
    def table(self):  # events-pool burst path detection
        return self.hot.table

    @property
    def device(self):
        return self.hot.device

    def apply_event_batches(self, batches, token_processor=None) -> None:
        self.hot.apply_event_batches(batches, token_processor)
        # mirror into the capacity tier through the CPU digest path
        from ..kvevents.pool import digest_events

        if token_processor is None:
            from .token_processor import ChunkedTokenDatabase

            token_processor = ChunkedTokenDatabase()
        for pod, model, events in batches:
            digest_events(self.cold, token_processor, pod, model, events)

    def fused_scores(self, hashes, counts_or_offsets, model_name,
                     pod_identifier_set, weights=None, max_k=None):
        """Read fast path with cold-tier correctness: one hot-tier lookup
        kernel produces per-key masks AND found flags; when every key is
        hot-resident (the common case) the scores come straight from the
        masks on-device. Any hot miss routes the batch through the
        generic merged lookup (which also promotes the cold hits back
        into HBM) and the Python scorer."""
        import torch

        from ..scorer import LongestPrefixScorer
        from .gpu_index import _to_u64

        hot = self.hot
        model_id = hot.registry.model_id(model_name)
        num_pods = hot._num_pods_padded()
        filt = hot._filter_tensor(pod_identifier_set, num_pods)
        if weights is None:
            weights = hot.tier_weights()
        n_tiers = max(1, len(hot.registry.id_to_tier))
        found, masks = hot.table.lookup(hashes, model_id, filt, num_pods,
                                        n_tiers=n_tiers)
        if hot.table.is_cuda:
            offsets = counts_or_offsets
        else:  # CPU fused convention passes per-prompt counts
            counts = counts_or_offsets.to(torch.int32)
            offsets = torch.zeros(counts.numel() + 1, dtype=torch.int32)
            torch.cumsum(counts, 0, out=offsets[1:].view(counts.numel()))
        if int((found == 0).sum()) == 0:
            ops = hot.table.ops
            fn = (ops.gpu_score_from_masks if hot.table.is_cuda
                  else ops.cpu_score_from_masks)
            return fn(masks.contiguous(), offsets.to(found.device)
                      if hot.table.is_cuda else offsets.cpu(),
                      weights, num_pods)
        # hot miss: merged lookup (promotes) + reference scorer
        hashes_l = hashes.cpu().tolist()
        offs_l = offsets.cpu().tolist()
        scorer = LongestPrefixScorer(
            medium_weights={name: float(weights[i])
                            for i, name in enumerate(hot.registry.id_to_tier)})
        out = torch.zeros((len(offs_l) - 1, num_pods), dtype=torch.float32,
                          device=hot.device)
        for b in range(len(offs_l) - 1):
            keys = [Key(model_name, _to_u64(h))
                    for h in hashes_l[offs_l[b]:offs_l[b + 1]]]
            if not keys:
                continue
            merged = self.lookup(keys, pod_identifier_set)
            for pod, score in scorer.score(keys, merged).items():
                pid = hot.registry.pod_to_id.get(pod)
                if pid is not None and pid < num_pods:
                    out[b, pid] = score
        return out

    def tier_weights(self, *args, **kwargs):
        return self.hot.tier_weights(*args, **kwargs)

    def scores_to_map(self, scores):
        return self.hot.scores_to_map(scores)
