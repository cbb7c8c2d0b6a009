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

Read path: fused score on the hot tier first; if the result shows early
chain termination (a prompt whose keys fell out of the hot tier), the
generic lookup merges cold-tier entries for the missing keys.  The
common case (hot working set) costs exactly one kernel.
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

    def fused_scores(self, *args, **kwargs):
        return self.hot.fused_scores(*args, **kwargs)

    def tier_weights(self, *args, **kwargs):
        return self.hot.tier_weights(*args, **kwargs)

    def scores_to_map(self, scores):
        return self.hot.scores_to_map(scores)
