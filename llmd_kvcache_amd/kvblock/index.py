"""Index contract + backend selection.

Parity with reference pkg/kvcache/kvblock/index.go:
 - ``Index`` contract: Lookup(request_keys, pod_filter) -> {key: [PodEntry]},
   Add(engine_keys, request_keys, entries), Evict(engine_key, entries),
   GetRequestKey(engine_key)  (index.go:119-135);
 - backend selection order: in-memory -> GPU (new, MI355X-native) ->
   cost-aware -> valkey -> redis, first non-None config wins (index.go:67-92);
 - optional metrics decorator wrap (index.go:94-102).

The MI355X-native addition is ``GpuIndexConfig``: an HBM3E-resident
open-addressing hash table with wave-cooperative HIP kernels (gpu_index.py).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Set

from .keys import Key, PodEntry


class Index:
    """Abstract Index contract (thread-safe implementations required)."""

    def lookup(
        self, request_keys: Sequence[Key], pod_identifier_set: Set[str]
    ) -> Dict[Key, List[PodEntry]]:
        raise NotImplementedError

    def add(
        self,
        engine_keys: Sequence[Key],
        request_keys: Sequence[Key],
        entries: Sequence[PodEntry],
    ) -> None:
        raise NotImplementedError

    def evict(self, engine_key: Key, entries: Sequence[PodEntry]) -> None:
        raise NotImplementedError

    def get_request_key(self, engine_key: Key) -> Optional[Key]:
        """Returns the request key for an engine key, or None when unmapped
        (the Go version returns an error; callers only check presence)."""
        raise NotImplementedError


@dataclass
class IndexConfig:
    in_memory: Optional["InMemoryIndexConfig"] = None
    gpu: Optional["GpuIndexConfig"] = None
    tiered: Optional["TieredIndexConfig"] = None
    native: Optional["TableIndexConfig"] = None
    cost_aware: Optional["CostAwareMemoryIndexConfig"] = None
    valkey: Optional["RedisIndexConfig"] = None
    redis: Optional["RedisIndexConfig"] = None
    enable_metrics: bool = False
    metrics_logging_interval_s: float = 0.0

    @staticmethod
    def default() -> "IndexConfig":
        from .in_memory import InMemoryIndexConfig

        return IndexConfig(in_memory=InMemoryIndexConfig())


def new_index(cfg: Optional[IndexConfig] = None) -> Index:
    """Backend selection, first non-None wins (index.go:67-92)."""
    if cfg is None:
        cfg = IndexConfig.default()

    idx: Index
    if cfg.in_memory is not None:
        from .in_memory import InMemoryIndex

        idx = InMemoryIndex(cfg.in_memory)
    elif cfg.gpu is not None:
        from .gpu_index import GpuIndex

        idx = GpuIndex(cfg.gpu)
    elif cfg.tiered is not None:
        from .tiered import TieredIndex

        idx = TieredIndex(cfg.tiered)
    elif cfg.native is not None:
        from .gpu_index import NativeIndex

        idx = NativeIndex(cfg.native)
    elif cfg.cost_aware is not None:
        from .cost_aware import CostAwareMemoryIndex

        idx = CostAwareMemoryIndex(cfg.cost_aware)
    elif cfg.valkey is not None:
        from .redis_index import ValkeyIndex

        idx = ValkeyIndex(cfg.valkey)
    elif cfg.redis is not None:
        from .redis_index import RedisIndex

        idx = RedisIndex(cfg.redis)
    else:
        raise ValueError("no valid index configuration provided")

    if cfg.enable_metrics:
        from .instrumented import InstrumentedIndex
        from ..metrics import collector

        collector.register()
        idx = InstrumentedIndex(idx)
        if cfg.metrics_logging_interval_s > 0:
            collector.start_metrics_logging(cfg.metrics_logging_interval_s)

    return idx
