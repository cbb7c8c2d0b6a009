"""Cost-aware in-memory index backend.

Parity with reference pkg/kvcache/kvblock/cost_aware_memory.go:
 - entries admitted against a byte-cost budget (default 2 GiB, :33-37);
 - cost of a key = estimated byte footprint of its pod set
   (CalculateByteSize, :126-158): key overhead + per-entry string sizes;
 - a global RW lock guards the structure (:97);
 - eviction removes lowest-recency entries until under budget (ristretto's
   admission/eviction approximated with LRU-by-recency here).
"""

from __future__ import annotations

import sys
import threading
from collections import OrderedDict
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Set

from .index import Index
from .keys import Key, PodEntry

DEFAULT_MAX_COST_BYTES = 2 * 1024 * 1024 * 1024  # 2 GiB
DEFAULT_ENGINE_MAP_SIZE = 1_000_000  # caps the engine->request mapping
# (reference cost_aware_memory.go uses an LRU "to cap mapping size")


@dataclass
class CostAwareMemoryIndexConfig:
    max_cost_bytes: int = DEFAULT_MAX_COST_BYTES
    engine_map_size: int = DEFAULT_ENGINE_MAP_SIZE


def calculate_byte_size(key: Key, pods: Dict[PodEntry, None]) -> int:
    """Estimated footprint: key string + per-pod entry strings + overhead."""
    size = sys.getsizeof(key.model_name) + 8  # model + hash
    for e in pods:
        size += sys.getsizeof(e.pod_identifier) + sys.getsizeof(e.device_tier)
        size += 16  # map-entry overhead
    return size


class CostAwareMemoryIndex(Index):
    def __init__(self, cfg: Optional[CostAwareMemoryIndexConfig] = None):
        cfg = cfg or CostAwareMemoryIndexConfig()
        self.max_cost = cfg.max_cost_bytes
        self.engine_map_size = max(1, cfg.engine_map_size)
        self._data: "OrderedDict[Key, Dict[PodEntry, None]]" = OrderedDict()
        # engine->request is a bounded LRU (cost_aware_memory.go caps this
        # mapping with an LRU); _request_to_engines is the reverse set so
        # budget eviction of a request key can drop its engine mappings.
        self._engine_to_request: "OrderedDict[Key, Key]" = OrderedDict()
        self._request_to_engines: Dict[Key, Set[Key]] = {}
        self._costs: Dict[Key, int] = {}
        self._total_cost = 0
        self._mu = threading.RLock()

    def _set_cost(self, key: Key, pods: Dict[PodEntry, None]) -> None:
        new_cost = calculate_byte_size(key, pods)
        self._total_cost += new_cost - self._costs.get(key, 0)
        self._costs[key] = new_cost

    def _map_engine(self, engine_key: Key, request_key: Key) -> None:
        old = self._engine_to_request.get(engine_key)
        if old is not None and old != request_key:
            peers = self._request_to_engines.get(old)
            if peers is not None:
                peers.discard(engine_key)
                if not peers:
                    self._request_to_engines.pop(old, None)
        self._engine_to_request[engine_key] = request_key
        self._engine_to_request.move_to_end(engine_key)
        self._request_to_engines.setdefault(request_key, set()).add(engine_key)
        while len(self._engine_to_request) > self.engine_map_size:
            ek, rk = self._engine_to_request.popitem(last=False)
            peers = self._request_to_engines.get(rk)
            if peers is not None:
                peers.discard(ek)
                if not peers:
                    self._request_to_engines.pop(rk, None)

    def _unmap_request(self, request_key: Key) -> None:
        for ek in self._request_to_engines.pop(request_key, ()):
            self._engine_to_request.pop(ek, None)

    def _evict_over_budget(self) -> None:
        while self._total_cost > self.max_cost and self._data:
            key, _pods = self._data.popitem(last=False)
            self._total_cost -= self._costs.pop(key, 0)
            self._unmap_request(key)

    def lookup(
        self, request_keys: Sequence[Key], pod_identifier_set: Set[str]
    ) -> Dict[Key, List[PodEntry]]:
        if not request_keys:
            raise ValueError("no request keys provided for lookup")
        result: Dict[Key, List[PodEntry]] = {}
        with self._mu:
            for key in request_keys:
                pods = self._data.get(key)
                if pods is None:
                    continue
                if len(pods) == 0:
                    return result  # chain break (parity with in-memory)
                self._data.move_to_end(key)
                if not pod_identifier_set:
                    result[key] = list(pods.keys())
                else:
                    filtered = [
                        e for e in pods if e.pod_identifier in pod_identifier_set
                    ]
                    if filtered:
                        result[key] = filtered
        return result

    def add(
        self,
        engine_keys: Sequence[Key],
        request_keys: Sequence[Key],
        entries: Sequence[PodEntry],
    ) -> None:
        if not engine_keys or not request_keys or not entries:
            raise ValueError("no keys or entries provided for adding to index")
        if len(engine_keys) != len(request_keys):
            raise ValueError("mismatch between engine keys and request keys length")
        with self._mu:
            for engine_key, request_key in zip(engine_keys, request_keys):
                self._map_engine(engine_key, request_key)
                pods = self._data.get(request_key)
                if pods is None:
                    pods = {}
                    self._data[request_key] = pods
                else:
                    self._data.move_to_end(request_key)
                for e in entries:
                    pods[e] = None
                self._set_cost(request_key, pods)
            self._evict_over_budget()

    def evict(self, engine_key: Key, entries: Sequence[PodEntry]) -> None:
        if not entries:
            raise ValueError("no entries provided for eviction from index")
        with self._mu:
            request_key = self._engine_to_request.get(engine_key)
            if request_key is None:
                return
            pods = self._data.get(request_key)
            if pods is None:
                self._unmap_request(request_key)
                return
            for e in entries:
                pods.pop(e, None)
            if not pods:
                self._data.pop(request_key, None)
                self._total_cost -= self._costs.pop(request_key, 0)
                self._unmap_request(request_key)
            else:
                self._set_cost(request_key, pods)

    def get_request_key(self, engine_key: Key) -> Optional[Key]:
        with self._mu:
            return self._engine_to_request.get(engine_key)

    @property
    def total_cost_bytes(self) -> int:
        with self._mu:
            return self._total_cost
