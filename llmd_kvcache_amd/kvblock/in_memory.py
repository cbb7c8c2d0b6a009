"""Default CPU in-memory index backend.

Parity with reference pkg/kvcache/kvblock/in_memory.go:
 - two-level LRU: request-key -> pod-cache (LRU of PodEntry, default 10
   pods/key), default 1e8 keys (:32-35);
 - a separate engine-key -> request-key LRU of the same size (:77-95);
 - Lookup early-stops when the prefix chain breaks: the first *present but
   empty* key cuts the search (:118-121); absent keys are skipped;
 - Evict removes pods and cleans up empty keys (:212-260).

This backend runs everywhere (no GPU needed) and doubles as the behavioral
reference for the HIP-backed GpuIndex.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Set

from ..utils.lru import LRUCache
from .index import Index
from .keys import Key, PodEntry

DEFAULT_IN_MEMORY_INDEX_SIZE = int(1e8)
DEFAULT_PODS_PER_KEY = 10


@dataclass
class InMemoryIndexConfig:
    size: int = DEFAULT_IN_MEMORY_INDEX_SIZE
    pod_cache_size: int = DEFAULT_PODS_PER_KEY


class _PodCache:
    __slots__ = ("cache", "mu")

    def __init__(self, capacity: int):
        self.cache = LRUCache(capacity)
        self.mu = threading.Lock()


class InMemoryIndex(Index):
    def __init__(self, cfg: Optional[InMemoryIndexConfig] = None):
        cfg = cfg or InMemoryIndexConfig()
        self._data = LRUCache(cfg.size)
        self._engine_to_request = LRUCache(cfg.size)
        self._pod_cache_size = cfg.pod_cache_size

    def lookup(
        self, request_keys: Sequence[Key], pod_identifier_set: Set[str]
    ) -> Dict[Key, List[PodEntry]]:
        if not request_keys:
            raise ValueError("no request keys provided for lookup")

        pods_per_key: Dict[Key, List[PodEntry]] = {}
        for key in request_keys:
            pod_cache, found = self._data.get(key)
            if not found:
                continue  # absent key: skip, keep scanning (in_memory.go:141)
            if pod_cache is None or len(pod_cache.cache) == 0:
                return pods_per_key  # chain breaks here (in_memory.go:118-121)
            entries = list(pod_cache.cache.keys())
            if not pod_identifier_set:
                pods_per_key[key] = entries
            else:
                filtered = [
                    e for e in entries if e.pod_identifier in pod_identifier_set
                ]
                if filtered:
                    pods_per_key[key] = filtered
        return pods_per_key

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

        for engine_key, request_key in zip(engine_keys, request_keys):
            self._engine_to_request.add(engine_key, request_key)

            pod_cache, found = self._data.get(request_key)
            if not found:
                new_cache = _PodCache(self._pod_cache_size)
                contained, _ = self._data.contains_or_add(request_key, new_cache)
                if contained:
                    pod_cache, found = self._data.get(request_key)
                    if not found:  # evicted in between - re-add ours
                        self._data.add(request_key, new_cache)
                        pod_cache = new_cache
                else:
                    pod_cache = new_cache

            with pod_cache.mu:
                for entry in entries:
                    pod_cache.cache.add(entry, None)

    def evict(self, engine_key: Key, entries: Sequence[PodEntry]) -> None:
        if not entries:
            raise ValueError("no entries provided for eviction from index")

        request_key, found = self._engine_to_request.get(engine_key)
        if not found:
            return
        pod_cache, found = self._data.get(request_key)
        if not found or pod_cache is None:
            self._engine_to_request.remove(engine_key)
            return

        with pod_cache.mu:
            for entry in entries:
                pod_cache.cache.remove(entry)
            is_empty = len(pod_cache.cache) == 0

        if is_empty:
            current, still_exists = self._data.get(request_key)
            if still_exists and current is not None:
                with current.mu:
                    still_empty = len(current.cache) == 0
                if still_empty:
                    self._data.remove(request_key)
                    self._engine_to_request.remove(engine_key)

    def get_request_key(self, engine_key: Key) -> Optional[Key]:
        request_key, found = self._engine_to_request.get(engine_key)
        return request_key if found else None
