"""Small thread-safe LRU cache used by CPU-side components.

Semantics follow hashicorp/golang-lru as used by the reference
(pkg/kvcache/kvblock/in_memory.go, pkg/tokenization/prefixstore/lru_store.go):
Get refreshes recency, Add inserts/refreshes and evicts the oldest entry
when over capacity.
"""

from __future__ import annotations

import threading
from collections import OrderedDict
from typing import Any, Hashable, Iterator, Tuple


class LRUCache:
    def __init__(self, capacity: int):
        if capacity <= 0:
            raise ValueError("LRU capacity must be positive")
        self.capacity = capacity
        self._data: "OrderedDict[Hashable, Any]" = OrderedDict()
        self._lock = threading.Lock()

    def __len__(self) -> int:
        with self._lock:
            return len(self._data)

    def __contains__(self, key: Hashable) -> bool:
        with self._lock:
            return key in self._data

    def get(self, key: Hashable) -> Tuple[Any, bool]:
        with self._lock:
            if key not in self._data:
                return None, False
            self._data.move_to_end(key)
            return self._data[key], True

    def peek(self, key: Hashable) -> Tuple[Any, bool]:
        with self._lock:
            if key not in self._data:
                return None, False
            return self._data[key], True

    def add(self, key: Hashable, value: Any) -> bool:
        """Insert/refresh. Returns True if an eviction happened."""
        with self._lock:
            if key in self._data:
                self._data.move_to_end(key)
                self._data[key] = value
                return False
            self._data[key] = value
            if len(self._data) > self.capacity:
                self._data.popitem(last=False)
                return True
            return False

    def contains_or_add(self, key: Hashable, value: Any) -> Tuple[bool, bool]:
        """Returns (contained, evicted) - mirrors golang-lru ContainsOrAdd
        (no recency update when contained)."""
        with self._lock:
            if key in self._data:
                return True, False
            self._data[key] = value
            if len(self._data) > self.capacity:
                self._data.popitem(last=False)
                return False, True
            return False, False

    def remove(self, key: Hashable) -> bool:
        with self._lock:
            if key in self._data:
                del self._data[key]
                return True
            return False

    def keys(self) -> Iterator[Hashable]:
        with self._lock:
            return iter(list(self._data.keys()))

    def purge(self) -> None:
        with self._lock:
            self._data.clear()
