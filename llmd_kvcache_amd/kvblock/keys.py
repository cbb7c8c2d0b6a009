"""Core KV-block identity types.

Parity with reference pkg/kvcache/kvblock/index.go:138-159:
``Key = {ModelName, ChunkHash uint64}``, ``PodEntry = {PodIdentifier,
DeviceTier}``.  Implemented as frozen/slotted lightweight classes (hashable,
usable as dict keys) rather than Go structs.
"""

from __future__ import annotations

from typing import NamedTuple


class Key(NamedTuple):
    model_name: str
    chunk_hash: int  # uint64

    def __str__(self) -> str:  # matches Key.String() "model@hash"
        return f"{self.model_name}@{self.chunk_hash}"


class PodEntry(NamedTuple):
    pod_identifier: str
    device_tier: str

    def __str__(self) -> str:  # matches PodEntry.String() "pod@tier"
        return f"{self.pod_identifier}@{self.device_tier}"


DEFAULT_DEVICE_TIER = "gpu"  # kvevents/pool.go:35
