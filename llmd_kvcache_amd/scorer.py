"""Pod scoring strategies.

Parity with reference pkg/kvcache/kvblock_scorer.go:
 - LongestPrefixScorer (:108-151): starting at block 0, keep the set of pods
   active; each consecutive key a pod holds adds that key's per-pod max
   device-tier weight to the pod's score; a pod drops out (and stops
   accumulating) at the first key it misses; break when the active set is
   empty.
 - tier weights come from KVCacheBackendConfig (backend.go:19-31); defaults
   gpu=1.0, cpu=0.8; unknown tiers weigh 1.0 (kvblock_scorer.go:93-99).

On the GPU path the same computation is fused into the lookup kernel
(ops/csrc/kvidx_hip.hip) operating on per-key per-tier pod bitmasks; this
Python implementation is the golden reference for it.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

from .kvblock.keys import Key, PodEntry

LONGEST_PREFIX_MATCH = "LongestPrefix"


@dataclass
class KVCacheBackendConfig:
    name: str
    weight: float


def default_kv_cache_backend_configs() -> List[KVCacheBackendConfig]:
    return [
        KVCacheBackendConfig(name="gpu", weight=1.0),
        KVCacheBackendConfig(name="cpu", weight=0.8),
    ]


@dataclass
class KVBlockScorerConfig:
    scoring_strategy: str = LONGEST_PREFIX_MATCH
    backend_configs: List[KVCacheBackendConfig] = field(
        default_factory=default_kv_cache_backend_configs
    )


def _get_max_weight(
    entries: Sequence[PodEntry], pod_id: str, medium_weights: Dict[str, float]
) -> float:
    max_weight = 0.0
    for entry in entries:
        if entry.pod_identifier == pod_id:
            weight = medium_weights.get(entry.device_tier, 1.0)
            if weight > max_weight:
                max_weight = weight
    return max_weight


class LongestPrefixScorer:
    def __init__(self, medium_weights: Dict[str, float]):
        self.medium_weights = medium_weights

    @property
    def strategy(self) -> str:
        return LONGEST_PREFIX_MATCH

    def score(
        self,
        keys: Sequence[Key],
        key_to_pods: Dict[Key, List[PodEntry]],
    ) -> Dict[str, float]:
        pod_scores: Dict[str, float] = {}
        if not keys:
            return pod_scores

        pods_first = key_to_pods.get(keys[0], [])
        active = {p.pod_identifier for p in pods_first}
        for pod in active:
            pod_scores[pod] = _get_max_weight(pods_first, pod, self.medium_weights)

        for key in keys[1:]:
            if not active:
                break
            pods_for_key = key_to_pods.get(key, [])
            current = {p.pod_identifier for p in pods_for_key}
            active &= current
            for pod in active:
                pod_scores[pod] += _get_max_weight(
                    pods_for_key, pod, self.medium_weights
                )
        return pod_scores


def new_kv_block_scorer(config: Optional[KVBlockScorerConfig] = None):
    config = config or KVBlockScorerConfig()
    if config.scoring_strategy == LONGEST_PREFIX_MATCH:
        weights = {b.name: b.weight for b in config.backend_configs}
        return LongestPrefixScorer(weights)
    raise ValueError(f"unsupported scoring strategy: {config.scoring_strategy}")
