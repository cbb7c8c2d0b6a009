"""Scheduler-scorer plugin sketch.

Parity with the reference's EPP plugin sketch
(examples/kv_cache_aware_scorer/kvcache_aware_scorer.go - excluded from
its build too): the shape a llm-d inference-scheduler scorer plugin takes
when it embeds this indexer directly.  The scheduler calls
``score(request, pods)`` per routing decision; scores are normalized to
[0, 1] (max consecutive hits wins).

This file is a reference sketch, not wired into any scheduler here.
"""

import os
import sys
from typing import Dict, Sequence

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.indexer import Config, Indexer
from llmd_kvcache_amd.kvblock.index import IndexConfig


class KVCacheAwareScorer:
    """plugins.Scorer-shaped adapter over the Indexer."""

    NAME = "kvcache-aware-scorer"

    def __init__(self):
        config = Config()
        redis_addr = os.environ.get("KVCACHE_INDEXER_REDIS_ADDR")
        if redis_addr:
            from llmd_kvcache_amd.kvblock.redis_index import RedisIndexConfig

            config.kv_block_index = IndexConfig(
                redis=RedisIndexConfig(address=redis_addr)
            )
        elif os.environ.get("KVCACHE_INDEX_BACKEND") == "gpu":
            from llmd_kvcache_amd.kvblock.gpu_index import GpuIndexConfig

            config.kv_block_index = IndexConfig(gpu=GpuIndexConfig())
        self.indexer = Indexer(config)
        self.indexer.run()

    def score(
        self, prompt: str, model_name: str, pods: Sequence[str]
    ) -> Dict[str, float]:
        """Returns normalized [0,1] scores for the candidate pods (the
        reference normalizes by the max hit count)."""
        raw = self.indexer.get_pod_scores(None, prompt, model_name, pods)
        if not raw:
            return {p: 0.0 for p in pods}
        top = max(raw.values()) or 1.0
        return {p: raw.get(p, 0.0) / top for p in pods}


if __name__ == "__main__":
    scorer = KVCacheAwareScorer()
    print(f"{scorer.NAME} ready (backend: "
          f"{type(scorer.indexer.kv_block_index()).__name__})")
