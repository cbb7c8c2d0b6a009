"""Library-usage example (parity with reference
examples/kv_cache_index/main.go:76-139): construct the Indexer directly,
choose the index backend from env (Redis via REDIS_ADDR like the
reference, or the MI355X GPU table), feed it a tokenization + events, and
score.

    python examples/kv_cache_index.py
    REDIS_ADDR=redis://host:6379 python examples/kv_cache_index.py
    KVCACHE_INDEX_BACKEND=gpu python examples/kv_cache_index.py   # on MI355X
"""

import os
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.indexer import Config, Indexer
from llmd_kvcache_amd.kvblock.index import IndexConfig, new_index
from llmd_kvcache_amd.kvblock.keys import PodEntry
from llmd_kvcache_amd.kvblock.token_processor import TokenProcessorConfig
from llmd_kvcache_amd.kvevents.events import BlockStored
from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool


def pick_index():
    backend = os.environ.get("KVCACHE_INDEX_BACKEND")
    cfg = IndexConfig()
    if os.environ.get("REDIS_ADDR"):
        from llmd_kvcache_amd.kvblock.redis_index import RedisIndexConfig

        cfg.redis = RedisIndexConfig(address=os.environ["REDIS_ADDR"])
    elif backend == "gpu":
        from llmd_kvcache_amd.kvblock.gpu_index import GpuIndexConfig

        cfg.gpu = GpuIndexConfig()
    elif backend == "native":
        from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig

        cfg.native = TableIndexConfig()
    else:
        from llmd_kvcache_amd.kvblock.in_memory import InMemoryIndexConfig

        cfg.in_memory = InMemoryIndexConfig()
    return new_index(cfg)


def main():
    config = Config()
    config.token_processor = TokenProcessorConfig(
        block_size=int(os.environ.get("BLOCK_SIZE", "16")),
        hash_seed=os.environ.get("PYTHONHASHSEED", ""),
    )
    index = pick_index()
    indexer = Indexer(config, kv_block_index=index)
    indexer.run()
    print(f"indexer up with backend {type(index).__name__}")

    # simulate a vLLM pod storing a 64-token prompt through the events path
    pool = EventsPool(EventsConfig(), index, indexer.tokens_processor)
    tokens = list(range(64))
    pool.digest_events(
        "vllm-pod-7", "demo-model",
        [BlockStored([1, 2, 3, 4], None, tokens, 16)],
    )

    scores = indexer.score_tokens(tokens, "demo-model", [])
    print("pod scores:", scores)
    assert scores == {"vllm-pod-7": 4.0}, scores
    print("OK")
    indexer.shutdown()


if __name__ == "__main__":
    main()
