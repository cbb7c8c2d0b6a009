"""Online KV-cache indexer service (the shipped-container-binary parity).

Parity with reference examples/kv_events/online/main.go: HTTP scoring
endpoints + Prometheus /metrics + the ZMQ events pool, configured via the
same environment variables (main.go:41-58,167-225):

  HTTP_PORT              (default 8080)
  SNAPSHOT_PATH          optional: load the index snapshot at boot if the
                         file exists; save on SIGTERM (table backends)
  ZMQ_ENDPOINT           (default tcp://*:5557)
  ZMQ_TOPIC              (default kv@)
  POOL_CONCURRENCY       (default 4)
  BLOCK_SIZE             (default 16)
  PYTHONHASHSEED         (hash seed aligned with the vLLM fleet)
  KVCACHE_INDEX_BACKEND  in_memory | native | gpu | cost_aware | redis |
                         valkey  (MI355X-native addition; default
                         in_memory, "gpu" puts the index in HBM)
  REDIS_ADDR             redis://host:port for the redis/valkey backends

    python examples/online_service.py
"""

import logging
import os
import signal
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.indexer import Config, Indexer
from llmd_kvcache_amd.kvblock.index import IndexConfig, new_index
from llmd_kvcache_amd.kvblock.token_processor import (
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool
from llmd_kvcache_amd.metrics import collector
from llmd_kvcache_amd.service.http_server import HttpService

logging.basicConfig(level=os.environ.get("LOG_LEVEL", "INFO"))
logger = logging.getLogger("online_service")


def build_index_config() -> IndexConfig:
    backend = os.environ.get("KVCACHE_INDEX_BACKEND", "in_memory")
    cfg = IndexConfig(enable_metrics=True, metrics_logging_interval_s=60.0)
    if backend == "in_memory":
        from llmd_kvcache_amd.kvblock.in_memory import InMemoryIndexConfig

        cfg.in_memory = InMemoryIndexConfig()
    elif backend == "native":
        from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig

        cfg.native = TableIndexConfig()
    elif backend == "gpu":
        from llmd_kvcache_amd.kvblock.gpu_index import GpuIndexConfig

        cfg.gpu = GpuIndexConfig()
    elif backend == "cost_aware":
        from llmd_kvcache_amd.kvblock.cost_aware import (
            CostAwareMemoryIndexConfig,
        )

        cfg.cost_aware = CostAwareMemoryIndexConfig()
    elif backend in ("redis", "valkey"):
        from llmd_kvcache_amd.kvblock.redis_index import RedisIndexConfig

        rcfg = RedisIndexConfig(
            address=os.environ.get("REDIS_ADDR", "redis://127.0.0.1:6379")
        )
        if backend == "valkey":
            cfg.valkey = rcfg
        else:
            cfg.redis = rcfg
    else:
        raise SystemExit(f"unknown KVCACHE_INDEX_BACKEND {backend!r}")
    return cfg


def main():
    block_size = int(os.environ.get("BLOCK_SIZE", "16"))
    hash_seed = os.environ.get("PYTHONHASHSEED", "")

    config = Config()
    config.token_processor = TokenProcessorConfig(
        block_size=block_size, hash_seed=hash_seed
    )
    config.kv_block_index = build_index_config()

    collector.register()
    index = new_index(config.kv_block_index)
    snapshot = os.environ.get("SNAPSHOT_PATH")
    if snapshot and os.path.exists(snapshot) and hasattr(index, "load"):
        try:
            index.load(snapshot)
            logger.info("loaded index snapshot from %s", snapshot)
        except Exception:
            logger.exception("snapshot load failed; starting empty")
    indexer = Indexer(config, kv_block_index=index)
    indexer.run()
    logger.info("indexer running (block_size=%d)", block_size)

    events_cfg = EventsConfig(
        zmq_endpoint=os.environ.get("ZMQ_ENDPOINT", "tcp://*:5557"),
        topic_filter=os.environ.get("ZMQ_TOPIC", "kv@"),
        concurrency=int(os.environ.get("POOL_CONCURRENCY", "4")),
    )
    pool = EventsPool(
        events_cfg,
        index,
        ChunkedTokenDatabase(config.token_processor),
    )
    pool.start(with_subscriber=True)
    logger.info("events pool running on %s", events_cfg.zmq_endpoint)

    http = HttpService(
        indexer,
        host="0.0.0.0",
        port=int(os.environ.get("HTTP_PORT", "8080")),
    )
    http.start()
    logger.info("HTTP service on :%d - POST /score_completions, "
                "/score_chat_completions, GET /metrics", http.port)

    stop = []
    signal.signal(signal.SIGTERM, lambda *a: stop.append(1))
    signal.signal(signal.SIGINT, lambda *a: stop.append(1))
    try:
        while not stop:
            signal.pause()
    finally:
        logger.info("shutting down")
        if snapshot and hasattr(index, "save"):
            try:
                index.save(snapshot)
                logger.info("saved index snapshot to %s", snapshot)
            except Exception:
                logger.exception("snapshot save failed")
        http.stop()
        pool.shutdown()
        indexer.shutdown()


if __name__ == "__main__":
    main()
