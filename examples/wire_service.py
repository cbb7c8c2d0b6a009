"""Online indexer with the NATIVE wire front (C++ epoll HTTP server).

Same role as examples/online_service.py (the shipped-binary parity,
reference examples/kv_events/online/main.go) but the scoring surface is
the wirefront: requests parse and batch in C++, warm session prompts
never touch Python, and micro-batches run one fused kernel each
(docs/architecture.md "Wire front").  The ZMQ KVEvents pool is identical.

Environment:
  WIRE_PORT              (default 8080) scoring endpoint port
  WIRE_IO_THREADS        (default 4)
  METRICS_PORT           optional: also serve the full HTTP surface
                         (/metrics, /score_batch, chat endpoints) on
                         this port via the stdlib service
  ZMQ_ENDPOINT           (default tcp://*:5557)
  ZMQ_TOPIC              (default kv@)
  POOL_CONCURRENCY       (default 4)
  BLOCK_SIZE             (default 16)
  PYTHONHASHSEED         hash seed aligned with the vLLM fleet
  KVCACHE_INDEX_BACKEND  native | gpu | tiered  (table backends only -
                         the wirefront scores through the fused path;
                         default native, "gpu" puts the index in HBM)

Endpoints (HTTP/1.1 keep-alive + pipelining):
  POST /score   {"model": m, "prompt": "..."} or {"model": m,
                 "tokens": [...]}, optional {"pods": [...]}
  GET  /health

    python examples/wire_service.py
"""

import logging
import os
import signal
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.indexer import Config, Indexer
from llmd_kvcache_amd.kvblock.index import IndexConfig, new_index
from llmd_kvcache_amd.kvblock.token_processor import TokenProcessorConfig
from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool
from llmd_kvcache_amd.service.wirefront import WireIndexerService

logging.basicConfig(level=os.environ.get("LOG_LEVEL", "INFO"))
logger = logging.getLogger("wire_service")


def build_index_config() -> IndexConfig:
    backend = os.environ.get("KVCACHE_INDEX_BACKEND", "native")
    cfg = IndexConfig()
    if backend == "gpu":
        from llmd_kvcache_amd.kvblock.gpu_index import GpuIndexConfig

        cfg.gpu = GpuIndexConfig()
    elif backend == "tiered":
        from llmd_kvcache_amd.kvblock.tiered import TieredIndexConfig

        cfg.tiered = TieredIndexConfig()
    else:
        from llmd_kvcache_amd.kvblock.gpu_index import TableIndexConfig

        cfg.native = TableIndexConfig()
    return cfg


def main() -> None:
    config = Config(
        token_processor=TokenProcessorConfig(
            block_size=int(os.environ.get("BLOCK_SIZE", "16")),
            hash_seed=os.environ.get("PYTHONHASHSEED", ""),
        ),
        kv_block_index=build_index_config(),
    )
    indexer = Indexer(config, kv_block_index=new_index(config.kv_block_index))
    indexer.run()

    events = EventsPool(
        EventsConfig(
            zmq_endpoint=os.environ.get("ZMQ_ENDPOINT", "tcp://*:5557"),
            topic_filter=os.environ.get("ZMQ_TOPIC", "kv@"),
            concurrency=int(os.environ.get("POOL_CONCURRENCY", "4")),
        ),
        indexer.kv_block_index(),
        indexer.tokens_processor,
    )
    events.start()

    svc = WireIndexerService(indexer)
    port = svc.start(host="0.0.0.0",
                     port=int(os.environ.get("WIRE_PORT", "8080")),
                     n_io=int(os.environ.get("WIRE_IO_THREADS", "4")))
    logger.info("wirefront serving on :%d", port)

    http = None
    metrics_port = int(os.environ.get("METRICS_PORT", "0"))
    if metrics_port:
        from llmd_kvcache_amd.service.http_server import HttpService

        http = HttpService(indexer, host="0.0.0.0", port=metrics_port)
        http.start()
        logger.info("metrics/full HTTP surface on :%d", http.port)

    stop = {"flag": False}

    def on_term(signum, frame):
        stop["flag"] = True

    signal.signal(signal.SIGTERM, on_term)
    signal.signal(signal.SIGINT, on_term)
    try:
        while not stop["flag"]:
            time.sleep(0.5)
    finally:
        svc.stop()
        if http is not None:
            http.stop()
        events.shutdown()
        indexer.shutdown()
        reqs, batches = svc.stats()
        logger.info("served %d requests in %d batches", reqs, batches)


if __name__ == "__main__":
    main()
