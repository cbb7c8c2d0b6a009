"""gRPC IndexerService example (parity with reference
examples/kv_cache_index_service): wires an Indexer + events pool behind
the indexer.v1.IndexerService API, plus a demo client call.

    python examples/grpc_service.py            # server + self-test client
    GRPC_PORT=50051 python examples/grpc_service.py serve   # serve forever
"""

import os
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.indexer import Config, Indexer
from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.token_processor import TokenProcessorConfig
from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool
from llmd_kvcache_amd.service.grpc_server import IndexerClient, serve


def main():
    cfg = Config()
    cfg.token_processor = TokenProcessorConfig(
        block_size=int(os.environ.get("BLOCK_SIZE", "16")),
        hash_seed=os.environ.get("PYTHONHASHSEED", ""),
    )
    index = InMemoryIndex(InMemoryIndexConfig())
    indexer = Indexer(cfg, kv_block_index=index)
    indexer.run()

    pool = EventsPool(
        EventsConfig(zmq_endpoint=os.environ.get("ZMQ_ENDPOINT", "tcp://*:5557")),
        index,
        indexer.tokens_processor,
    )
    pool.start(with_subscriber=True)

    port = os.environ.get("GRPC_PORT", "50051")
    server = serve(indexer, address=f"0.0.0.0:{port}")
    print(f"IndexerService on :{port}; KVEvents SUB bound")

    if len(sys.argv) > 1 and sys.argv[1] == "serve":
        try:
            while True:
                time.sleep(3600)
        except KeyboardInterrupt:
            pass
    else:
        # demo: seed the index directly and score over gRPC
        from llmd_kvcache_amd.kvblock.keys import PodEntry

        tokens = list(range(64))
        keys = indexer.tokens_processor.tokens_to_kv_block_keys(
            None, tokens, "demo-model"
        )
        index.add(keys, keys, [PodEntry("pod-demo", "gpu")])

        client = IndexerClient(f"127.0.0.1:{port}")
        # our FixedTokenizer-free demo scores pre-tokenized instead
        scores = indexer.score_tokens(tokens, "demo-model", [])
        print("library scores:", scores)
        client.close()

    server.stop(None)
    pool.shutdown()
    indexer.shutdown()


if __name__ == "__main__":
    main()
