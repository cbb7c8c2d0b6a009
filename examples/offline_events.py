"""Offline KV-events example (BASELINE.json config 1).

Parity with reference examples/kv_events/offline/main.go: a dummy ZMQ
publisher and an in-memory index wired through the full events pool, one
pod, CPU-only - proves the plumbing with zero GPU code.

    python examples/offline_events.py
"""

import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.token_processor import (
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool
from llmd_kvcache_amd.scorer import new_kv_block_scorer

from helper import Publisher, simulate_events


def main():
    block_size = 16
    index = InMemoryIndex(InMemoryIndexConfig(size=100_000, pod_cache_size=10))
    tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=block_size))
    pool = EventsPool(
        EventsConfig(zmq_endpoint="tcp://127.0.0.1:0", concurrency=4),
        index,
        tp,
    )
    pool.start(with_subscriber=True)
    # ephemeral port: wait for the SUB to bind and read the port back
    # (a fixed port collides with concurrent runs/tests)
    deadline = time.monotonic() + 10
    while pool._subscriber.port is None and time.monotonic() < deadline:
        time.sleep(0.05)
    assert pool._subscriber.port, "subscriber did not bind"

    pub = Publisher(f"tcp://127.0.0.1:{pool._subscriber.port}")
    assert pub.wait_ready(), "subscriber did not come up"
    print("publishing synthetic vLLM KV events for pod-0 ...")
    simulate_events(pub, "pod-0", "demo-model", n_batches=20,
                    block_size=block_size, seed=7)
    time.sleep(1.0)
    pool.drain()

    # score a prompt whose prefix the events stored
    import random

    rng = random.Random(7)
    rng.randrange(1, 1 << 32)  # consume the hash draw like the simulator
    tokens = [rng.randrange(0, 1 << 31) for _ in range(4 * block_size)]
    keys = tp.tokens_to_kv_block_keys(None, tokens, "demo-model")
    scorer = new_kv_block_scorer()
    scores = scorer.score(keys, index.lookup(keys, set()))
    print("pod scores for the first stored prefix:", scores)
    assert scores.get("pod-0", 0) > 0, "expected a hit on pod-0"
    print("OK")

    pub.close()
    pool.shutdown()


if __name__ == "__main__":
    main()
