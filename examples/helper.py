"""Test/demo publisher: frames KV event batches the way a vLLM pod does.

Parity with reference examples/helper/publisher.go:59-83 (3-part ZMQ
message ``[topic, seq BE-u64, msgpack payload]``, topic
``kv@<pod-id>@<model>``) and helper/events.go:32-92 (synthetic
BlockStored/BlockRemoved streams).
"""

from __future__ import annotations

import random
import struct
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.kvevents.events import BlockRemoved, BlockStored, EventBatch
from llmd_kvcache_amd.kvevents.zmtp import PubSocket


class Publisher:
    def __init__(self, endpoint: str):
        self.pub = PubSocket()
        self.pub.connect(endpoint)
        self.seq = 0

    def wait_ready(self, timeout: float = 10.0) -> bool:
        return self.pub.wait_for_subscriber(timeout)

    def publish(self, pod_id: str, model: str, batch: EventBatch) -> None:
        topic = f"kv@{pod_id}@{model}".encode()
        self.pub.send_multipart(
            [topic, struct.pack(">Q", self.seq), batch.encode()]
        )
        self.seq += 1

    def close(self) -> None:
        self.pub.close()


def simulate_events(
    publisher: Publisher,
    pod_id: str,
    model: str,
    n_batches: int = 10,
    blocks_per_batch: int = 4,
    block_size: int = 16,
    seed: int = 0,
    remove_ratio: float = 0.2,
):
    """Emits chained BlockStored batches with occasional removals."""
    rng = random.Random(seed)
    next_hash = rng.randrange(1, 1 << 32)
    parent = None
    stored = []
    for _ in range(n_batches):
        if stored and rng.random() < remove_ratio:
            victim = stored.pop(rng.randrange(len(stored)))
            batch = EventBatch(ts=time.time(), events=[BlockRemoved([victim])])
        else:
            hashes = list(range(next_hash, next_hash + blocks_per_batch))
            next_hash += blocks_per_batch
            tokens = [rng.randrange(0, 1 << 31)
                      for _ in range(blocks_per_batch * block_size)]
            batch = EventBatch(
                ts=time.time(),
                events=[BlockStored(hashes, parent, tokens, block_size)],
            )
            parent = hashes[-1]
            stored.extend(hashes)
        publisher.publish(pod_id, model, batch)
