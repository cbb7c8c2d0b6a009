"""Backend comparison on one identical workload (CPU-host backends).

    python scripts/compare_backends.py
"""
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.cost_aware import (
    CostAwareMemoryIndex,
    CostAwareMemoryIndexConfig,
)
from llmd_kvcache_amd.kvblock.fake_redis import FakeRedisServer
from llmd_kvcache_amd.kvblock.gpu_index import NativeIndex, TableIndexConfig
from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
from llmd_kvcache_amd.kvblock.redis_index import RedisIndex, RedisIndexConfig

MODEL = "m"
N_KEYS = 20_000
LOOKUPS = 2_000
KEYS_PER_LOOKUP = 64


def run(index, label):
    entries = [PodEntry("pod-a", "gpu")]
    t0 = time.perf_counter()
    for lo in range(0, N_KEYS, 64):
        ks = [Key(MODEL, 1 + lo + i) for i in range(64)]
        index.add(ks, ks, entries)
    t_add = time.perf_counter() - t0

    t0 = time.perf_counter()
    found = 0
    for q in range(LOOKUPS):
        base = 1 + (q * 37) % (N_KEYS - KEYS_PER_LOOKUP)
        ks = [Key(MODEL, base + i) for i in range(KEYS_PER_LOOKUP)]
        found += len(index.lookup(ks, set()))
    t_lookup = time.perf_counter() - t0

    print(f"{label:<28} add {N_KEYS / t_add:>12,.0f} keys/s   "
          f"lookup {LOOKUPS * KEYS_PER_LOOKUP / t_lookup:>12,.0f} keys/s   "
          f"(hits {found})", flush=True)


def main():
    run(InMemoryIndex(InMemoryIndexConfig(size=10**6, pod_cache_size=10)),
        "in_memory (pure Python)")
    run(NativeIndex(TableIndexConfig(capacity=1 << 16, pods_per_key=10)),
        "native (C++ table, CPU)")
    run(CostAwareMemoryIndex(CostAwareMemoryIndexConfig()),
        "cost_aware")
    server = FakeRedisServer()
    server.start()
    try:
        run(RedisIndex(RedisIndexConfig(
            address=f"redis://127.0.0.1:{server.port}")),
            "redis (loopback RESP)")
    finally:
        server.stop()


if __name__ == "__main__":
    main()
