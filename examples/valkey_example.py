"""Valkey-backed distributed index example (parity with reference
examples/valkey_example/main.go:72-109): the index lives in a
Valkey/Redis server shared by many indexer replicas.

Runs self-contained against the in-process fake server unless
VALKEY_ADDR points at a real one:

    python examples/valkey_example.py
    VALKEY_ADDR=valkey://10.0.0.5:6379 python examples/valkey_example.py
"""

import os
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
from llmd_kvcache_amd.kvblock.redis_index import RedisIndexConfig, ValkeyIndex


def main():
    addr = os.environ.get("VALKEY_ADDR")
    server = None
    if addr is None:
        from llmd_kvcache_amd.kvblock.fake_redis import FakeRedisServer

        server = FakeRedisServer()
        server.start()
        addr = f"valkey://127.0.0.1:{server.port}"
        print(f"using in-process fake valkey at {addr}")

    index = ValkeyIndex(RedisIndexConfig(address=addr))

    keys = [Key("demo-model", h) for h in (11, 12, 13)]
    index.add(keys, keys, [PodEntry("pod-a", "gpu"), PodEntry("pod-b", "cpu")])
    print("lookup all:", index.lookup(keys, set()))
    print("lookup filtered:", index.lookup(keys, {"pod-b"}))
    index.evict(keys[1], [PodEntry("pod-a", "gpu")])
    print("after evict:", index.lookup(keys, set()))
    print("engine->request:", index.get_request_key(keys[0]))

    if server:
        server.stop()
    print("OK")


if __name__ == "__main__":
    main()
