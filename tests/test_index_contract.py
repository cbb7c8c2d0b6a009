"""Shared behavioral contract suite for every Index backend.

Mirrors the reference's factory-injected common suite
(pkg/kvcache/kvblock/index_test.go:35-63): BasicAddAndLookup,
DuplicatePodHandling, FilteredLookup, EvictBasic, ConcurrentOperations -
run against each backend via a pytest fixture parameterized by factory.
"""

import threading

import pytest

from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.keys import Key, PodEntry


def _in_memory_factory():
    return InMemoryIndex(InMemoryIndexConfig(size=10_000, pod_cache_size=10))


def _cost_aware_factory():
    from llmd_kvcache_amd.kvblock.cost_aware import (
        CostAwareMemoryIndex,
        CostAwareMemoryIndexConfig,
    )

    return CostAwareMemoryIndex(
        CostAwareMemoryIndexConfig(max_cost_bytes=64 * 1024 * 1024)
    )


def _fake_redis_factory():
    from llmd_kvcache_amd.kvblock.redis_index import RedisIndex, RedisIndexConfig
    from llmd_kvcache_amd.kvblock.fake_redis import FakeRedisServer

    server = FakeRedisServer()
    server.start()
    return RedisIndex(
        RedisIndexConfig(address=f"redis://127.0.0.1:{server.port}")
    )


def _native_factory():
    from llmd_kvcache_amd.kvblock.gpu_index import NativeIndex, TableIndexConfig

    return NativeIndex(TableIndexConfig(capacity=1 << 14, pods_per_key=10))


FACTORIES = {
    "in_memory": _in_memory_factory,
    "native": _native_factory,
    "cost_aware": _cost_aware_factory,
    "redis": _fake_redis_factory,
}


@pytest.fixture(params=list(FACTORIES.keys()))
def index(request):
    try:
        yield FACTORIES[request.param]()
    except ImportError:
        pytest.skip(f"backend {request.param} not available yet")


MODEL = "test-model"


def k(h):
    return Key(MODEL, h)


def pe(pod, tier="gpu"):
    return PodEntry(pod, tier)


class TestIndexContract:
    def test_basic_add_and_lookup(self, index):
        keys = [k(1), k(2), k(3)]
        index.add(keys, keys, [pe("pod-a")])
        result = index.lookup(keys, set())
        assert set(result.keys()) == set(keys)
        for key in keys:
            assert result[key] == [pe("pod-a")]

    def test_lookup_empty_keys_raises(self, index):
        with pytest.raises(ValueError):
            index.lookup([], set())

    def test_duplicate_pod_handling(self, index):
        keys = [k(10)]
        index.add(keys, keys, [pe("pod-a")])
        index.add(keys, keys, [pe("pod-a")])
        result = index.lookup(keys, set())
        assert result[k(10)] == [pe("pod-a")]

    def test_multiple_pods_and_tiers(self, index):
        keys = [k(20)]
        index.add(keys, keys, [pe("pod-a", "gpu"), pe("pod-b", "cpu")])
        result = index.lookup(keys, set())
        assert sorted(result[k(20)]) == sorted(
            [pe("pod-a", "gpu"), pe("pod-b", "cpu")]
        )

    def test_filtered_lookup(self, index):
        keys = [k(30)]
        index.add(keys, keys, [pe("pod-a"), pe("pod-b"), pe("pod-c")])
        result = index.lookup(keys, {"pod-b"})
        assert result[k(30)] == [pe("pod-b")]
        # filter with no matching pod: key yields no entries
        result = index.lookup(keys, {"pod-zzz"})
        assert k(30) not in result or result[k(30)] == []

    def test_evict_basic(self, index):
        keys = [k(40)]
        index.add(keys, keys, [pe("pod-a"), pe("pod-b")])
        index.evict(k(40), [pe("pod-a")])
        result = index.lookup(keys, set())
        assert result.get(k(40), []) == [pe("pod-b")]

    def test_evict_last_pod_removes_key(self, index):
        keys = [k(50)]
        index.add(keys, keys, [pe("pod-a")])
        index.evict(k(50), [pe("pod-a")])
        result = index.lookup([k(50), k(51)], set())
        assert k(50) not in result

    def test_evict_unknown_key_noop(self, index):
        index.evict(k(999), [pe("pod-a")])  # must not raise

    def test_evict_empty_entries_raises(self, index):
        with pytest.raises(ValueError):
            index.evict(k(1), [])

    def test_get_request_key_mapping(self, index):
        engine_keys = [k(60)]
        request_keys = [k(61)]
        index.add(engine_keys, request_keys, [pe("pod-a")])
        assert index.get_request_key(k(60)) == k(61)
        assert index.get_request_key(k(12345)) is None

    def test_dual_key_eviction_goes_through_engine_key(self, index):
        engine_keys = [k(70)]
        request_keys = [k(71)]
        index.add(engine_keys, request_keys, [pe("pod-a")])
        # data is stored under the request key
        assert index.lookup([k(71)], set())[k(71)] == [pe("pod-a")]
        # eviction addresses the engine key
        index.evict(k(70), [pe("pod-a")])
        assert k(71) not in index.lookup([k(71), k(72)], set())

    def test_lookup_early_stop_on_chain_break(self, index):
        # keys 1,2 present; 3 absent; 4 present -> early-stop semantics:
        # absent keys are skipped but present keys after a gap still count
        # (reference in_memory.go:105-146: only an empty-present key cuts).
        keys = [k(80), k(81), k(83)]
        index.add(keys, keys, [pe("pod-a")])
        lookup_keys = [k(80), k(81), k(82), k(83)]
        result = index.lookup(lookup_keys, set())
        assert k(80) in result and k(81) in result and k(83) in result

    def test_add_mismatched_key_lengths_raises(self, index):
        with pytest.raises(ValueError):
            index.add([k(1), k(2)], [k(1)], [pe("pod-a")])

    def test_add_empty_raises(self, index):
        with pytest.raises(ValueError):
            index.add([], [], [])

    def test_concurrent_operations(self, index):
        errors = []

        def writer(base):
            try:
                for i in range(50):
                    keys = [k(base + i)]
                    index.add(keys, keys, [pe(f"pod-{base}")])
            except Exception as e:  # pragma: no cover
                errors.append(e)

        def reader():
            try:
                for i in range(50):
                    index.lookup([k(1000 + i)], set())
            except Exception as e:  # pragma: no cover
                errors.append(e)

        threads = [
            threading.Thread(target=writer, args=(1000,)),
            threading.Thread(target=writer, args=(2000,)),
            threading.Thread(target=reader),
            threading.Thread(target=reader),
        ]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert not errors
        result = index.lookup([k(1000)], set())
        assert result[k(1000)] == [pe("pod-1000")]


class TestCostAwareEngineMapBounds:
    """ADVICE round-1 (medium): the engine->request map must be bounded
    and budget eviction must clean up mappings for dropped request keys
    (reference cost_aware_memory.go caps the mapping with an LRU)."""

    def _make(self, budget=4096, emap=8):
        from llmd_kvcache_amd.kvblock.cost_aware import (
            CostAwareMemoryIndex,
            CostAwareMemoryIndexConfig,
        )

        return CostAwareMemoryIndex(
            CostAwareMemoryIndexConfig(max_cost_bytes=budget,
                                       engine_map_size=emap)
        )

    def test_budget_eviction_cleans_engine_mappings(self):
        idx = self._make(budget=600)  # fits only a few keys
        entries = [PodEntry("pod-a", "gpu")]
        for i in range(50):
            ek, rk = Key("m", 10_000 + i), Key("m", 20_000 + i)
            idx.add([ek], [rk], entries)
        # every request key evicted by budget lost its engine mapping too
        live_requests = set(idx._data.keys())
        for ek, rk in idx._engine_to_request.items():
            assert rk in live_requests
        assert len(idx._engine_to_request) <= len(live_requests)
        # evicted engine keys resolve to None now
        assert idx.get_request_key(Key("m", 10_000)) is None

    def test_engine_map_capacity_bounded(self):
        idx = self._make(budget=1 << 30, emap=8)
        entries = [PodEntry("pod-a", "gpu")]
        for i in range(100):
            idx.add([Key("m", 1000 + i)], [Key("m", 5000 + i)], entries)
        assert len(idx._engine_to_request) <= 8
        # newest mappings survive, oldest were LRU-evicted
        assert idx.get_request_key(Key("m", 1099)) == Key("m", 5099)
        assert idx.get_request_key(Key("m", 1000)) is None

    def test_evict_to_empty_drops_all_aliases(self):
        idx = self._make()
        entries = [PodEntry("pod-a", "gpu")]
        rk = Key("m", 7777)
        eks = [Key("m", 100), Key("m", 101)]
        idx.add(eks, [rk, rk], entries)
        idx.evict(eks[0], entries)  # removes the only pod -> key dropped
        assert idx.get_request_key(eks[0]) is None
        assert idx.get_request_key(eks[1]) is None
        assert idx._request_to_engines == {}
