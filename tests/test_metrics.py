"""Metrics subsystem tests: collector registration idempotency, the
instrumented-index decorator counters, and the periodic metrics beat
(mirrors pkg/kvcache/metrics/collector_test.go and
instrumented_index.go)."""

import time

import pytest

from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.instrumented import InstrumentedIndex
from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
from llmd_kvcache_amd.metrics import collector


@pytest.fixture(autouse=True)
def registered():
    collector.register()
    yield


def k(h):
    return Key("m", h)


class TestCollector:
    def test_register_idempotent(self):
        collector.register()
        collector.register()  # second call must not raise duplicate error
        assert collector.admissions is not None

    def test_observe_helpers(self):
        collector.observe_tokenization("hf", 0.01, 42)
        collector.observe_render_latency(0.005)

    def test_metrics_beat(self):
        collector.start_metrics_logging(0.05)
        time.sleep(0.15)
        collector.stop_metrics_logging()


class TestInstrumentedIndex:
    def make(self):
        inner = InMemoryIndex(InMemoryIndexConfig(size=1000, pod_cache_size=5))
        return InstrumentedIndex(inner), inner

    def test_add_lookup_evict_counters_move(self):
        idx, _ = self.make()
        a0 = collector.admissions._value.get()
        l0 = collector.lookup_requests._value.get()
        e0 = collector.evictions._value.get()

        keys = [k(1), k(2)]
        idx.add(keys, keys, [PodEntry("pod-a", "gpu")])
        assert collector.admissions._value.get() == a0 + 2

        result = idx.lookup(keys, set())
        assert len(result) == 2
        assert collector.lookup_requests._value.get() == l0 + 1

        idx.evict(k(1), [PodEntry("pod-a", "gpu")])
        assert collector.evictions._value.get() == e0 + 1

    def test_delegates_get_request_key(self):
        idx, _ = self.make()
        keys = [k(10)]
        idx.add(keys, [k(11)], [PodEntry("pod-a", "gpu")])
        assert idx.get_request_key(k(10)) == k(11)

    def test_passthrough_attributes(self):
        idx, inner = self.make()
        # decorator exposes the inner backend's extra attributes
        assert idx._pod_cache_size == inner._pod_cache_size


class TestInstrumentedOverTableIndex:
    def test_decorator_transparent_for_fast_paths(self):
        """InstrumentedIndex must forward the table-backend fast-path
        surface (table, fused_scores, apply_event_batches detection) so
        the events pool and Indexer keep their native paths."""
        import pytest

        torch = pytest.importorskip("torch")
        from llmd_kvcache_amd.kvblock.gpu_index import (
            NativeIndex,
            TableIndexConfig,
        )
        from llmd_kvcache_amd.ops import cpu_ext

        if cpu_ext.maybe_load() is None:
            pytest.skip("native extension not built")
        inner = NativeIndex(TableIndexConfig(capacity=1 << 10, pods_per_key=4))
        idx = InstrumentedIndex(inner)
        # counters move through the decorator
        a0 = collector.admissions._value.get()
        keys = [k(100), k(101)]
        idx.add(keys, keys, [PodEntry("pod-a", "gpu")])
        assert collector.admissions._value.get() == a0 + 2
        # fast-path surface forwarded
        assert idx.table is inner.table
        assert idx.tier_weights() is not None
        from llmd_kvcache_amd.kvblock.gpu_index import _to_i64

        hashes = torch.tensor([_to_i64(x.chunk_hash) for x in keys],
                              dtype=torch.int64)
        counts = torch.tensor([2], dtype=torch.int32)
        maps = idx.scores_to_map(idx.fused_scores(hashes, counts, "m", set()))
        assert maps[0] == {"pod-a": 2.0}


class TestTraceLogging:
    def test_trace_level_and_helpers(self):
        import logging

        from llmd_kvcache_amd.utils.logging import (
            TRACE,
            enable_trace,
            get_logger,
            trace,
        )

        assert logging.getLevelName(TRACE) == "TRACE"
        log = get_logger("test")
        records = []

        class Capture(logging.Handler):
            def emit(self, record):
                records.append(record.getMessage())

        h = Capture()
        log.addHandler(h)
        try:
            log.setLevel(TRACE + 1)
            trace(log, "hidden %d", 1)
            assert records == []
            enable_trace()
            log.setLevel(TRACE)
            trace(log, "visible %d", 2)
            assert records == ["visible 2"]
        finally:
            log.removeHandler(h)
