"""Config (de)serialization + events-pool GPU burst grouping tests."""

import json
import queue
import time

import pytest

from llmd_kvcache_amd.config import (
    config_from_dict,
    config_from_json,
    config_to_dict,
    config_to_json,
)
from llmd_kvcache_amd.indexer import Config


class TestConfigSerialization:
    def test_defaults_round_trip(self):
        cfg = Config()
        blob = config_to_json(cfg)
        data = json.loads(blob)
        assert data["token_processor"]["block_size"] == 16
        assert data["prefix_store"]["block_size"] == 256
        cfg2 = config_from_json(blob)
        assert cfg2.token_processor.block_size == 16

    def test_partial_overlay_keeps_defaults(self):
        cfg = config_from_dict({"token_processor": {"block_size": 4}})
        assert cfg.token_processor.block_size == 4
        assert cfg.token_processor.hash_seed == ""
        assert cfg.prefix_store.block_size == 256

    def test_backend_configs(self):
        cfg = config_from_dict({
            "backend_configs": [
                {"name": "gpu", "weight": 1.0},
                {"name": "disk", "weight": 0.5},
            ]
        })
        assert cfg.backend_configs[1].name == "disk"
        assert cfg.scorer.backend_configs[1].weight == 0.5

    def test_index_backend_selection(self):
        cfg = config_from_dict({
            "kv_block_index": {"cost_aware": {"max_cost_bytes": 1024}}
        })
        assert cfg.kv_block_index.cost_aware.max_cost_bytes == 1024
        assert cfg.kv_block_index.in_memory is None

    def test_index_defaults_to_in_memory(self):
        cfg = config_from_dict({"kv_block_index": {}})
        assert cfg.kv_block_index.in_memory is not None

    def test_unknown_keys_ignored(self):
        cfg = config_from_dict({"no_such_section": {"x": 1}})
        assert cfg.token_processor.block_size == 16


class FakeGpuIndex:
    """Quacks like GpuIndex for the events-pool burst path."""

    class _Table:
        is_cuda = True

    table = _Table()

    def __init__(self):
        self.applied = []

    def apply_event_batches(self, batches, token_processor=None):
        self.applied.append(list(batches))


class TestEventsPoolBurst:
    def test_burst_groups_messages_into_one_apply(self):
        from llmd_kvcache_amd.kvevents.events import BlockStored, EventBatch
        from llmd_kvcache_amd.kvevents.pool import (
            EventsConfig,
            EventsPool,
            Message,
        )

        index = FakeGpuIndex()
        pool = EventsPool(EventsConfig(concurrency=1), index)
        # enqueue before starting so the first worker drain sees a burst
        for i in range(5):
            batch = EventBatch(
                ts=0.0,
                events=[BlockStored([i + 1], None, list(range(16)), 16)],
            )
            pool.add_task(
                Message(f"kv@pod-{i % 2}@m", batch.encode(), i,
                        f"pod-{i % 2}", "m")
            )
        pool.start(with_subscriber=False)
        pool.drain()
        pool.shutdown()
        total_msgs = sum(len(b) for b in index.applied)
        assert total_msgs == 5
        # burst coalescing: far fewer apply calls than messages
        assert len(index.applied) <= 2

    def test_poison_pill_skipped_in_burst(self):
        from llmd_kvcache_amd.kvevents.pool import (
            EventsConfig,
            EventsPool,
            Message,
        )

        index = FakeGpuIndex()
        pool = EventsPool(EventsConfig(concurrency=1), index)
        pool.add_task(Message("kv@pod-0@m", b"garbage!!", 0, "pod-0", "m"))
        pool.start(with_subscriber=False)
        pool.drain()
        pool.shutdown()
        assert index.applied == []


class TestNestedNoneOverlay:
    def test_none_default_nested_dataclass_constructed(self):
        cfg = config_from_dict({
            "tokenizers_pool": {"uds": {"socket_path": "/tmp/x.sock"}}
        })
        from llmd_kvcache_amd.tokenization.uds import UdsTokenizerConfig

        assert isinstance(cfg.tokenizers_pool.uds, UdsTokenizerConfig)
        assert cfg.tokenizers_pool.uds.socket_path == "/tmp/x.sock"
        assert cfg.tokenizers_pool.uds.timeout_s == 5.0  # default kept


class TestMultiModelBurst:
    def test_mixed_model_burst_single_apply(self):
        """Round 2: the event kernels take a per-event model id
        (model_of), so a mixed-model burst reaches apply_event_batches
        as ONE call - no host-side model split, one launch set."""
        from llmd_kvcache_amd.kvevents.events import BlockStored, EventBatch
        from llmd_kvcache_amd.kvevents.pool import (
            EventsConfig,
            EventsPool,
            Message,
        )

        index = FakeGpuIndex()
        pool = EventsPool(EventsConfig(concurrency=1), index)
        for i, model in enumerate(["model-a", "model-b", "model-a"]):
            batch = EventBatch(
                ts=0.0,
                events=[BlockStored([i + 1], None, list(range(16)), 16)],
            )
            pool.add_task(Message(f"kv@p@{model}", batch.encode(), i, "p",
                                  model))
        pool.start(with_subscriber=False)
        pool.drain()
        pool.shutdown()
        # all three messages (two models) landed in one apply call
        assert len(index.applied) == 1
        assert {m for _, m, _ in index.applied[0]} == {"model-a",
                                                       "model-b"}


class TestTieredConfig:
    def test_tiered_backend_from_dict(self):
        """ADVICE round-1: 'tiered' must be selectable from a dict/JSON
        service config like every other backend."""
        from llmd_kvcache_amd.config import index_config_from_dict

        cfg = index_config_from_dict({
            "tiered": {
                "hot": {"capacity": 1 << 12, "pods_per_key": 4},
                "cold": {"capacity": 1 << 14},
            }
        })
        assert cfg.tiered is not None
        assert cfg.in_memory is None  # no fallback default kicked in
        assert cfg.tiered.hot.capacity == 1 << 12
        assert cfg.tiered.hot.pods_per_key == 4
        assert cfg.tiered.cold.capacity == 1 << 14
        assert cfg.tiered.cold.device == "cpu"

    def test_tiered_empty_uses_defaults(self):
        from llmd_kvcache_amd.config import index_config_from_dict

        cfg = index_config_from_dict({"tiered": {}})
        assert cfg.tiered is not None
        assert cfg.tiered.cold.capacity >= cfg.tiered.hot.capacity
