"""Write-path tests: msgpack wire format, hash coercion, sharded pool
semantics (mirrors pkg/kvcache/kvevents behavior)."""

import struct

import msgpack
import pytest

from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
from llmd_kvcache_amd.kvblock.token_processor import (
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from llmd_kvcache_amd.kvevents.events import (
    AllBlocksCleared,
    BlockRemoved,
    BlockStored,
    DecodeError,
    EventBatch,
    decode_event_batch,
    get_hash_as_uint64,
)
from llmd_kvcache_amd.kvevents.pool import EventsConfig, EventsPool, Message


class TestWireFormat:
    def test_roundtrip_block_stored(self):
        batch = EventBatch(
            ts=123.5,
            events=[
                BlockStored(
                    block_hashes=[1, 2],
                    parent_block_hash=None,
                    token_ids=list(range(32)),
                    block_size=16,
                )
            ],
        )
        decoded = decode_event_batch(batch.encode())
        assert decoded.ts == 123.5
        assert len(decoded.events) == 1
        ev = decoded.events[0]
        assert isinstance(ev, BlockStored)
        assert ev.block_hashes == [1, 2]
        assert ev.parent_block_hash is None
        assert ev.token_ids == list(range(32))
        assert ev.block_size == 16

    def test_batch_is_msgpack_array_with_raw_events(self):
        """The wire layout must be [ts, [raw msgpack events], dp_rank?] with
        events encoded as nested msgpack byte strings (events.go:38-43)."""
        batch = EventBatch(ts=1.0, events=[AllBlocksCleared()])
        raw = msgpack.unpackb(batch.encode(), raw=False)
        assert isinstance(raw, list) and len(raw) == 2
        assert raw[0] == 1.0
        inner = msgpack.unpackb(raw[1][0], raw=False)
        assert inner == ["AllBlocksCleared"]

    def test_data_parallel_rank(self):
        batch = EventBatch(ts=1.0, events=[AllBlocksCleared()], data_parallel_rank=3)
        decoded = decode_event_batch(batch.encode())
        assert decoded.data_parallel_rank == 3

    def test_tagged_union_order_matches_reference(self):
        ev = BlockStored([1], 5, [1, 2], 2, 7, "CPU")
        assert ev.to_tagged_union() == ["BlockStored", [1], 5, [1, 2], 2, 7, "CPU"]
        ev2 = BlockRemoved([9], "gpu")
        assert ev2.to_tagged_union() == ["BlockRemoved", [9], "gpu"]

    def test_unknown_tag_skipped(self):
        raw_ev = msgpack.packb(["FancyNewEvent", 1, 2], use_bin_type=True)
        payload = msgpack.packb([1.0, [raw_ev]], use_bin_type=True)
        decoded = decode_event_batch(payload)
        assert decoded.events == []

    def test_poison_pill_raises_decode_error(self):
        with pytest.raises(DecodeError):
            decode_event_batch(b"\x00\x01garbage")

    def test_malformed_event_in_batch_skipped(self):
        good = msgpack.packb(["AllBlocksCleared"], use_bin_type=True)
        bad = b"\xc1\xc1"  # invalid msgpack
        payload = msgpack.packb([1.0, [bad, good]], use_bin_type=True)
        decoded = decode_event_batch(payload)
        assert len(decoded.events) == 1

    def test_bytes_hashes(self):
        h = struct.pack(">Q", 0xDEADBEEFCAFEBABE)
        batch = EventBatch(
            ts=0.0,
            events=[BlockStored([h], None, list(range(16)), 16)],
        )
        decoded = decode_event_batch(batch.encode())
        assert get_hash_as_uint64(decoded.events[0].block_hashes[0]) == 0xDEADBEEFCAFEBABE


class TestHashCoercion:
    def test_uint64_passthrough(self):
        assert get_hash_as_uint64(42) == 42

    def test_negative_int64_wraps(self):
        assert get_hash_as_uint64(-1) == 0xFFFFFFFFFFFFFFFF

    def test_long_bytes_takes_last_8_big_endian(self):
        b = bytes(range(1, 13))  # 12 bytes
        assert get_hash_as_uint64(b) == int.from_bytes(b[-8:], "big")

    def test_short_bytes_zero_padded(self):
        assert get_hash_as_uint64(b"\x01\x02") == 0x0102

    def test_empty_bytes_raises(self):
        with pytest.raises(DecodeError):
            get_hash_as_uint64(b"")

    def test_unsupported_type_raises(self):
        with pytest.raises(DecodeError):
            get_hash_as_uint64(1.5)


def make_pool(block_size=4, concurrency=2):
    index = InMemoryIndex(InMemoryIndexConfig(size=1000, pod_cache_size=10))
    tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=block_size))
    pool = EventsPool(EventsConfig(concurrency=concurrency), index, tp)
    return pool, index, tp


class TestEventsPool:
    def test_block_stored_adds_dual_keys(self):
        pool, index, tp = make_pool()
        tokens = [1, 2, 3, 4, 5, 6, 7, 8]
        request_keys = tp.tokens_to_kv_block_keys(None, tokens, "m")
        ev = BlockStored(
            block_hashes=[100, 200],
            parent_block_hash=None,
            token_ids=tokens,
            block_size=4,
        )
        pool.digest_events("pod-a", "m", [ev])
        # lookup by request keys finds the pod on gpu tier
        result = index.lookup(request_keys, set())
        for k in request_keys:
            assert result[k] == [PodEntry("pod-a", "gpu")]
        # engine keys map to request keys
        assert index.get_request_key(Key("m", 100)) == request_keys[0]
        assert index.get_request_key(Key("m", 200)) == request_keys[1]

    def test_parent_chain_stitching(self):
        pool, index, tp = make_pool()
        t1 = [1, 2, 3, 4]
        ev1 = BlockStored([100], None, t1, 4)
        pool.digest_events("pod-a", "m", [ev1])
        t2 = [5, 6, 7, 8]
        ev2 = BlockStored([200], 100, t2, 4)
        pool.digest_events("pod-a", "m", [ev2])
        # the second block's request key must continue the first chain -
        # identical to hashing all 8 tokens at once
        full = tp.tokens_to_kv_block_keys(None, t1 + t2, "m")
        result = index.lookup(full, set())
        assert set(result.keys()) == set(full)

    def test_unknown_parent_starts_fresh_chain(self):
        pool, index, tp = make_pool()
        ev = BlockStored([200], 999, [5, 6, 7, 8], 4)
        pool.digest_events("pod-a", "m", [ev])
        # parent unknown -> request key computed from root
        fresh = tp.tokens_to_kv_block_keys(None, [5, 6, 7, 8], "m")
        assert index.lookup(fresh, set())[fresh[0]] == [PodEntry("pod-a", "gpu")]

    def test_medium_sets_tier(self):
        pool, index, tp = make_pool()
        ev = BlockStored([100], None, [1, 2, 3, 4], 4, medium="CPU")
        pool.digest_events("pod-a", "m", [ev])
        keys = tp.tokens_to_kv_block_keys(None, [1, 2, 3, 4], "m")
        assert index.lookup(keys, set())[keys[0]] == [PodEntry("pod-a", "cpu")]

    def test_block_removed_evicts(self):
        pool, index, tp = make_pool()
        pool.digest_events(
            "pod-a", "m", [BlockStored([100], None, [1, 2, 3, 4], 4)]
        )
        pool.digest_events("pod-a", "m", [BlockRemoved([100])])
        keys = tp.tokens_to_kv_block_keys(None, [1, 2, 3, 4], "m")
        assert keys[0] not in index.lookup(keys + [Key("m", 1)], set())

    def test_all_blocks_cleared_is_noop(self):
        pool, index, tp = make_pool()
        pool.digest_events(
            "pod-a", "m", [BlockStored([100], None, [1, 2, 3, 4], 4)]
        )
        pool.digest_events("pod-a", "m", [AllBlocksCleared()])
        keys = tp.tokens_to_kv_block_keys(None, [1, 2, 3, 4], "m")
        assert keys[0] in index.lookup(keys, set())

    def test_sharding_is_stable_per_pod(self):
        pool, _, _ = make_pool(concurrency=4)
        from llmd_kvcache_amd.utils.hashing import fnv1a_32

        shard = fnv1a_32(b"pod-a") % 4
        for _ in range(5):
            assert fnv1a_32(b"pod-a") % 4 == shard

    def test_worker_pool_end_to_end(self):
        pool, index, tp = make_pool()
        pool.start(with_subscriber=False)
        try:
            batch = EventBatch(
                ts=0.0,
                events=[BlockStored([100], None, [1, 2, 3, 4], 4)],
            )
            pool.add_task(
                Message(
                    topic="kv@pod-a@m",
                    payload=batch.encode(),
                    seq=1,
                    pod_identifier="pod-a",
                    model_name="m",
                )
            )
            pool.drain()
            keys = tp.tokens_to_kv_block_keys(None, [1, 2, 3, 4], "m")
            assert index.lookup(keys, set())[keys[0]] == [
                PodEntry("pod-a", "gpu")
            ]
        finally:
            pool.shutdown()

    def test_poison_pill_dropped_not_retried(self):
        pool, index, _ = make_pool()
        pool.start(with_subscriber=False)
        try:
            pool.add_task(
                Message("kv@pod-a@m", b"\x00garbage", 1, "pod-a", "m")
            )
            pool.drain()  # completes without hanging
        finally:
            pool.shutdown()
