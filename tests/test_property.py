"""Property-based tests (hypothesis) for the compatibility-critical
codecs: proto3 wire format, msgpack event schema, hash coercion, CBOR
canonical encoding, and the LRU against a model implementation."""

from collections import OrderedDict

import msgpack
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from llmd_kvcache_amd.kvevents.events import (
    BlockRemoved,
    BlockStored,
    EventBatch,
    decode_event_batch,
    get_hash_as_uint64,
)
from llmd_kvcache_amd.service import proto
from llmd_kvcache_amd.utils import hashing
from llmd_kvcache_amd.utils.lru import LRUCache

pod_names = st.text(
    alphabet=st.characters(min_codepoint=33, max_codepoint=126),
    min_size=1, max_size=24,
)


class TestProtoProperties:
    @given(
        prompt=st.text(max_size=500),
        model=st.text(max_size=60),
        pods=st.lists(pod_names, max_size=8),
    )
    @settings(max_examples=200, deadline=None)
    def test_request_roundtrip(self, prompt, model, pods):
        req = proto.GetPodScoresRequest(prompt, model, pods)
        assert proto.GetPodScoresRequest.decode(req.encode()) == req

    @given(
        scores=st.lists(
            st.tuples(pod_names, st.floats(allow_nan=False)),
            max_size=16,
        )
    )
    @settings(max_examples=200, deadline=None)
    def test_response_roundtrip(self, scores):
        resp = proto.GetPodScoresResponse(
            scores=[proto.PodScore(p, s) for p, s in scores]
        )
        back = proto.GetPodScoresResponse.decode(resp.encode())
        assert back == resp

    @given(v=st.integers(min_value=0, max_value=2**64 - 1))
    @settings(max_examples=300, deadline=None)
    def test_varint_roundtrip(self, v):
        buf = proto._encode_varint(v)
        out, pos = proto._decode_varint(buf, 0)
        assert out == v and pos == len(buf)


hash_values = st.one_of(
    st.integers(min_value=0, max_value=2**64 - 1),
    st.integers(min_value=-(2**63), max_value=-1),
    st.binary(min_size=1, max_size=16),
)


class TestEventProperties:
    @given(
        hashes=st.lists(st.integers(min_value=0, max_value=2**64 - 1),
                        min_size=1, max_size=8),
        parent=st.none() | st.integers(min_value=0, max_value=2**64 - 1),
        tokens=st.lists(st.integers(min_value=0, max_value=2**32 - 1),
                        max_size=64),
        block_size=st.integers(min_value=1, max_value=64),
        medium=st.none() | st.sampled_from(["GPU", "cpu", "Disk"]),
    )
    @settings(max_examples=200, deadline=None)
    def test_block_stored_roundtrip(self, hashes, parent, tokens,
                                    block_size, medium):
        batch = EventBatch(
            ts=1.5,
            events=[BlockStored(hashes, parent, tokens, block_size,
                                medium=medium)],
        )
        out = decode_event_batch(batch.encode())
        assert len(out.events) == 1
        ev = out.events[0]
        assert [get_hash_as_uint64(h) for h in ev.block_hashes] == hashes
        assert ev.token_ids == tokens
        assert ev.block_size == block_size
        assert (ev.parent_block_hash is None) == (parent is None)
        assert ev.medium == medium

    @given(h=hash_values)
    @settings(max_examples=300, deadline=None)
    def test_hash_coercion_total_on_supported_types(self, h):
        v = get_hash_as_uint64(h)
        assert 0 <= v < 2**64
        if isinstance(h, int):
            assert v == h % 2**64
        else:
            assert v == int.from_bytes(h[-8:], "big")

    @given(payload=st.binary(max_size=200))
    @settings(max_examples=300, deadline=None)
    def test_decode_never_crashes_on_garbage(self, payload):
        from llmd_kvcache_amd.kvevents.events import DecodeError

        try:
            decode_event_batch(payload)
        except DecodeError:
            pass  # rejected cleanly


class TestCborProperties:
    @given(v=st.integers(min_value=0, max_value=2**64 - 1))
    @settings(max_examples=300, deadline=None)
    def test_shortest_form(self, v):
        enc = hashing.cbor_encode_uint(v)
        # canonical shortest form lengths per RFC 8949 s4.2.1
        if v < 24:
            assert len(enc) == 1
        elif v <= 0xFF:
            assert len(enc) == 2
        elif v <= 0xFFFF:
            assert len(enc) == 3
        elif v <= 0xFFFFFFFF:
            assert len(enc) == 5
        else:
            assert len(enc) == 9

    @given(
        parent=st.integers(min_value=0, max_value=2**64 - 1),
        tokens=st.lists(st.integers(min_value=0, max_value=2**32 - 1),
                        min_size=1, max_size=32),
    )
    @settings(max_examples=200, deadline=None)
    def test_native_matches_python(self, parent, tokens):
        from llmd_kvcache_amd.ops import cpu_ext

        mod = cpu_ext.maybe_load()
        if mod is None:
            pytest.skip("native extension not built")
        py = hashing.chunk_hash(parent, tokens)
        a = mod.tokens_to_chunk_hashes(tokens, parent, len(tokens))
        b = mod.tokens_to_chunk_hashes_fast(tokens, parent, len(tokens))
        assert list(a) == [py] == list(b)


class TestLruModel:
    @given(
        ops=st.lists(
            st.tuples(st.sampled_from(["add", "get", "remove"]),
                      st.integers(min_value=0, max_value=12)),
            max_size=60,
        ),
        capacity=st.integers(min_value=1, max_value=6),
    )
    @settings(max_examples=200, deadline=None)
    def test_against_ordereddict_model(self, ops, capacity):
        lru = LRUCache(capacity)
        model: "OrderedDict[int, int]" = OrderedDict()
        for op, k in ops:
            if op == "add":
                lru.add(k, k * 10)
                if k in model:
                    model.move_to_end(k)
                model[k] = k * 10
                if len(model) > capacity:
                    model.popitem(last=False)
            elif op == "get":
                got, found = lru.get(k)
                assert found == (k in model)
                if found:
                    assert got == model[k]
                    model.move_to_end(k)
            else:
                assert lru.remove(k) == (model.pop(k, None) is not None)
            assert len(lru) == len(model)
            assert list(lru.keys()) == list(model.keys())


@given(st.integers(0, 2**31), st.integers(10, 120))
@settings(max_examples=25, deadline=None)
def test_compact_is_lookup_invariant(seed, n_ops):
    """Property: for ANY op sequence, TableIndex.compact() (including a
    grow-then-shrink cycle) preserves every lookup result."""
    import random as _random

    from llmd_kvcache_amd.kvblock.gpu_index import (NativeIndex,
                                                    TableIndexConfig)
    from llmd_kvcache_amd.kvblock.keys import Key
    from llmd_kvcache_amd.ops import cpu_ext
    from tests.test_native_index import random_workload, run_op

    if cpu_ext.maybe_load() is None:
        return  # extension not built in this env
    rng = _random.Random(seed)
    nat = NativeIndex(TableIndexConfig(capacity=1 << 10, pods_per_key=4))
    for op in random_workload(rng, n_ops=n_ops, key_space=64, n_pods=5):
        run_op(nat, op)
    keys = [Key("m", 1000 + i) for i in range(64)]
    before = {k: sorted(map(tuple, v))
              for k, v in nat.lookup(keys, set()).items()}
    nat.compact(new_capacity=1 << 11)
    nat.compact(new_capacity=1 << 10)
    after = {k: sorted(map(tuple, v))
             for k, v in nat.lookup(keys, set()).items()}
    assert after == before


@settings(max_examples=60, deadline=None)
@given(st.data())
def test_chain_cache_transparent(data):
    """Property: chunk_hashes with the session chain cache enabled is
    bit-identical to the uncached chain for ANY interleaving of prompts
    that share prefixes (the cache must be a pure accelerator)."""
    from llmd_kvcache_amd.kvblock.token_processor import (
        ChunkedTokenDatabase, TokenProcessorConfig)

    bs = data.draw(st.sampled_from([4, 16]))
    seg = data.draw(st.sampled_from([2, 4]))
    cached = ChunkedTokenDatabase(TokenProcessorConfig(
        block_size=bs, chain_cache_seg_chunks=seg,
        chain_cache_entries=32))
    plain = ChunkedTokenDatabase(TokenProcessorConfig(
        block_size=bs, chain_cache_entries=0))
    base = data.draw(st.lists(st.integers(0, 2**31 - 1),
                              min_size=bs * seg, max_size=bs * seg * 6))
    prompts = []
    for _ in range(data.draw(st.integers(1, 6))):
        cut = data.draw(st.integers(0, len(base)))
        tail = data.draw(st.lists(st.integers(0, 2**31 - 1),
                                  min_size=0, max_size=bs * seg * 2))
        prompts.append(base[:cut] + tail)
    init = cached.config.init_hash()
    for p in prompts:
        assert cached.chunk_hashes(init, p) == plain.chunk_hashes(init, p)
