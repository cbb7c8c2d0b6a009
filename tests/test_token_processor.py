"""Token processor (chunked chained hashing) behavior tests.

Mirrors reference semantics from pkg/kvcache/kvblock/token_processor.go:
chunking with dropped partial tails, chain continuation from a parent key,
seed handling.
"""

import pytest

from llmd_kvcache_amd.kvblock.keys import Key
from llmd_kvcache_amd.kvblock.token_processor import (
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from llmd_kvcache_amd.utils import hashing


def db(block_size=4, seed=""):
    return ChunkedTokenDatabase(
        TokenProcessorConfig(block_size=block_size, hash_seed=seed)
    )


class TestChunking:
    def test_partial_tail_dropped(self):
        d = db(block_size=4)
        keys = d.tokens_to_kv_block_keys(None, [1, 2, 3, 4, 5, 6], "m")
        assert len(keys) == 1

    def test_too_few_tokens_returns_empty(self):
        d = db(block_size=4)
        assert d.tokens_to_kv_block_keys(None, [1, 2, 3], "m") == []

    def test_exact_multiple(self):
        d = db(block_size=4)
        keys = d.tokens_to_kv_block_keys(None, list(range(8)), "m")
        assert len(keys) == 2


class TestChain:
    def test_chain_links(self):
        d = db(block_size=2)
        keys = d.tokens_to_kv_block_keys(None, [1, 2, 3, 4], "m")
        root = hashing.init_hash("")
        h1 = hashing.chunk_hash(root, [1, 2])
        h2 = hashing.chunk_hash(h1, [3, 4])
        assert [k.chunk_hash for k in keys] == [h1, h2]
        assert all(k.model_name == "m" for k in keys)

    def test_parent_key_continuation(self):
        d = db(block_size=2)
        full = d.tokens_to_kv_block_keys(None, [1, 2, 3, 4], "m")
        head = d.tokens_to_kv_block_keys(None, [1, 2], "m")
        tail = d.tokens_to_kv_block_keys(head[-1], [3, 4], "m")
        assert head + tail == full

    def test_seed_changes_chain(self):
        d0 = db(block_size=2, seed="")
        d1 = db(block_size=2, seed="42")
        k0 = d0.tokens_to_kv_block_keys(None, [1, 2], "m")
        k1 = d1.tokens_to_kv_block_keys(None, [1, 2], "m")
        assert k0 != k1
        assert k1[0].chunk_hash == hashing.chunk_hash(
            hashing.fnv1a_64(b"42"), [1, 2]
        )

    def test_vllm_default_block_size(self):
        d = ChunkedTokenDatabase()
        assert d.block_size == 16

    def test_deterministic(self):
        d = db(block_size=16)
        tokens = list(range(64))
        a = d.tokens_to_kv_block_keys(None, tokens, "m")
        b = d.tokens_to_kv_block_keys(None, tokens, "m")
        assert a == b


class TestNativeParity:
    """If the C++ extension is built, its chain must match Python exactly."""

    def test_native_matches_python(self):
        from llmd_kvcache_amd.ops import cpu_ext

        mod = cpu_ext.maybe_load()
        if mod is None:
            import pytest

            pytest.skip("native extension not built")
        import random

        rng = random.Random(7)
        tokens = [rng.randrange(0, 2**32) for _ in range(160)]
        py = []
        h = hashing.init_hash("")
        for i in range(10):
            h = hashing.chunk_hash(h, tokens[i * 16 : (i + 1) * 16])
            py.append(h)
        native = mod.tokens_to_chunk_hashes(tokens, hashing.init_hash(""), 16)
        assert list(native) == py


class TestBranchlessParity:
    """chunk_hash_fast (the GPU hot path) must be bit-identical to the
    branchy reference on every CBOR shortest-form boundary."""

    def test_boundary_values(self):
        from llmd_kvcache_amd.ops import cpu_ext

        mod = cpu_ext.maybe_load()
        if mod is None:
            import pytest

            pytest.skip("native extension not built")
        boundary_tokens = [0, 1, 23, 24, 255, 256, 65535, 65536,
                           2**31 - 1, 2**32 - 1]
        # parents spanning every u64 encoding length
        parents = [0, 5, 23, 24, 200, 256, 65535, 70000, 2**32 - 1, 2**32,
                   2**63, 2**64 - 1, hashing.init_hash("")]
        import itertools

        for parent in parents:
            toks = boundary_tokens[:8]
            a = mod.tokens_to_chunk_hashes(toks, parent, 4)
            b = mod.tokens_to_chunk_hashes_fast(toks, parent, 4)
            assert list(a) == list(b), parent

    def test_random_parity(self):
        from llmd_kvcache_amd.ops import cpu_ext

        mod = cpu_ext.maybe_load()
        if mod is None:
            import pytest

            pytest.skip("native extension not built")
        import random

        rng = random.Random(3)
        toks = [rng.randrange(0, 2**32) for _ in range(320)]
        a = mod.tokens_to_chunk_hashes(toks, hashing.init_hash(""), 16)
        b = mod.tokens_to_chunk_hashes_fast(toks, hashing.init_hash(""), 16)
        assert list(a) == list(b)


class TestPluggableHash:
    def test_sha256_algo_chains_differently_but_deterministically(self):
        from llmd_kvcache_amd.utils import hashing as hh

        fnv = ChunkedTokenDatabase(TokenProcessorConfig(block_size=4))
        sha = ChunkedTokenDatabase(TokenProcessorConfig(
            block_size=4, hash_algo="sha256-cbor-64"))
        toks = list(range(12))
        a = sha.tokens_to_kv_block_keys(None, toks, "m")
        b = sha.tokens_to_kv_block_keys(None, toks, "m")
        assert a == b and len(a) == 3
        assert a != fnv.tokens_to_kv_block_keys(None, toks, "m")
        # chain property: prefix determines prefix
        c = sha.tokens_to_kv_block_keys(None, toks[:8], "m")
        assert a[:2] == c
        # link function is SHA-256 over the SAME canonical CBOR payload
        import hashlib
        payload = hh.cbor_chunk_payload(sha.config.init_hash(), toks[:4])
        expect = int.from_bytes(hashlib.sha256(payload).digest()[:8], "big")
        assert a[0].chunk_hash == expect

    def test_unknown_algo_raises(self):
        cfg = TokenProcessorConfig(hash_algo="md5")
        with pytest.raises(KeyError):
            cfg.init_hash()


class TestChainCache:
    """Host-side session chain cache (kvblock/chain_cache.py): warm
    prefixes skip the serial chain, results stay bit-identical."""

    def _tp(self, **kw):
        from llmd_kvcache_amd.kvblock.token_processor import (
            ChunkedTokenDatabase, TokenProcessorConfig)

        return ChunkedTokenDatabase(TokenProcessorConfig(**kw))

    def test_warm_hit_bit_identical(self):
        tp = self._tp(chain_cache_seg_chunks=4)
        cold = self._tp(chain_cache_entries=0)
        toks = list(range(16 * 16))  # 16 chunks
        a = tp.chunk_hashes(tp.config.init_hash(), toks)
        assert tp.chain_cache.misses == 1
        b = tp.chunk_hashes(tp.config.init_hash(), toks)
        assert tp.chain_cache.hits >= 1
        assert a == b == cold.chunk_hashes(cold.config.init_hash(), toks)

    def test_partial_prefix_hit(self):
        tp = self._tp(chain_cache_seg_chunks=4)
        cold = self._tp(chain_cache_entries=0)
        base = list(range(16 * 16))
        tp.chunk_hashes(tp.config.init_hash(), base)
        # shares the first 8 chunks (2 segments), then diverges
        variant = base[: 8 * 16] + [99999 + i for i in range(8 * 16)]
        got = tp.chunk_hashes(tp.config.init_hash(), variant)
        assert tp.chain_cache.hits == 1
        assert got == cold.chunk_hashes(cold.config.init_hash(), variant)
        # the diverged tail differs from the base chain
        assert got[8:] != tp.chunk_hashes(tp.config.init_hash(), base)[8:]

    def test_growing_session_extends_chain(self):
        tp = self._tp(chain_cache_seg_chunks=4)
        cold = self._tp(chain_cache_entries=0)
        s1 = list(range(8 * 16))
        s2 = s1 + list(range(1000, 1000 + 8 * 16))  # session grows
        tp.chunk_hashes(tp.config.init_hash(), s1)
        got = tp.chunk_hashes(tp.config.init_hash(), s2)
        assert got == cold.chunk_hashes(cold.config.init_hash(), s2)
        # and the longer chain is now cached in full
        tp.chain_cache.hits = 0
        tp.chunk_hashes(tp.config.init_hash(), s2)
        assert tp.chain_cache.hits == 1

    def test_non_root_parent_bypasses_cache(self):
        tp = self._tp(chain_cache_seg_chunks=4)
        toks = list(range(8 * 16))
        tp.chunk_hashes(12345, toks)  # event-style explicit parent
        assert tp.chain_cache.hits == 0
        assert tp.chain_cache.misses == 0
        assert len(tp.chain_cache) == 0

    def test_eviction_bounded(self):
        tp = self._tp(chain_cache_seg_chunks=4, chain_cache_entries=8)
        for i in range(20):
            toks = [i * 1000 + j for j in range(4 * 16)]
            tp.chunk_hashes(tp.config.init_hash(), toks)
        assert len(tp.chain_cache) <= 8

    def test_disabled(self):
        tp = self._tp(chain_cache_entries=0)
        assert tp.chain_cache is None
        toks = list(range(4 * 16))
        assert len(tp.chunk_hashes(tp.config.init_hash(), toks)) == 4

    def test_short_prompt_bypasses(self):
        tp = self._tp(chain_cache_seg_chunks=32)
        toks = list(range(4 * 16))  # 4 chunks < seg_chunks
        tp.chunk_hashes(tp.config.init_hash(), toks)
        assert len(tp.chain_cache) == 0
