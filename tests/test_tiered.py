"""Two-tier index tests (CPU-injected tiers; the GPU-hot variant shares
all logic and is exercised by the gpu-marked suite)."""

import pytest

torch = pytest.importorskip("torch")

from llmd_kvcache_amd.kvblock.gpu_index import NativeIndex, TableIndexConfig
from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
from llmd_kvcache_amd.kvblock.tiered import TieredIndex
from llmd_kvcache_amd.ops import cpu_ext

pytestmark = pytest.mark.skipif(
    cpu_ext.maybe_load() is None, reason="native extension not built"
)

MODEL = "m"


def make_tiered(hot_capacity=256, cold_capacity=1 << 14):
    hot = NativeIndex(TableIndexConfig(capacity=hot_capacity, pods_per_key=4))
    cold = NativeIndex(
        TableIndexConfig(capacity=cold_capacity, pods_per_key=10),
        registry=hot.registry,
    )
    return TieredIndex(hot=hot, cold=cold)


class TestTieredIndex:
    def test_basic_add_goes_to_both_tiers(self):
        t = make_tiered()
        keys = [Key(MODEL, 1), Key(MODEL, 2)]
        t.add(keys, keys, [PodEntry("pod-a", "gpu")])
        assert t.hot.lookup(keys, set())[keys[0]] == [PodEntry("pod-a", "gpu")]
        assert t.cold.lookup(keys, set())[keys[0]] == [PodEntry("pod-a", "gpu")]
        assert t.lookup(keys, set())[keys[1]] == [PodEntry("pod-a", "gpu")]

    def test_cold_tier_serves_hot_evictions(self):
        """Overfill the tiny hot tier: stolen slots must still resolve
        from the capacity tier."""
        t = make_tiered(hot_capacity=256)
        all_keys = []
        for h in range(3000):
            k = [Key(MODEL, 10_000 + h)]
            t.add(k, k, [PodEntry("pod-a", "gpu")])
            all_keys.append(k[0])
        # early keys likely evicted from hot; tiered lookup still finds them
        early = all_keys[:50]
        before = t.hot.lookup(early, set())
        assert len(before) < len(early)  # hot tier really did evict
        merged = t.lookup(early, set())
        assert len(merged) == len(early)
        # ...and the cold hits were promoted back into the hot tier
        after = t.hot.lookup(early, set())
        assert len(after) == len(early)

    def test_eviction_removes_from_both(self):
        t = make_tiered()
        keys = [Key(MODEL, 77)]
        t.add(keys, keys, [PodEntry("pod-a", "gpu")])
        t.evict(keys[0], [PodEntry("pod-a", "gpu")])
        assert keys[0] not in t.lookup(keys + [Key(MODEL, 78)], set())

    def test_get_request_key_falls_back_to_cold(self):
        t = make_tiered(hot_capacity=256)
        for h in range(3000):
            ek = [Key(MODEL, 50_000 + h)]
            rk = [Key(MODEL, 90_000 + h)]
            t.add(ek, rk, [PodEntry("pod-a", "gpu")])
        # all engine mappings resolvable through the tier stack
        misses = sum(
            1
            for h in range(0, 3000, 37)
            if t.get_request_key(Key(MODEL, 50_000 + h)) != Key(MODEL, 90_000 + h)
        )
        assert misses == 0

    def test_fused_scores_routes_to_hot(self):
        t = make_tiered()
        keys = [Key(MODEL, 5), Key(MODEL, 6)]
        t.add(keys, keys, [PodEntry("pod-a", "gpu")])
        from llmd_kvcache_amd.kvblock.gpu_index import _to_i64

        hashes = torch.tensor([_to_i64(k.chunk_hash) for k in keys],
                              dtype=torch.int64)
        counts = torch.tensor([2], dtype=torch.int32)
        maps = t.scores_to_map(t.fused_scores(hashes, counts, MODEL, set()))
        assert maps[0] == {"pod-a": 2.0}


class TestValkeyBackedTier:
    """BASELINE config 5: distributed Valkey index fronted by the
    fast-tier table (GPU in production; CPU table here) - composition
    through the Index contract."""

    def test_table_hot_valkey_cold(self):
        from llmd_kvcache_amd.kvblock.fake_redis import FakeRedisServer
        from llmd_kvcache_amd.kvblock.redis_index import (
            RedisIndexConfig,
            ValkeyIndex,
        )

        server = FakeRedisServer()
        server.start()
        try:
            hot = NativeIndex(TableIndexConfig(capacity=128, pods_per_key=4))
            cold = ValkeyIndex(
                RedisIndexConfig(address=f"valkey://127.0.0.1:{server.port}")
            )
            t = TieredIndex(hot=hot, cold=cold)
            all_keys = [Key(MODEL, 70_000 + h) for h in range(600)]
            for lo in range(0, 600, 50):
                ks = all_keys[lo:lo + 50]
                t.add(ks, ks, [PodEntry("pod-v", "gpu")])
            # hot tier evicted early keys; valkey still serves them
            early = all_keys[:40]
            before = t.hot.lookup(early, set())
            assert len(before) < len(early)  # hot tier really did evict
            merged = t.lookup(early, set())
            assert len(merged) == len(early)
            after = t.hot.lookup(early, set())
            assert len(after) == len(early)  # promoted
            # dual keys resolve through the stack
            assert t.get_request_key(all_keys[0]) == all_keys[0]
        finally:
            server.stop()


class TestPromotion:
    def test_cold_hit_promotes_to_hot(self):
        """A cold-tier hit is re-inserted into the hot tier so the next
        lookup for that prefix is hot again (roadmap #12)."""
        t = make_tiered()
        keys = [Key(MODEL, 77), Key(MODEL, 78)]
        entries = [PodEntry("pod-p", "gpu")]
        # bypass TieredIndex.add: place the keys ONLY in the cold tier
        t.cold.add(keys, keys, entries)
        assert t.hot.lookup(keys, set()) == {}
        merged = t.lookup(keys, set())
        assert merged[keys[0]] == entries and merged[keys[1]] == entries
        # now resident in the hot tier
        hot = t.hot.lookup(keys, set())
        assert hot[keys[0]] == entries and hot[keys[1]] == entries

    def test_promotion_does_not_pollute_engine_map(self):
        """write_emap=False: promoting request keys must not create
        engine->request self-mappings in the hot tier."""
        t = make_tiered()
        k = Key(MODEL, 501)
        t.cold.add([k], [k], [PodEntry("pod-q", "cpu")])
        t.lookup([k], set())
        assert t.hot.lookup([k], set())  # promoted
        assert t.hot.get_request_key(k) is None  # emap untouched
        # real engine->request mappings still work end to end
        ek, rk = Key(MODEL, 900), Key(MODEL, 901)
        t.add([ek], [rk], [PodEntry("pod-q", "gpu")])
        assert t.get_request_key(ek) == rk


class TestTieredFusedColdCorrectness:
    def _hashes(self, keys, device="cpu"):
        from llmd_kvcache_amd.kvblock.gpu_index import _to_i64

        return torch.tensor([_to_i64(k.chunk_hash) for k in keys],
                            dtype=torch.int64, device=device)

    def test_fused_scores_sees_cold_only_keys(self):
        """The read fast path must NOT lose cold-tier hits: keys present
        only in the capacity tier score correctly and get promoted."""
        t = make_tiered()
        keys = [Key(MODEL, 900 + i) for i in range(4)]
        entries = [PodEntry("pod-cold", "gpu")]
        t.cold.add(keys, keys, entries)  # cold-only (hot never saw them)
        counts = torch.tensor([len(keys)], dtype=torch.int32)
        scores = t.fused_scores(self._hashes(keys), counts, MODEL, set())
        m = t.scores_to_map(scores)[0]
        assert m == {"pod-cold": pytest.approx(4.0)}
        # and the lookup promoted them: next call is pure hot fast path
        assert len(t.hot.lookup(keys, set())) == 4
        scores2 = t.fused_scores(self._hashes(keys), counts, MODEL, set())
        assert t.scores_to_map(scores2)[0] == {"pod-cold": pytest.approx(4.0)}

    def test_fused_scores_mixed_hot_cold_prefix(self):
        """Prefix split across tiers: hot has key0, only cold has key1 -
        the merged walk must still credit both."""
        t = make_tiered()
        k0, k1 = Key(MODEL, 70), Key(MODEL, 71)
        t.add([k0], [k0], [PodEntry("pod-m", "gpu")])      # both tiers
        t.cold.add([k1], [k1], [PodEntry("pod-m", "gpu")])  # cold only
        counts = torch.tensor([2], dtype=torch.int32)
        scores = t.fused_scores(self._hashes([k0, k1]), counts, MODEL, set())
        assert t.scores_to_map(scores)[0] == {"pod-m": pytest.approx(2.0)}

    def test_indexer_fast_path_through_tiered(self):
        """End to end: Indexer._score_keys fused path on a TieredIndex
        backend reaches cold-tier data."""
        from llmd_kvcache_amd.indexer import Config, Indexer
        from llmd_kvcache_amd.kvblock.token_processor import (
            ChunkedTokenDatabase, TokenProcessorConfig)

        t = make_tiered()
        cfg = Config()
        cfg.token_processor = TokenProcessorConfig(block_size=4)
        tp = ChunkedTokenDatabase(cfg.token_processor)
        tokens = list(range(8))
        keys = tp.tokens_to_kv_block_keys(None, tokens, MODEL)
        t.cold.add(keys, keys, [PodEntry("pod-deep", "cpu")])
        idx = Indexer(cfg, kv_block_index=t)
        scores = idx.score_tokens(tokens, MODEL, [])
        assert scores == {"pod-deep": pytest.approx(2 * 0.8)}
