"""LongestPrefixScorer table tests (mirrors pkg/kvcache/kvblock_scorer_test.go)."""

import pytest

from llmd_kvcache_amd.kvblock.keys import Key, PodEntry
from llmd_kvcache_amd.scorer import (
    KVBlockScorerConfig,
    KVCacheBackendConfig,
    new_kv_block_scorer,
)


def k(h):
    return Key("m", h)


def pe(pod, tier="gpu"):
    return PodEntry(pod, tier)


@pytest.fixture
def scorer():
    return new_kv_block_scorer()


class TestLongestPrefixScorer:
    def test_empty_keys(self, scorer):
        assert scorer.score([], {}) == {}

    def test_no_pods_for_first_key(self, scorer):
        assert scorer.score([k(1)], {}) == {}

    def test_single_key_single_pod(self, scorer):
        scores = scorer.score([k(1)], {k(1): [pe("a")]})
        assert scores == {"a": 1.0}

    def test_consecutive_hits_accumulate(self, scorer):
        keys = [k(1), k(2), k(3)]
        mapping = {key: [pe("a")] for key in keys}
        assert scorer.score(keys, mapping) == {"a": 3.0}

    def test_break_in_chain_stops_scoring(self, scorer):
        keys = [k(1), k(2), k(3)]
        mapping = {k(1): [pe("a")], k(3): [pe("a")]}  # gap at key 2
        assert scorer.score(keys, mapping) == {"a": 1.0}

    def test_pod_drops_out_at_first_miss(self, scorer):
        keys = [k(1), k(2), k(3)]
        mapping = {
            k(1): [pe("a"), pe("b")],
            k(2): [pe("a")],
            k(3): [pe("a"), pe("b")],  # b returning later doesn't resume
        }
        assert scorer.score(keys, mapping) == {"a": 3.0, "b": 1.0}

    def test_pod_not_in_first_key_scores_zero(self, scorer):
        keys = [k(1), k(2)]
        mapping = {k(1): [pe("a")], k(2): [pe("a"), pe("b")]}
        scores = scorer.score(keys, mapping)
        assert scores == {"a": 2.0}
        assert "b" not in scores

    def test_tier_weights(self):
        scorer = new_kv_block_scorer(
            KVBlockScorerConfig(
                backend_configs=[
                    KVCacheBackendConfig("gpu", 1.0),
                    KVCacheBackendConfig("cpu", 0.8),
                ]
            )
        )
        keys = [k(1), k(2)]
        mapping = {
            k(1): [pe("a", "gpu"), pe("b", "cpu")],
            k(2): [pe("a", "cpu"), pe("b", "cpu")],
        }
        scores = scorer.score(keys, mapping)
        assert scores["a"] == pytest.approx(1.8)
        assert scores["b"] == pytest.approx(1.6)

    def test_max_weight_across_tiers_per_key(self):
        scorer = new_kv_block_scorer()
        keys = [k(1)]
        mapping = {k(1): [pe("a", "cpu"), pe("a", "gpu")]}
        assert scorer.score(keys, mapping) == {"a": 1.0}

    def test_unknown_tier_weighs_one(self):
        scorer = new_kv_block_scorer()
        assert scorer.score([k(1)], {k(1): [pe("a", "disk")]}) == {"a": 1.0}

    def test_unsupported_strategy_raises(self):
        with pytest.raises(ValueError):
            new_kv_block_scorer(KVBlockScorerConfig(scoring_strategy="nope"))
