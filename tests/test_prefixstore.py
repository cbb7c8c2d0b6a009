"""Prefix-store tests (mirrors pkg/tokenization/prefixstore/lru_store_test.go)."""

from llmd_kvcache_amd.tokenization.prefixstore import (
    LRUStoreConfig,
    LRUTokenStore,
    TrieTokenStore,
)


def simple_offsets(prompt, token_len=4):
    """Tokens covering fixed-width character spans."""
    tokens, offsets = [], []
    for i, start in enumerate(range(0, len(prompt), token_len)):
        end = min(start + token_len, len(prompt))
        tokens.append(1000 + i)
        offsets.append((start, end))
    return tokens, offsets


class TestLRUTokenStore:
    def store(self, block_size=16, cache_size=100):
        return LRUTokenStore(
            LRUStoreConfig(cache_size=cache_size, block_size=block_size)
        )

    def test_add_and_retrieve_full(self):
        s = self.store(block_size=16)
        prompt = "a" * 64
        tokens, offsets = simple_offsets(prompt)
        s.add_tokenization(prompt, tokens, offsets)
        found, ratio = s.find_longest_contained_tokens(prompt)
        assert found == tokens
        assert ratio == 1.0

    def test_prefix_match(self):
        s = self.store(block_size=16)
        prompt = "a" * 64
        tokens, offsets = simple_offsets(prompt)
        s.add_tokenization(prompt, tokens, offsets)
        # extended prompt: only the original prefix blocks match
        found, ratio = s.find_longest_contained_tokens(prompt + "b" * 64)
        assert found == tokens
        assert 0 < ratio <= 0.5

    def test_mismatch_returns_empty(self):
        s = self.store(block_size=16)
        prompt = "a" * 64
        tokens, offsets = simple_offsets(prompt)
        s.add_tokenization(prompt, tokens, offsets)
        found, ratio = s.find_longest_contained_tokens("z" * 64)
        assert found == []
        assert ratio == 0.0

    def test_partial_block_dropped(self):
        s = self.store(block_size=16)
        prompt = "a" * 20  # one full block + partial
        tokens, offsets = simple_offsets(prompt)
        s.add_tokenization(prompt, tokens, offsets)
        found, ratio = s.find_longest_contained_tokens(prompt)
        # only tokens whose end-offset falls inside the single full block
        assert found == [t for t, o in zip(tokens, offsets) if o[1] <= 16]
        assert ratio == 16 / 20

    def test_chain_key_depends_on_previous_block(self):
        s = self.store(block_size=4)
        s.add_tokenization("aaaabbbb", [1, 2], [(0, 4), (4, 8)])
        # same second block content, different first block -> no match
        found, _ = s.find_longest_contained_tokens("zzzzbbbb")
        assert found == []

    def test_lru_eviction(self):
        s = self.store(block_size=4, cache_size=2)
        s.add_tokenization("aaaabbbbcccc", [1, 2, 3], [(0, 4), (4, 8), (8, 12)])
        # 3 blocks into a 2-cap cache: the first block was evicted
        found, ratio = s.find_longest_contained_tokens("aaaabbbbcccc")
        assert found == []

    def test_empty_inputs(self):
        s = self.store()
        s.add_tokenization("", [], [])
        s.add_tokenization("abc", [], [])
        found, ratio = s.find_longest_contained_tokens("")
        assert found == [] and ratio == 0.0


class TestTrieTokenStore:
    def test_add_and_retrieve(self):
        s = TrieTokenStore()
        prompt = "hello world!"
        tokens = [10, 20, 30]
        offsets = [(0, 5), (5, 11), (11, 12)]
        s.add_tokenization(prompt, tokens, offsets)
        found, ratio = s.find_longest_contained_tokens(prompt)
        assert found == tokens
        assert ratio == 1.0

    def test_partial_prefix(self):
        s = TrieTokenStore()
        prompt = "hello world!"
        tokens = [10, 20, 30]
        offsets = [(0, 5), (5, 11), (11, 12)]
        s.add_tokenization(prompt, tokens, offsets)
        found, ratio = s.find_longest_contained_tokens("hello worXYZ")
        assert found == [10]

    def test_no_match(self):
        s = TrieTokenStore()
        s.add_tokenization("abc", [1], [(0, 3)])
        found, ratio = s.find_longest_contained_tokens("xyz")
        assert found == [] and ratio == 0.0
