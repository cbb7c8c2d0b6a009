"""Text-prefix -> token LRU cache.

Parity with reference pkg/tokenization/prefixstore/lru_store.go:
 - prompt bytes chunked into 256-byte blocks (:29-34), partial tail dropped;
 - chained keys: block_hash_i = xxhash64(LE64(prev_hash) || chunk_bytes)
   (:109-141), prev_hash starts at 0;
 - a block stores the tokens whose [_, high) byte offset ends inside the
   chunk (:131-139);
 - lookup walks the chain until the first miss and returns the concatenated
   tokens plus the covered-bytes / prompt-bytes ratio (:153-190);
 - default 500k-block LRU (:31-33).
"""

from __future__ import annotations

import struct
import threading
from array import array
from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

import xxhash

from ...utils.lru import LRUCache

DEFAULT_BLOCK_SIZE = 256
DEFAULT_MAX_CACHE_SIZE = 500_000


@dataclass
class LRUStoreConfig:
    cache_size: int = DEFAULT_MAX_CACHE_SIZE
    block_size: int = DEFAULT_BLOCK_SIZE


class LRUTokenStore:
    """In-memory prefix-to-tokens cache with xxhash chain keys."""

    def __init__(self, config: Optional[LRUStoreConfig] = None):
        config = config or LRUStoreConfig()
        self.block_size = config.block_size
        self.cache = LRUCache(config.cache_size)
        self._mu = threading.Lock()

    @staticmethod
    def _block_hash(prev_hash: int, chunk: bytes) -> int:
        return xxhash.xxh64(struct.pack("<Q", prev_hash) + chunk).intdigest()

    def add_tokenization(
        self,
        prompt: str,
        tokens: Sequence[int],
        offsets: Sequence[Tuple[int, int]],
    ) -> None:
        """offsets[i] = (low, high) byte offsets of token i in the prompt."""
        if not prompt or not tokens:
            return
        with self._mu:
            prompt_bytes = prompt.encode("utf-8")
            token_idx = 0
            prev_hash = 0
            for start in range(0, len(prompt_bytes), self.block_size):
                end = start + self.block_size
                if end > len(prompt_bytes):
                    break  # no partial blocks
                block_hash = self._block_hash(prev_hash, prompt_bytes[start:end])
                prev_hash = block_hash

                block_tokens: List[int] = []
                while token_idx < len(tokens):
                    if offsets[token_idx][1] <= end:
                        block_tokens.append(tokens[token_idx])
                        token_idx += 1
                    else:
                        break
                # stored as a typed array: a Python int list costs
                # ~28 B/token, which at the reference-default 500k-block
                # cap added up to >1 GB RSS under unique-prompt load
                # (measured); array("q") is 8 B/token in one object
                self.cache.add(block_hash, array("q", block_tokens))

    def find_longest_contained_tokens(
        self, prompt: str
    ) -> Tuple[List[int], float]:
        contained: List[int] = []
        prompt_bytes = prompt.encode("utf-8")
        prev_hash = 0
        overlap_ratio = 0.0
        for start in range(0, len(prompt_bytes), self.block_size):
            end = start + self.block_size
            if end > len(prompt_bytes):
                break
            block_hash = self._block_hash(prev_hash, prompt_bytes[start:end])
            prev_hash = block_hash
            block, ok = self.cache.get(block_hash)
            if not ok:
                break  # early-stop
            contained.extend(block)
            overlap_ratio = end / len(prompt_bytes)
        return contained, overlap_ratio
