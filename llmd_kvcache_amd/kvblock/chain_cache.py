"""Host-side session/prefix chain cache for the read path.

Routers see the same session prefix repeatedly (growing chat prompts,
shared system prompts).  The block-key chain is inherently serial - the
only way to make a warm prefix FREE is to not recompute it at all
(ROADMAP round-1 #1: the in-HBM memo variant traded ALU for HBM latency
and lost; this host-side cache removes the work instead).

Design:
 - prompts are signed at SEGMENT boundaries (seg_chunks chunks = 512
   tokens by default) with a STREAMING xxh3-128 over the raw token
   bytes: one pass over the prompt yields every boundary signature
   (~2 us for 8k tokens, vs ~160 us for the FNV/CBOR chain itself);
 - the cache maps boundary signature -> (shared chain-hash list,
   n_chunks covered).  A stored prompt registers every boundary, all
   pointing at ONE shared list, so a later prompt that shares only the
   first j segments still hits at its own boundary j;
 - lookup probes boundaries longest-first and returns the longest
   cached prefix; the caller computes only the remaining chunks, seeded
   by the last cached hash.

128-bit signatures make collisions negligible (~2^-64 at billions of
entries); a collision would mis-score one routing decision, never
corrupt the index.  The reference has no analog (its CPU chain is the
whole read path); the closest structure is the text-level prefix store
(pkg/tokenization/prefixstore/lru_store.go: 256-char chunked xxhash
chain), applied here one level lower, at block-key granularity.
"""

from __future__ import annotations

import threading
from typing import List, Optional, Sequence, Tuple

import numpy as np
import xxhash

DEFAULT_SEG_CHUNKS = 32  # 32 chunks x 16 tokens = 512 tokens per segment
DEFAULT_MAX_ENTRIES = 1 << 16


class ChainCache:
    """LRU of chain-hash prefixes keyed by token-prefix signature."""

    def __init__(self, max_entries: int = DEFAULT_MAX_ENTRIES,
                 seg_chunks: int = DEFAULT_SEG_CHUNKS):
        self.max_entries = max_entries
        self.seg_chunks = seg_chunks
        self._lock = threading.Lock()
        # sig -> (shared hash list, n_chunks); OrderedDict-free LRU via
        # a monotonically bumped clock would be overkill: dict preserves
        # insertion order and move-to-end is O(1).
        self._data: dict = {}
        self.hits = 0
        self.misses = 0

    # -- signatures ----------------------------------------------------
    def _boundary_sigs(self, tokens_b: bytes, block_size: int,
                       n_chunks: int) -> List[bytes]:
        """Streaming xxh3-128 digests at every segment boundary
        (segment j covers chunks [0, (j+1)*seg_chunks))."""
        seg_bytes = self.seg_chunks * block_size * 4  # int32 tokens
        h = xxhash.xxh3_128()
        sigs: List[bytes] = []
        n_segs = n_chunks // self.seg_chunks
        for j in range(n_segs):
            h.update(tokens_b[j * seg_bytes:(j + 1) * seg_bytes])
            sigs.append(h.digest())
        return sigs

    @staticmethod
    def _token_bytes(tokens: Sequence[int]) -> bytes:
        """Canonical int32 byte form of the token prefix.  ndarray input
        is ~free (one memcpy); list input pays the C conversion once
        (array.array beats np.asarray ~1.3x on plain lists)."""
        if isinstance(tokens, np.ndarray):
            if tokens.dtype == np.int32:
                return tokens.tobytes()
            return tokens.astype(np.int32, copy=False).tobytes()
        import array

        return array.array("i", tokens).tobytes()

    # -- cache ops -----------------------------------------------------
    def lookup(self, tokens_b: bytes, block_size: int,
               n_chunks: int) -> Tuple[int, Optional[List[int]]]:
        """Longest cached prefix for this prompt: returns
        (n_cached_chunks, full-chain list to slice) or (0, None)."""
        sigs = self._boundary_sigs(tokens_b, block_size, n_chunks)
        with self._lock:
            for j in range(len(sigs) - 1, -1, -1):
                ent = self._data.get(sigs[j])
                if ent is None:
                    continue
                chain, covered = ent
                need = (j + 1) * self.seg_chunks
                if covered >= need:
                    # LRU touch
                    self._data.pop(sigs[j])
                    self._data[sigs[j]] = ent
                    self.hits += 1
                    return need, chain
            self.misses += 1
        return 0, None

    def store(self, tokens_b: bytes, block_size: int,
              chain: List[int]) -> None:
        """Register every boundary of this prompt's chain (one shared
        list; boundary j's entry covers (j+1)*seg_chunks chunks)."""
        n_chunks = len(chain)
        sigs = self._boundary_sigs(tokens_b, block_size, n_chunks)
        if not sigs:
            return
        with self._lock:
            for j, sig in enumerate(sigs):
                need = (j + 1) * self.seg_chunks
                ent = self._data.get(sig)
                if ent is not None and ent[1] >= need:
                    continue  # existing entry already covers this prefix
                self._data.pop(sig, None)
                self._data[sig] = (chain, need)
            while len(self._data) > self.max_entries:
                self._data.pop(next(iter(self._data)))

    def __len__(self) -> int:
        with self._lock:
            return len(self._data)
