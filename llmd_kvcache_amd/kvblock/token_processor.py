"""Tokens -> chained KV-block keys (vLLM-compatible content addressing).

Behavioral parity with reference pkg/kvcache/kvblock/token_processor.go:
 - tokens are chunked into BlockSize-token chunks (default 16, the vLLM
   default; token_processor.go:31), partial tail chunks dropped (:126-138);
 - hash chain h_i = FNV-64a(canonical-CBOR([h_{i-1}, chunk, null]))
   (:94-112); root = FNV-64a(hash_seed) (:81-90), which must be aligned with
   the vLLM fleet's PYTHONHASHSEED-derived seed;
 - a chain may be continued from an explicit parent key (:141-162), which the
   event write path uses to stitch chains (kvevents/pool.go:279-296).

Fast paths: the C++ extension (ops._C.tokens_to_chunk_hashes) computes the
same chain natively; the HIP kernel (ops on gfx950) batches many prompts per
launch, one lane per prompt chain.  This Python implementation is the golden
reference for both and the fallback when the extension is unavailable on
CPU-only hosts.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Sequence

from ..utils import hashing
from .keys import Key

DEFAULT_BLOCK_SIZE = 16


@dataclass
class TokenProcessorConfig:
    block_size: int = DEFAULT_BLOCK_SIZE
    # Aligned with vLLM's PYTHONHASHSEED; see token_processor.go:36-40.
    hash_seed: str = ""
    # Pluggable chain hash (utils.hashing.CHAIN_ALGOS): "fnv-64a" is
    # current vLLM/reference behavior and the only algo with C++/HIP
    # fast paths; alternatives run on the Python path.
    hash_algo: str = "fnv-64a"
    # Host-side session/prefix chain cache (kvblock/chain_cache.py):
    # warm shared prefixes skip the serial FNV/CBOR chain entirely.
    # 0 disables.  Applies only to root-parented chains (the read path);
    # event chains with explicit parents bypass it.
    chain_cache_entries: int = 1 << 16
    chain_cache_seg_chunks: int = 32
    _init_hash: Optional[int] = field(default=None, repr=False)

    def init_hash(self) -> int:
        if self._init_hash is None:
            self._init_hash = hashing.CHAIN_ALGOS[self.hash_algo][1](
                self.hash_seed)
        return self._init_hash


class ChunkedTokenDatabase:
    """Converts token sequences to chained block keys."""

    def __init__(self, config: Optional[TokenProcessorConfig] = None):
        self.config = config or TokenProcessorConfig()
        self._native = None
        try:  # optional C++ fast path
            from ..ops import cpu_ext

            self._native = cpu_ext.maybe_load()
        except Exception:  # pragma: no cover - ops package always importable
            self._native = None
        self.chain_cache = None
        if self.config.chain_cache_entries > 0:
            from .chain_cache import ChainCache

            self.chain_cache = ChainCache(
                max_entries=self.config.chain_cache_entries,
                seg_chunks=self.config.chain_cache_seg_chunks,
            )

    @property
    def block_size(self) -> int:
        return self.config.block_size

    def chunk_hashes(self, parent_hash: int, tokens: Sequence[int]) -> List[int]:
        """Full-chunk chain hashes starting from parent_hash.  Warm
        root-parented prefixes are served from the session chain cache
        (only the uncached tail is computed, seeded by the last cached
        hash)."""
        bs = self.config.block_size
        n_chunks = len(tokens) // bs
        if n_chunks == 0:
            return []
        cache = self.chain_cache
        if (cache is not None
                and parent_hash == self.config.init_hash()
                and n_chunks >= cache.seg_chunks):
            from .chain_cache import ChainCache

            tokens_b = ChainCache._token_bytes(tokens[: n_chunks * bs])
            covered, chain = cache.lookup(tokens_b, bs, n_chunks)
            if covered >= n_chunks:
                return list(chain[:n_chunks])
            if covered:
                tail = self._chain(chain[covered - 1],
                                   tokens[covered * bs: n_chunks * bs])
                full = list(chain[:covered]) + tail
            else:
                full = self._chain(parent_hash, tokens[: n_chunks * bs])
            cache.store(tokens_b, bs, full)
            return full
        return self._chain(parent_hash, tokens[: n_chunks * bs])

    def _chain(self, parent_hash: int, tokens: Sequence[int]) -> List[int]:
        """Uncached chain over full chunks (C++ fast path or Python)."""
        bs = self.config.block_size
        n_chunks = len(tokens) // bs
        if n_chunks == 0:
            return []
        if self._native is not None and self.config.hash_algo == "fnv-64a":
            return self._native.tokens_to_chunk_hashes(
                list(tokens[: n_chunks * bs]), parent_hash, bs
            )
        link = hashing.CHAIN_ALGOS[self.config.hash_algo][0]
        hashes = []
        h = parent_hash
        for i in range(n_chunks):
            h = link(h, tokens[i * bs : (i + 1) * bs])
            hashes.append(h)
        return hashes

    def tokens_to_kv_block_keys(
        self,
        parent_key: Optional[Key],
        tokens: Sequence[int],
        model_name: str,
    ) -> List[Key]:
        """Parity with TokensToKVBlockKeys (token_processor.go:141-162)."""
        parent_hash = (
            parent_key.chunk_hash if parent_key is not None else self.config.init_hash()
        )
        return [
            Key(model_name, h) for h in self.chunk_hashes(parent_hash, tokens)
        ]
