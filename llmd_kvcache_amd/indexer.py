"""KV-cache Indexer orchestrator: the read path.

Parity with reference pkg/kvcache/indexer.go:
 - Config aggregates prefix-store / token-processor / index / scorer /
   tokenization-pool configs (:36-43) with a defaults constructor (:47-60);
 - get_pod_scores (:132-166): tokenize (blocking) -> block keys -> index
   lookup (filtered by the candidate pod set; empty set = all pods) ->
   longest-prefix score; empty key list returns no scores;
 - run() starts the tokenization pool workers (:116-118);
 - kv_block_index() exposes the index for the events write path (:121-123).

MI355X note: with a GpuIndex backend the lookup+score steps run fused in a
single HIP kernel (ops/csrc/kvidx_hip.hip); with the sharded parallel index
(parallel/sharded.py) per-shard partial hit masks are merged over RCCL.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

from .kvblock.index import Index, IndexConfig, new_index
from .utils.logging import get_logger, trace
from .kvblock.keys import Key
from .kvblock.token_processor import ChunkedTokenDatabase, TokenProcessorConfig
from .scorer import (
    KVBlockScorerConfig,
    KVCacheBackendConfig,
    default_kv_cache_backend_configs,
    new_kv_block_scorer,
)
from .tokenization.pool import TokenizationConfig, TokenizationPool
from .tokenization.prefixstore import LRUStoreConfig, LRUTokenStore


@dataclass
class Config:
    prefix_store: LRUStoreConfig = field(default_factory=LRUStoreConfig)
    token_processor: TokenProcessorConfig = field(
        default_factory=TokenProcessorConfig
    )
    kv_block_index: IndexConfig = field(default_factory=IndexConfig.default)
    scorer: KVBlockScorerConfig = field(default_factory=KVBlockScorerConfig)
    tokenizers_pool: TokenizationConfig = field(
        default_factory=TokenizationConfig
    )
    backend_configs: List[KVCacheBackendConfig] = field(
        default_factory=default_kv_cache_backend_configs
    )


_log = get_logger("indexer")


class Indexer:
    def __init__(
        self,
        config: Optional[Config] = None,
        tokenization_pool: Optional[TokenizationPool] = None,
        kv_block_index: Optional[Index] = None,
    ):
        self.config = config or Config()

        self.tokens_indexer = LRUTokenStore(self.config.prefix_store)
        self.tokens_processor = ChunkedTokenDatabase(self.config.token_processor)
        self._kv_block_index = kv_block_index or new_index(
            self.config.kv_block_index
        )
        # scorer backend configs are overridden by top-level backend configs
        # (indexer.go:96-97)
        self.config.scorer.backend_configs = self.config.backend_configs
        self.kv_block_scorer = new_kv_block_scorer(self.config.scorer)
        self.tokenizers_pool = tokenization_pool or TokenizationPool(
            self.config.tokenizers_pool, self.tokens_indexer
        )

    def run(self) -> None:
        self.tokenizers_pool.run()

    def shutdown(self) -> None:
        self.tokenizers_pool.shutdown()

    def kv_block_index(self) -> Index:
        return self._kv_block_index

    def get_pod_scores(
        self,
        render_req,
        prompt: str,
        model_name: str,
        pod_identifiers: Sequence[str],
    ) -> Dict[str, float]:
        # 1. tokenize prompt (blocking on the pool)
        tokens = self.tokenizers_pool.tokenize(render_req, prompt, model_name)
        trace(_log, "tokenized prompt: %d tokens", len(tokens))

        # 2. block keys
        block_keys = self.tokens_processor.tokens_to_kv_block_keys(
            None, tokens, model_name
        )
        if not block_keys:
            trace(_log, "no block keys found, returning empty scores")
            return {}
        trace(_log, "block keys: %d (first=%s)", len(block_keys),
              block_keys[0])
        return self._score_keys(block_keys, pod_identifiers)

    def _score_keys(
        self, block_keys: List[Key], pod_identifiers: Sequence[str]
    ) -> Dict[str, float]:
        # Fast path: table-backed indexes (CPU C++ or gfx950 HIP) fuse
        # lookup + longest-prefix scoring into one native call/kernel.
        fused = getattr(self._kv_block_index, "fused_scores", None)
        if fused is not None:
            import torch

            from .kvblock.gpu_index import _to_i64

            idx = self._kv_block_index
            device = idx.device
            hashes = torch.tensor(
                [_to_i64(k.chunk_hash) for k in block_keys],
                dtype=torch.int64, device=device,
            )
            if idx.table.is_cuda:
                counts = torch.tensor([0, len(block_keys)],
                                      dtype=torch.int32, device=device)
            else:
                counts = torch.tensor([len(block_keys)], dtype=torch.int32)
            weights = idx.tier_weights(
                {b.name: b.weight for b in self.config.backend_configs}
            )
            scores = fused(hashes, counts, block_keys[0].model_name,
                           set(pod_identifiers), weights,
                           max_k=len(block_keys))
            return idx.scores_to_map(scores)[0]

        # 3. generic index lookup (empty filter set = all pods)
        key_to_pods = self._kv_block_index.lookup(
            block_keys, set(pod_identifiers)
        )
        trace(_log, "lookup hit %d/%d keys", len(key_to_pods),
              len(block_keys))
        # 4. score
        scores = self.kv_block_scorer.score(block_keys, key_to_pods)
        trace(_log, "pod scores: %s", scores)
        return scores

    def score_tokens(
        self,
        tokens: Sequence[int],
        model_name: str,
        pod_identifiers: Sequence[str],
    ) -> Dict[str, float]:
        """Pre-tokenized scoring entry point (MI355X-native addition for
        callers that already hold token ids; skips the tokenization pool)."""
        block_keys = self.tokens_processor.tokens_to_kv_block_keys(
            None, tokens, model_name
        )
        if not block_keys:
            return {}
        return self._score_keys(block_keys, pod_identifiers)

    def score_tokens_batch(
        self,
        token_lists: Sequence[Sequence[int]],
        model_name: str,
        pod_identifiers: Sequence[str],
    ) -> List[Dict[str, float]]:
        """Batched scoring: N prompts through ONE fused kernel call
        (the designed hot interface - bench.py measures this shape at
        22M scores/s). Falls back to per-prompt scoring on generic
        backends.

        The chain step runs through the PARALLEL C++ op
        (ops.hash_chain_batch, at::parallel_for across prompts) rather
        than a per-prompt Python loop - at wire-level request rates the
        chains are the dominant host cost (round-2: the coalescing
        service front feeds this method)."""
        fused = getattr(self._kv_block_index, "fused_scores", None)
        tp = self.tokens_processor
        native = (fused is not None and tp._native is not None
                  and tp.config.hash_algo == "fnv-64a")
        if not native:
            keys_per_prompt = [
                tp.tokens_to_kv_block_keys(None, t, model_name)
                for t in token_lists
            ]
            return self._score_keys_batch(keys_per_prompt, model_name,
                                          pod_identifiers)

        import numpy as np
        import torch

        B = len(token_lists)
        lens = [len(t) for t in token_lists]
        flat = np.empty(sum(lens), dtype=np.int64)
        off = np.zeros(B + 1, dtype=np.int64)
        pos = 0
        for i, t in enumerate(token_lists):
            n = lens[i]
            flat[pos:pos + n] = np.asarray(t, dtype=np.int64)
            pos += n
            off[i + 1] = pos
        scores = self.score_flat_tokens(
            torch.from_numpy(flat), torch.from_numpy(off), model_name,
            pod_identifiers)
        return self._kv_block_index.scores_to_map(scores)

    def score_flat_tokens(
        self,
        tokens_flat,
        offsets,
        model_name: str,
        pod_identifiers: Sequence[str],
    ):
        """Core batched scorer over flat int64 token tensors: parallel
        C++ chain -> one fused probe/score.  Returns the [B, num_pods]
        float32 score tensor (CPU or device).  This is the entry the C++
        wirefront calls once per micro-batch."""
        import torch

        from .kvblock.gpu_index import _to_i64

        tp = self.tokens_processor
        idx = self._kv_block_index
        bs = tp.block_size
        B = int(offsets.numel()) - 1
        init = _to_i64(tp.config.init_hash())
        parents = torch.full((B,), init, dtype=torch.int64)
        hashes, chunk_off = idx.table.ops.hash_chain_batch(
            tokens_flat, offsets, parents, bs)
        counts = (chunk_off[1:] - chunk_off[:-1]).to(torch.int32)
        max_k = int(counts.max()) if B else 0
        num_pods = idx._num_pods_padded()
        if max_k == 0:
            return torch.zeros((B, num_pods), dtype=torch.float32)
        weights = idx.tier_weights(
            {b.name: b.weight for b in self.config.backend_configs}
        )
        fused = idx.fused_scores
        if idx.table.is_cuda:
            device = idx.device
            return fused(hashes.to(device),
                         chunk_off.to(torch.int32).to(device),
                         model_name, set(pod_identifiers), weights,
                         max_k=max_k)
        return fused(hashes, counts, model_name,
                     set(pod_identifiers), weights, max_k=max_k)

    def _score_keys_batch(
        self, keys_per_prompt, model_name: str,
        pod_identifiers: Sequence[str],
    ) -> List[Dict[str, float]]:
        """Generic-backend batched scoring (per-prompt calls)."""
        fused = getattr(self._kv_block_index, "fused_scores", None)
        if fused is None or not any(keys_per_prompt):
            return [self._score_keys(ks, pod_identifiers) if ks else {}
                    for ks in keys_per_prompt]

        import torch

        from .kvblock.gpu_index import _to_i64

        idx = self._kv_block_index
        device = idx.device
        flat = [_to_i64(k.chunk_hash) for ks in keys_per_prompt for k in ks]
        hashes = torch.tensor(flat, dtype=torch.int64, device=device)
        weights = idx.tier_weights(
            {b.name: b.weight for b in self.config.backend_configs}
        )
        counts = [len(ks) for ks in keys_per_prompt]
        if idx.table.is_cuda:
            offs = [0]
            for c in counts:
                offs.append(offs[-1] + c)
            counts_t = torch.tensor(offs, dtype=torch.int32, device=device)
        else:
            counts_t = torch.tensor(counts, dtype=torch.int32)
        scores = fused(hashes, counts_t, model_name, set(pod_identifiers),
                       weights, max_k=max(counts))
        return idx.scores_to_map(scores)
