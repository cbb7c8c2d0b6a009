"""Python wrapper for the native (C++) wire front.

The wirefront (ops/csrc/wirefront.cpp) owns the sockets: epoll io
threads parse HTTP/1.1 + JSON natively and a batcher thread crosses
into Python exactly once per micro-batch through the two callbacks
below.  Python's role per batch is the already-hot scoring path:
parallel C++ chain (hash_chain_batch) + one fused probe/score kernel.

    svc = WireIndexerService(indexer)
    port = svc.start(port=8080, n_io=4)
    ...
    svc.stop()

Endpoints (JSON over HTTP/1.1 keep-alive, pipelining supported):
    POST /score   {"model": m, "tokens": [...]}  pre-tokenized fast path
                  {"model": m, "prompt": "..."}  text (tokenization incl.)
                  optional {"pods": ["p1", ...]} candidate filter
    GET  /health

The reference's shipped service is the Go HTTP binary
(examples/kv_events/online/main.go:269-365); this is its MI355X-native
equivalent with the request path in C++ instead of Go.
"""

from __future__ import annotations

from typing import List, Sequence, Tuple

from ..indexer import Indexer


class WireIndexerService:
    def __init__(self, indexer: Indexer, max_batch: int = 4096,
                 n_batchers: int = 2):
        from ..ops import cpu_ext

        ops = cpu_ext.require()
        if not hasattr(ops, "WireFront"):
            raise RuntimeError(
                "native extension built without the wirefront - rebuild")
        if indexer.tokens_processor.config.hash_algo != "fnv-64a":
            raise ValueError(
                "wirefront requires the fnv-64a chain (the native "
                "scoring op implements that algorithm)")
        self.indexer = indexer
        self._front = ops.WireFront(self._score_tokens_cb,
                                    self._score_text_cb, max_batch,
                                    n_batchers)
        self._running = False

    # -- lifecycle -----------------------------------------------------
    def start(self, host: str = "", port: int = 0, n_io: int = 2) -> int:
        """Binds and serves; returns the bound port (port=0 -> ephemeral).
        host="" binds loopback; use "0.0.0.0" to expose."""
        bound = self._front.start(host, port, n_io)
        self._running = True
        return bound

    def stop(self) -> None:
        if self._running:
            self._running = False
            self._front.stop()

    @property
    def port(self) -> int:
        return self._front.port()

    def stats(self) -> Tuple[int, int]:
        """(requests served, micro-batches dispatched)."""
        return self._front.requests(), self._front.batches()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.stop()
        return False

    # -- batch callbacks (invoked by the C++ batcher) ------------------
    # The Python part of a batch is only id/filter/weight staging (~us);
    # the heavy work runs inside ops.wire_score_flat, which RELEASES the
    # GIL - so multiple C++ batcher threads overlap and io threads are
    # never stalled behind scoring.
    def _stage(self, model: str, pods: Sequence[str]):
        idx = self.indexer.kv_block_index()
        tp = self.indexer.tokens_processor
        model_id = idx.registry.model_id(model)
        num_pods = idx._num_pods_padded()
        filt = idx._filter_tensor(set(pods), num_pods)
        weights = idx.tier_weights(
            {b.name: b.weight
             for b in self.indexer.config.backend_configs})
        n_tiers = max(1, len(idx.registry.id_to_tier))
        from ..kvblock.gpu_index import _to_i64

        return (idx, model_id, filt, weights, num_pods, n_tiers,
                _to_i64(tp.config.init_hash()), tp.block_size)

    @staticmethod
    def _run_flat(idx, *args):
        if idx.table.is_cuda:
            return idx.table.ops.wire_score_flat(*args)
        # CPU tables are single-writer/locked-reader (gpu_index.py
        # fused_scores takes the same lock); the op releases the GIL so
        # the lock is what serializes against concurrent event applies
        with idx._write_lock:
            return idx.table.ops.wire_score_flat(*args)

    def _score_tokens_cb(self, model: str, pods: Sequence[str],
                         tokens_flat, offsets):
        (idx, model_id, filt, weights, num_pods, n_tiers, init,
         bs) = self._stage(model, pods)
        scores = self._run_flat(
            idx, *idx.table._t(), tokens_flat, offsets, model_id, filt,
            weights, num_pods, idx.table.next_epoch(), init, bs, n_tiers)
        return scores, list(idx.registry.id_to_pod)

    def _score_text_cb(self, model: str, pods: Sequence[str],
                       prompts: List[str]):
        import numpy as np
        import torch

        pool = self.indexer.tokenizers_pool
        if hasattr(pool, "tokenize_batch"):
            token_lists = pool.tokenize_batch(list(prompts), model)
        else:  # custom pool injected without the batch API
            token_lists = [pool.tokenize(None, p, model) for p in prompts]
        lens = [len(t) for t in token_lists]
        flat = np.empty(sum(lens), dtype=np.int64)
        off = np.zeros(len(token_lists) + 1, dtype=np.int64)
        pos = 0
        for i, t in enumerate(token_lists):
            flat[pos:pos + lens[i]] = np.asarray(t, dtype=np.int64)
            pos += lens[i]
            off[i + 1] = pos
        flat_t = torch.from_numpy(flat)
        off_t = torch.from_numpy(off)
        (idx, model_id, filt, weights, num_pods, n_tiers, init,
         bs) = self._stage(model, pods)
        scores = self._run_flat(
            idx, *idx.table._t(), flat_t, off_t, model_id, filt, weights,
            num_pods, idx.table.next_epoch(), init, bs, n_tiers)
        # 4-tuple: the C++ side feeds elements 2/3 (int32 tokens +
        # offsets) into its prompt cache so repeats skip Python
        return (scores, list(idx.registry.id_to_pod),
                flat_t.to(torch.int32), off_t)
