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

from typing import List, Optional, Sequence, Tuple

from ..indexer import Indexer


class WireIndexerService:
    def __init__(self, indexer: Indexer, max_batch: int = 4096):
        from ..ops import cpu_ext

        ops = cpu_ext.require()
        if not hasattr(ops, "WireFront"):
            raise RuntimeError(
                "native extension built without the wirefront - rebuild")
        self.indexer = indexer
        self._front = ops.WireFront(self._score_tokens_cb,
                                    self._score_text_cb, max_batch)
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

    # -- batch callbacks (invoked by the C++ batcher, GIL held) --------
    def _score_tokens_cb(self, model: str, pods: Sequence[str],
                         tokens_flat, offsets):
        scores = self.indexer.score_flat_tokens(
            tokens_flat, offsets, model, list(pods))
        return scores.to("cpu", non_blocking=False).contiguous(), \
            list(self.indexer.kv_block_index().registry.id_to_pod)

    def _score_text_cb(self, model: str, pods: Sequence[str],
                       prompts: List[str]):
        import numpy as np
        import torch

        pool = self.indexer.tokenizers_pool
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
        scores = self.indexer.score_flat_tokens(flat_t, off_t, model,
                                                list(pods))
        # 4-tuple: the C++ side feeds elements 2/3 (int32 tokens +
        # offsets) into its prompt cache so repeats skip Python
        return (scores.to("cpu", non_blocking=False).contiguous(),
                list(self.indexer.kv_block_index().registry.id_to_pod),
                flat_t.to(torch.int32), off_t)
