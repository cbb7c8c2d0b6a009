"""Tokenization worker pool (sync + async modes).

Parity with reference pkg/tokenization/pool.go:
 - default 5 workers (:32), min-prefix-overlap ratio 0.8 (:33);
 - ``tokenize`` blocks on a per-task result channel (:149-161);
 - ``enqueue_tokenization`` is fire-and-forget (:140-146);
 - process_task (:198-237): optional chat-template render, then prefix-store
   FindLongestContainedTokens; if coverage >= threshold return cached
   tokens, else full encode + AddTokenization;
 - the tokenizer is a composite fallback chain local -> uds -> hf
   (pool.go:103-127).

Implemented with a thread pool over a queue; HF tokenizers release the GIL
during Rust-side encode, so N>1 workers give real concurrency.
"""

from __future__ import annotations

import queue
import threading
import time
from dataclasses import dataclass, field
from typing import List, Optional

from .prefixstore import LRUTokenStore
from .uds import UdsTokenizerConfig
from .tokenizer import (
    CompositeTokenizer,
    HFTokenizerConfig,
    LocalTokenizerConfig,
    Tokenizer,
    new_cached_hf_tokenizer,
    new_cached_local_tokenizer,
)

DEFAULT_WORKERS = 5
DEFAULT_MIN_PREFIX_OVERLAP_RATIO = 0.8


@dataclass
class TokenizationConfig:
    workers_count: int = DEFAULT_WORKERS
    min_prefix_overlap_ratio: float = DEFAULT_MIN_PREFIX_OVERLAP_RATIO
    local: Optional[LocalTokenizerConfig] = field(
        default_factory=LocalTokenizerConfig.from_env
    )
    uds: Optional[UdsTokenizerConfig] = None
    hf: Optional[HFTokenizerConfig] = field(default_factory=HFTokenizerConfig)


@dataclass
class _Task:
    render_req: Optional[object]
    prompt: str
    model_name: str
    result: Optional["queue.Queue"]  # None => fire-and-forget
    attempts: int = 0


class TokenizationPool:
    def __init__(
        self,
        config: Optional[TokenizationConfig] = None,
        indexer: Optional[LRUTokenStore] = None,
        tokenizer: Optional[Tokenizer] = None,
    ):
        self.config = config or TokenizationConfig()
        self.indexer = indexer if indexer is not None else LRUTokenStore()
        self.min_prefix_overlap_ratio = self.config.min_prefix_overlap_ratio
        self.tokenizer = tokenizer or self._build_composite(self.config)
        self._queue: "queue.Queue[Optional[_Task]]" = queue.Queue()
        self._threads: List[threading.Thread] = []
        self._running = False

    @staticmethod
    def _build_composite(config: TokenizationConfig) -> Tokenizer:
        chain: List[Tokenizer] = []
        if config.local is not None and config.local.is_enabled():
            chain.append(new_cached_local_tokenizer(config.local))
        if config.uds is not None and config.uds.is_enabled():
            from .uds import UdsTokenizer

            chain.append(UdsTokenizer(config.uds))
        if config.hf is not None and config.hf.is_enabled():
            chain.append(new_cached_hf_tokenizer(config.hf))
        if not chain:
            raise ValueError("no tokenizer backends configured")
        return CompositeTokenizer(chain)

    # -- lifecycle -----------------------------------------------------
    def run(self) -> None:
        if self._running:
            return
        self._running = True
        for i in range(self.config.workers_count):
            t = threading.Thread(
                target=self._worker, name=f"tokenize-worker-{i}", daemon=True
            )
            t.start()
            self._threads.append(t)

    def shutdown(self) -> None:
        if not self._running:
            return
        self._running = False
        for _ in self._threads:
            self._queue.put(None)
        for t in self._threads:
            t.join(timeout=2.0)
        self._threads.clear()

    # -- API -----------------------------------------------------------
    def tokenize(
        self, render_req, prompt: str, model_name: str
    ) -> List[int]:
        """Synchronous tokenization (blocks on the worker result)."""
        result: "queue.Queue" = queue.Queue(maxsize=1)
        task = _Task(render_req, prompt, model_name, result)
        if not self._running:
            # inline mode when the pool isn't started (tests/library
            # usage): process THIS task directly - going through the
            # shared queue could dequeue a concurrent caller's task and
            # leave this caller's result queue empty forever.
            self._process(task)
        else:
            self._queue.put(task)
        outcome, payload = result.get()
        if outcome == "err":
            raise payload
        return payload

    def enqueue_tokenization(self, render_req, prompt: str, model_name: str) -> None:
        task = _Task(render_req, prompt, model_name, None)
        if not self._running:
            self._process(task)
        else:
            self._queue.put(task)

    def tokenize_batch(
        self, prompts: List[str], model_name: str
    ) -> List[List[int]]:
        """Batched synchronous tokenization (no chat templating): prefix
        store serves warm prompts; the misses encode in ONE backend
        batch call (HF tokenizers' Rust encode_batch parallelizes across
        its rayon pool with the GIL released - a per-prompt loop
        serializes ~0.7 ms/prompt of Python on cold bursts).  Used by
        the native wirefront's text micro-batches."""
        results: List[Optional[List[int]]] = [None] * len(prompts)
        miss_idx: List[int] = []
        for i, p in enumerate(prompts):
            tokens, overlap = self.indexer.find_longest_contained_tokens(p)
            if overlap >= self.min_prefix_overlap_ratio:
                results[i] = tokens
            else:
                miss_idx.append(i)
        if miss_idx:
            encoded = self.tokenizer.encode_batch(
                [prompts[i] for i in miss_idx], model_name)
            for i, (token_ids, offsets) in zip(miss_idx, encoded):
                self.indexer.add_tokenization(prompts[i], token_ids, offsets)
                results[i] = token_ids
        return results  # type: ignore[return-value]

    def _worker(self) -> None:
        while True:
            task = self._queue.get()
            if task is None:
                return
            self._process(task)

    MAX_ASYNC_RETRIES = 2

    def _process(self, task: _Task) -> None:
        try:
            tokens = self._process_task(task)
            if task.result is not None:
                task.result.put(("ok", tokens))
        except Exception as e:
            if task.result is not None:
                task.result.put(("err", e))
            elif task.attempts < self.MAX_ASYNC_RETRIES:
                # fire-and-forget tasks are re-queued with backoff, the
                # reference's rate-limited retry (pool.go:187-191)
                task.attempts += 1

                def requeue():
                    time.sleep(0.05 * (2 ** task.attempts))
                    if self._running:
                        self._queue.put(task)

                threading.Thread(target=requeue, daemon=True).start()

    def _process_task(self, task: _Task) -> List[int]:
        prompt = task.prompt
        if task.render_req is not None:
            prompt = self.tokenizer.render_chat_template(task.render_req)

        tokens, overlap = self.indexer.find_longest_contained_tokens(prompt)
        if overlap >= self.min_prefix_overlap_ratio:
            return tokens

        token_ids, offsets = self.tokenizer.encode(prompt, task.model_name)
        self.indexer.add_tokenization(prompt, token_ids, offsets)
        return token_ids
