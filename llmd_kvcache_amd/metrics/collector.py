"""Prometheus metrics + periodic metrics beat.

Parity with reference pkg/kvcache/metrics/collector.go:28-157:
 - counters/histograms: kvcache_index_{admissions,evictions,lookup_requests,
   lookup_hits}_total, kvcache_index_max_pod_hit_count,
   kvcache_index_lookup_latency_seconds,
   kvcache_tokenization_{tokenization_latency,tokenized_tokens,
   render_chat_template_latency};
 - register() is idempotent (collector.go:86-93);
 - start_metrics_logging emits a periodic log beat (collector.go:97-157).
"""

from __future__ import annotations

import logging
import threading
import time

logger = logging.getLogger("llmd_kvcache_amd.metrics")

monotonic = time.monotonic

_registered = False
_lock = threading.Lock()

# metric handles (populated by register())
admissions = None
evictions = None
lookup_requests = None
lookup_hits = None
max_pod_hit_count = None
lookup_latency = None
tokenization_latency = None
tokenized_tokens = None
render_chat_template_latency = None


def register() -> None:
    """Idempotently create and register the prometheus collectors."""
    global _registered, admissions, evictions, lookup_requests, lookup_hits
    global max_pod_hit_count, lookup_latency, tokenization_latency
    global tokenized_tokens, render_chat_template_latency
    with _lock:
        if _registered:
            return
        from prometheus_client import Counter, Histogram

        admissions = Counter(
            "kvcache_index_admissions_total",
            "Total number of KV-block admissions into the index",
        )
        evictions = Counter(
            "kvcache_index_evictions_total",
            "Total number of KV-block evictions from the index",
        )
        lookup_requests = Counter(
            "kvcache_index_lookup_requests_total",
            "Total number of index lookup requests",
        )
        lookup_hits = Counter(
            "kvcache_index_lookup_hits_total",
            "Total number of keys hit across lookups",
        )
        max_pod_hit_count = Histogram(
            "kvcache_index_max_pod_hit_count",
            "Per-lookup maximum consecutive hit count across pods",
            buckets=(0, 1, 2, 4, 8, 16, 32, 64, 128, 256, 512, 1024),
        )
        lookup_latency = Histogram(
            "kvcache_index_lookup_latency_seconds",
            "Index lookup latency in seconds",
        )
        tokenization_latency = Histogram(
            "kvcache_tokenization_tokenization_latency_seconds",
            "Tokenization latency in seconds",
            labelnames=("backend",),
        )
        tokenized_tokens = Counter(
            "kvcache_tokenization_tokenized_tokens_total",
            "Total number of tokens produced by tokenization",
            labelnames=("backend",),
        )
        render_chat_template_latency = Histogram(
            "kvcache_tokenization_render_chat_template_latency_seconds",
            "Chat template rendering latency in seconds",
        )
        _registered = True


def observe_tokenization(backend: str, latency_s: float, n_tokens: int) -> None:
    if not _registered:
        return
    tokenization_latency.labels(backend=backend).observe(latency_s)
    tokenized_tokens.labels(backend=backend).inc(n_tokens)


def observe_render_latency(latency_s: float) -> None:
    if _registered:
        render_chat_template_latency.observe(latency_s)


_beat_thread = None
_beat_stop = threading.Event()


def start_metrics_logging(interval_s: float) -> None:
    """Non-blocking periodic metrics log beat."""
    global _beat_thread
    if _beat_thread is not None or interval_s <= 0:
        return
    _beat_stop.clear()

    def _beat():
        while not _beat_stop.wait(interval_s):
            try:
                parts = []
                for name, c in (
                    ("admissions", admissions),
                    ("evictions", evictions),
                    ("lookup_requests", lookup_requests),
                    ("lookup_hits", lookup_hits),
                ):
                    if c is not None:
                        parts.append(f"{name}={c._value.get():.0f}")
                logger.info("metrics beat: %s", " ".join(parts))
            except Exception:  # never kill the beat
                logger.exception("metrics beat failed")

    _beat_thread = threading.Thread(target=_beat, name="metrics-beat", daemon=True)
    _beat_thread.start()


def stop_metrics_logging() -> None:
    global _beat_thread
    _beat_stop.set()
    if _beat_thread is not None:
        _beat_thread.join(timeout=1.0)
        _beat_thread = None
