"""Isolate event-upload staging costs: pageable vs pinned H2D for the
exact shapes the ingest path moves per 512-event batch (4 MB tokens +
0.25 MB hashes + small metadata), plus the full apply_event_batches.

Run on a GPU box:  python scripts/bench_upload.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters * 1000.0


def main():
    d = torch.device("cuda:0")
    torch.cuda.set_device(d)
    n_tok = 512 * 64 * 16          # tokens per batch (int64)
    n_h = 512 * 64                 # hashes per batch

    rng = np.random.default_rng(0)
    tok_np = rng.integers(0, 128256, size=n_tok, dtype=np.int64)
    h_np = rng.integers(1, 2**62, size=n_h, dtype=np.int64)

    # a) pageable .to(non_blocking)
    def pageable():
        torch.from_numpy(tok_np).to(d, non_blocking=True)
        torch.from_numpy(h_np).to(d, non_blocking=True)
    print(f"pageable to(): {timeit(pageable):.3f} ms")

    # b) pinned staging (alloc once)
    buf_t = torch.empty(n_tok, dtype=torch.int64, pin_memory=True)
    buf_h = torch.empty(n_h, dtype=torch.int64, pin_memory=True)

    def pinned_copy_only():
        buf_t.copy_(torch.from_numpy(tok_np))
        buf_h.copy_(torch.from_numpy(h_np))
    print(f"pinned host copy_ only: {timeit(pinned_copy_only):.3f} ms")

    def pinned_full():
        buf_t.copy_(torch.from_numpy(tok_np))
        buf_h.copy_(torch.from_numpy(h_np))
        buf_t.to(d, non_blocking=True)
        buf_h.to(d, non_blocking=True)
    print(f"pinned copy_+to(): {timeit(pinned_full):.3f} ms")

    def pinned_np_copy():
        np.copyto(buf_t.numpy(), tok_np)
        np.copyto(buf_h.numpy(), h_np)
        buf_t.to(d, non_blocking=True)
        buf_h.to(d, non_blocking=True)
    print(f"pinned np.copyto+to(): {timeit(pinned_np_copy):.3f} ms")

    ev = torch.cuda.Event()

    def pinned_evented():
        ev.synchronize()
        buf_t.copy_(torch.from_numpy(tok_np))
        buf_h.copy_(torch.from_numpy(h_np))
        buf_t.to(d, non_blocking=True)
        buf_h.to(d, non_blocking=True)
        ev.record()
    print(f"pinned + event synchronize(): {timeit(pinned_evented):.3f} ms")

    def pinned_query_spin():
        while not ev.query():
            time.sleep(0)
        buf_t.copy_(torch.from_numpy(tok_np))
        buf_h.copy_(torch.from_numpy(h_np))
        buf_t.to(d, non_blocking=True)
        buf_h.to(d, non_blocking=True)
        ev.record()
    print(f"pinned + event query-spin: {timeit(pinned_query_spin):.3f} ms")

    def event_create_each():
        e = torch.cuda.Event()
        e.record()
        e.query()
    print(f"event create+record per call: {timeit(event_create_each):.3f} ms")

    # c) the real thing: apply_event_batches via bench-identical batches
    from llmd_kvcache_amd.kvblock.gpu_index import (GpuIndex, GpuIndexConfig)
    from llmd_kvcache_amd.kvblock.token_processor import (
        ChunkedTokenDatabase, TokenProcessorConfig)
    from llmd_kvcache_amd.kvevents.events import BlockStored

    tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=16))
    idx = GpuIndex(GpuIndexConfig(capacity=1 << 22, pods_per_key=10))
    for i in range(64):
        idx.registry.pod_id(f"pod-{i}")
    batches = []
    eh = 1
    for e in range(512):
        hs = np.arange(eh, eh + 64, dtype=np.uint64)
        eh += 64
        toks = rng.integers(0, 128256, size=64 * 16, dtype=np.int64)
        batches.append((f"pod-{e % 64}", "m", [BlockStored(hs, None, toks, 16)]))

    def apply_batch():
        idx.apply_event_batches(batches, tp)
        torch.cuda.synchronize()
    for mode in ("1", "0"):
        os.environ["KVIDX_PINNED"] = mode
        ms = timeit(apply_batch, iters=10, warmup=3)
        print(f"apply_event_batches(512 ev) pinned={mode}: {ms:.3f} ms "
              f"({512 * 64 / ms * 1000:.0f} blocks/s)")


if __name__ == "__main__":
    main()
