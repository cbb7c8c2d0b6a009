"""Wirefront scaling sweep: io threads x batcher threads (GPU host).

    python scripts/bench_wire_sweep.py
"""

import sys

sys.path.insert(0, ".")

import torch

import bench
from llmd_kvcache_amd.kvblock.gpu_index import GpuIndex, GpuIndexConfig
from llmd_kvcache_amd.service.wirefront import WireIndexerService


def main():
    import os

    assert torch.cuda.is_available()
    # KVIDX_SWEEP_TOKENS overrides the request length (A/B: is the wire
    # ceiling the per-request CPU chain work - 512 chunks at 8192 - or
    # the parse/socket path?)
    bench.PROMPT_TOKENS = int(os.environ.get("KVIDX_SWEEP_TOKENS", "8192"))
    bench.NUM_BLOCKS = 1 << 18  # 256k blocks: faster populate, same probe shape
    idx = GpuIndex(GpuIndexConfig(capacity=1 << 20, pods_per_key=10))
    from llmd_kvcache_amd.kvblock.gpu_index import GpuIndex as _G

    idx.apply_event_batches = _G.apply_event_batches.__get__(idx)
    chains, _ = bench.populate_index(idx, torch.device("cuda:0"), 0)
    indexer = bench.build_wire_indexer(idx)

    for n_io, n_batchers, n_procs in [(16, 4, 16), (16, 4, 24), (16, 4, 32), (24, 6, 32)]:
        svc = WireIndexerService(indexer, max_batch=8192,
                                 n_batchers=n_batchers)
        port = svc.start(port=0, n_io=n_io)
        try:
            # reuse bench's measure_mode machinery via measure_wire with a
            # patched server: simplest is to re-implement the quick loop
            import json as _json
            import multiprocessing as _mp
            import time

            import numpy as np

            rng = np.random.default_rng(777)
            reqs = []
            for _ in range(32):
                chain = chains[rng.integers(len(chains))]
                toks = np.empty(bench.PROMPT_TOKENS, dtype=np.int64)
                reuse = bench.PROMPT_TOKENS // 2
                toks[:reuse] = np.tile(chain, reuse // len(chain) + 1)[:reuse]
                toks[reuse:] = rng.integers(0, bench.VOCAB,
                                            size=bench.PROMPT_TOKENS - reuse)
                body = _json.dumps({"model": bench.MODEL,
                                    "tokens": toks.tolist()}).encode()
                reqs.append((f"POST /score HTTP/1.1\r\nhost: b\r\n"
                             f"content-length: {len(body)}\r\n\r\n"
                             ).encode() + body)
            depth = 96
            blob = b"".join((reqs * (depth // len(reqs) + 1))[:depth])
            ctx = _mp.get_context("spawn")
            q = ctx.SimpleQueue()
            n_blobs = 48
            barrier = ctx.Barrier(n_procs)
            procs = [ctx.Process(target=bench._wire_client_proc,
                                 args=(port, blob, n_blobs, depth, q,
                                       barrier))
                     for _ in range(n_procs)]
            for p in procs:
                p.start()
            total, starts, ends = 0, [], []
            for _ in procs:
                n, ts, te = q.get()
                total += n
                starts.append(ts)
                ends.append(te)
            for p in procs:
                p.join(timeout=60)
            wall = max(ends) - min(starts)
            reqs_srv, batches = svc.stats()
            print(f"io={n_io} batchers={n_batchers} clients={n_procs}: "
                  f"{total / wall / 1000:.1f}k req/s "
                  f"(avg batch {reqs_srv / max(batches, 1):.1f})",
                  flush=True)
        finally:
            svc.stop()


if __name__ == "__main__":
    main()
