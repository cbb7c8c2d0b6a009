"""Ingest breakdown: host prep vs kernel time for apply_event_batches.

Feeds the flagship event shape (512 events x 64 blocks x 1024 tokens per
batch) and reports blocks/sec plus where the time goes - the host-side
Python/numpy staging loop vs the on-device kernels (sync'd separately).
Run on a GPU host:  python scripts/bench_ingest.py [--batches N]
"""

import argparse
import sys
import time

sys.path.insert(0, ".")

import numpy as np
import torch

from llmd_kvcache_amd.kvblock.gpu_index import GpuIndex, GpuIndexConfig
from llmd_kvcache_amd.kvblock.token_processor import (ChunkedTokenDatabase,
                                                      TokenProcessorConfig)
from llmd_kvcache_amd.kvevents.events import BlockStored

BLOCK_SIZE = 16
BLOCKS_PER_EVENT = 64
EVENTS_PER_BATCH = 512
NUM_PODS = 64
VOCAB = 128256


def make_batches(n_batches, rng):
    batches = []
    next_hash = 1
    for b in range(n_batches):
        batch = []
        for e in range(EVENTS_PER_BATCH):
            toks = rng.integers(0, VOCAB,
                                size=BLOCKS_PER_EVENT * BLOCK_SIZE,
                                dtype=np.int64)
            hs = np.arange(next_hash, next_hash + BLOCKS_PER_EVENT,
                           dtype=np.uint64)
            next_hash += BLOCKS_PER_EVENT
            batch.append((f"pod-{e % NUM_PODS}", "m",
                          [BlockStored(hs, None, toks, BLOCK_SIZE)]))
        batches.append(batch)
    return batches


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batches", type=int, default=24)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    rng = np.random.default_rng(7)
    tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=BLOCK_SIZE))
    idx = GpuIndex(GpuIndexConfig(capacity=1 << 22, pods_per_key=10))
    for p in range(NUM_PODS):
        idx.registry.pod_id(f"pod-{p}")
    batches = make_batches(args.batches, rng)

    # warmup
    idx.apply_event_batches(batches[0], tp)
    torch.cuda.synchronize()

    # end-to-end pipelined (fresh inserts; host staging of batch i+1
    # overlaps batch i's kernels - the events pool's deployment shape)
    t0 = time.monotonic()
    for b in batches[1:]:
        idx.apply_event_batches(b, tp)
    torch.cuda.synchronize()
    wall = time.monotonic() - t0
    blocks = (len(batches) - 1) * EVENTS_PER_BATCH * BLOCKS_PER_EVENT
    print(f"end-to-end pipelined: {blocks / wall / 1e6:.2f}M blocks/s "
          f"({wall * 1000 / (len(batches) - 1):.2f} ms/batch)")

    # per-batch synced on a FRESH index (re-inserting existing keys is
    # cheaper and cache-warm, so reusing the table above would flatter
    # this number)
    idx2 = GpuIndex(GpuIndexConfig(capacity=1 << 22, pods_per_key=10))
    for p in range(NUM_PODS):
        idx2.registry.pod_id(f"pod-{p}")
    idx2.apply_event_batches(batches[0], tp)
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for b in batches[1:]:
        idx2.apply_event_batches(b, tp)
        torch.cuda.synchronize()
    synced = time.monotonic() - t0
    print(f"per-batch synced (fresh): {blocks / synced / 1e6:.2f}M blocks/s "
          f"({synced * 1000 / (len(batches) - 1):.2f} ms/batch)")


if __name__ == "__main__":
    main()
