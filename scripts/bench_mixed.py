"""Mixed read+write benchmark: Score() batches with CONCURRENT KVEvents
ingest on a second HIP stream - the realistic router steady state (the
fleet streams BlockStored/BlockRemoved while scoring runs hot).

    python scripts/bench_mixed.py [--steps 10]
"""
import argparse
import json
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import numpy as np
import torch

from llmd_kvcache_amd.kvblock.gpu_index import GpuIndex, GpuIndexConfig
from llmd_kvcache_amd.kvblock.token_processor import (
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from llmd_kvcache_amd.kvevents.events import BlockStored

BS = 16
K = 512
PROMPT = K * BS
PODS = 64


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--events-per-step", type=int, default=512)
    args = ap.parse_args()

    assert torch.cuda.is_available()
    tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=BS))
    idx = GpuIndex(GpuIndexConfig(capacity=1 << 22, pods_per_key=10))
    rng = np.random.default_rng(7)
    pods = [f"pod-{i}" for i in range(PODS)]
    for p in pods:
        idx.registry.pod_id(p)

    def make_events(n, start_hash):
        batch = []
        h = start_hash
        for e in range(n):
            toks = rng.integers(0, 1 << 31, size=64 * BS, dtype=np.int64)
            hs = np.arange(h, h + 64, dtype=np.uint64)
            h += 64
            batch.append((pods[e % PODS], "m",
                          [BlockStored(hs, None, toks, BS)]))
        return batch, h

    # warm index: 1M blocks
    h = 1
    for _ in range(32):
        batch, h = make_events(512, h)
        idx.apply_event_batches(batch, tp)
    torch.cuda.synchronize()

    # prompt workload (transposed int32)
    toks = torch.randint(0, 1 << 31, (PROMPT, args.batch), dtype=torch.int32,
                         device="cuda")
    from llmd_kvcache_amd.kvblock.gpu_index import _to_i64

    parents = torch.full((args.batch,), _to_i64(tp.config.init_hash()),
                         dtype=torch.int64, device="cuda")
    nch = torch.full((args.batch,), K, dtype=torch.int32, device="cuda")
    offs = torch.arange(0, (args.batch + 1) * K, K, dtype=torch.int32,
                        device="cuda")
    weights = idx.tier_weights()
    no_filter = torch.zeros(0, dtype=torch.int64, device="cuda")
    model_id = idx.registry.model_id("m")
    num_pods = idx._num_pods_padded()
    ops = idx.table.ops

    ingest_stream = torch.cuda.Stream()

    def read_call():
        hashes = ops.gpu_hash_chain_tr(toks, parents, nch, BS, K, 0, 1)
        hashes = hashes.view(-1)
        scores = ops.gpu_fused_score(
            *idx.table._t(), hashes, offs, model_id, no_filter, weights,
            num_pods, idx.table.next_epoch(), K,
            max(1, len(idx.registry.id_to_tier)))
        return scores.argmax(dim=1).cpu()

    # pre-stage event batches (host prep off the timed path to isolate
    # GPU-side contention)
    staged = []
    for _ in range(args.steps):
        batch, h = make_events(args.events_per_step, h)
        staged.append(batch)

    for mode in ("reads-only", "mixed"):
        for _ in range(2):  # per-mode warmup
            read_call()
        torch.cuda.synchronize()
        t0 = time.monotonic()
        blocks = 0
        for s in range(args.steps):
            if mode == "mixed":
                with torch.cuda.stream(ingest_stream):
                    idx.apply_event_batches(staged[s], tp)
                blocks += args.events_per_step * 64
            read_call()
        torch.cuda.synchronize()
        dt = time.monotonic() - t0
        qps = args.batch * args.steps / dt
        print(json.dumps({
            "mode": mode,
            "score_qps": round(qps, 1),
            "concurrent_ingest_blocks_per_s": round(blocks / dt, 1),
            "ms_per_step": round(dt / args.steps * 1e3, 3),
        }), flush=True)


if __name__ == "__main__":
    main()
