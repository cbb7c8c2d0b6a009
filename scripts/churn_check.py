"""Large-scale churn integrity check (GPU vs pure-Python reference).

Applies one randomized KV-event stream - chained BlockStored events,
removals, multiple pods/tiers - to BOTH the GPU index (burst kernel
path) and the pure-Python InMemoryIndex (the behavioral reference), then
compares a large sample of lookups and dual-key mappings exactly.

    python scripts/churn_check.py [--events 20000]
"""
import argparse
import random
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch

from llmd_kvcache_amd.kvblock import InMemoryIndex, InMemoryIndexConfig
from llmd_kvcache_amd.kvblock.gpu_index import GpuIndex, GpuIndexConfig
from llmd_kvcache_amd.kvblock.keys import Key
from llmd_kvcache_amd.kvblock.token_processor import (
    ChunkedTokenDatabase,
    TokenProcessorConfig,
)
from llmd_kvcache_amd.kvevents.events import BlockRemoved, BlockStored
from llmd_kvcache_amd.kvevents.pool import digest_events

BS = 16
MODEL = "m"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--events", type=int, default=20000)
    ap.add_argument("--seed", type=int, default=1234)
    ap.add_argument("--compact-every", type=int, default=0,
                    help="run TableIndex.compact() on the GPU index every "
                         "N events mid-stream (0 = never): proves "
                         "compaction preserves semantics under the live "
                         "write path")
    args = ap.parse_args()
    assert torch.cuda.is_available()

    rng = random.Random(args.seed)
    tp = ChunkedTokenDatabase(TokenProcessorConfig(block_size=BS))
    gpu = GpuIndex(GpuIndexConfig(capacity=1 << 22, pods_per_key=10))
    ref = InMemoryIndex(InMemoryIndexConfig(size=10**7, pod_cache_size=10))

    pods = [f"pod-{i}" for i in range(16)]
    chains = {p: None for p in pods}  # pod -> last engine hash
    stored = {p: [] for p in pods}
    next_hash = 1
    n_removed = 0

    batch = []
    for e in range(args.events):
        pod = pods[rng.randrange(len(pods))]
        if stored[pod] and rng.random() < 0.15:
            victim = stored[pod].pop(rng.randrange(len(stored[pod])))
            ev = BlockRemoved(list(victim))
            n_removed += len(victim)
            if chains[pod] in victim:
                # never chain a future event to a removed parent: the
                # reference restarts such chains (mapping dropped) while
                # the table retains mappings (documented divergence) -
                # this check targets the COMMON semantics
                chains[pod] = None
        else:
            n_blocks = rng.randrange(1, 6)
            toks = [rng.randrange(0, 1 << 31) for _ in range(n_blocks * BS)]
            hs = list(range(next_hash, next_hash + n_blocks))
            next_hash += n_blocks
            parent = chains[pod] if rng.random() < 0.7 else None
            ev = BlockStored(hs, parent, toks, BS,
                             medium=rng.choice([None, "CPU", None]))
            chains[pod] = hs[-1]
            stored[pod].append(hs)
        batch.append((pod, MODEL, [ev]))
        if len(batch) >= 64 or e == args.events - 1:
            # identical stream to both indexes; bursts mirror the pool's
            # GPU burst size
            gpu.apply_event_batches(batch, tp)
            for p, m, evs in batch:
                digest_events(ref, tp, p, m, evs)
            batch = []
            if args.compact_every and e // args.compact_every != \
                    (e - 63) // args.compact_every:
                torch.cuda.synchronize()
                gpu.compact()
                print(f"compacted at event {e}", flush=True)
    torch.cuda.synchronize()
    print(f"applied {args.events} events ({next_hash - 1} blocks stored, "
          f"{n_removed} removed)", flush=True)

    # compare dual-key mappings over the full engine-hash space (sampled)
    mism_map = 0
    sample = rng.sample(range(1, next_hash), min(40000, next_hash - 1))
    for h in sample:
        k = Key(MODEL, h)
        a = ref.get_request_key(k)
        b = gpu.get_request_key(k)
        # table backends deliberately retain mappings post-eviction, so
        # gpu may have a mapping where ref does not; where ref has one
        # they must agree
        if a is not None and a != b:
            mism_map += 1
    print(f"dual-key sample {len(sample)}: {mism_map} mismatches", flush=True)

    # compare lookups: walk each pod's live chains
    mism_lookup = 0
    checked = 0
    for pod in pods:
        for hs in stored[pod][:200]:
            rks = [ref.get_request_key(Key(MODEL, h)) for h in hs]
            rks = [k for k in rks if k is not None]
            if not rks:
                continue
            a = ref.lookup(rks, set())
            b = gpu.lookup(rks, set())
            checked += 1
            if set(a.keys()) != set(b.keys()):
                mism_lookup += 1
                continue
            for k in a:
                if set(a[k]) != set(b[k]):
                    mism_lookup += 1
                    break
    print(f"lookup chains checked {checked}: {mism_lookup} mismatches",
          flush=True)
    assert mism_map == 0, "dual-key divergence"
    assert mism_lookup == 0, "lookup divergence"
    print("CHURN CHECK OK")


if __name__ == "__main__":
    main()
