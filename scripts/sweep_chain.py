"""Hash-chain kernel sweep on MI355X: batch x ILP grid.

    python scripts/sweep_chain.py
"""
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])
import torch

from llmd_kvcache_amd.ops import cpu_ext


def main():
    mod = cpu_ext.require()
    K = 512  # chunks per prompt (8k tokens / bs 16)
    BS = 16
    results = []
    for B in (512, 1024, 2048, 4096, 8192, 16384):
        toks = torch.randint(0, 1 << 31, (K * BS, B), dtype=torch.int32,
                             device="cuda")
        parents = torch.full((B,), -3750763034362895579, dtype=torch.int64,
                             device="cuda")  # init_hash("") as i64
        nch = torch.full((B,), K, dtype=torch.int32, device="cuda")
        for ilp in (1, 9, 2, 4):
            if B // ilp < 64:
                continue
            mod.gpu_hash_chain_tr(toks, parents, nch, BS, K, ilp)  # warmup
            torch.cuda.synchronize()
            t0 = time.monotonic()
            iters = 5
            for _ in range(iters):
                mod.gpu_hash_chain_tr(toks, parents, nch, BS, K, ilp)
            torch.cuda.synchronize()
            dt = (time.monotonic() - t0) / iters
            results.append((B, ilp, dt * 1e3, B / dt))
            print(f"B={B:6d} ilp={ilp} {dt*1e3:8.3f} ms  "
                  f"{B/dt/1e6:8.3f} M prompts/s  "
                  f"{B*K/dt/1e9:6.2f} G chunks/s", flush=True)
    best = {}
    for B, ilp, ms, qps in results:
        if B not in best or qps > best[B][1]:
            best[B] = (ilp, qps)
    print("best ILP per batch:", {b: v[0] for b, v in sorted(best.items())})


if __name__ == "__main__":
    main()
