// hip_ops.hip - gfx950 (MI355X/CDNA4) kernels for the KV-block index.
//
// Design (MI355X-first, not a port - the reference does all of this on CPU
// with Go maps + LRU locks):
//  - the block->pod table is HBM3E-resident (288 GB/GPU allows >1e9 keys);
//    probing is lock-free open addressing with device-scope atomics
//    (cross-XCD safe: atomicCAS on global memory is device scope);
//  - read path: one workgroup per prompt; all K key probes issue in
//    PARALLEL (the prefix early-stop is applied after the fact during the
//    in-LDS mask walk), so per-prompt latency is ~2 dependent HBM reads,
//    not K of them - the CPU reference walks keys serially;
//  - per-(key,tier) pod-bitmask accumulation in LDS, then a single wave
//    walks the masks in key order computing longest-prefix scores with the
//    active pod set in registers (lane l owns pod bit l of each 64-pod
//    word) - scoring parity with kvblock_scorer.go:108-151;
//  - write path: events arrive grouped by pod (per-pod ordering guarantee
//    of kvevents/pool.go:132-144 preserved); one wave per pod-group
//    processes its events serially; within a BlockStored event lane 0
//    streams the serial FNV/CBOR chain (token_processor.go:94-123) while
//    block inserts fan out across all 64 lanes;
//  - hash-chain batch kernel: one lane per prompt, chain in registers
//    (streaming CBOR->FNV, no scratch), wave64-dense.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "kvidx_common.h"

namespace kvidx {

#define DEV __device__ __forceinline__

struct DevTable {
  uint64_t* __restrict__ keys;
  uint32_t* __restrict__ meta;
  int32_t* __restrict__ stamp;
  uint32_t* __restrict__ pods;
  uint64_t* __restrict__ e_keys;
  uint32_t* __restrict__ e_meta;
  uint64_t* __restrict__ e_vals;
  uint64_t cap_mask;
  int pods_per_key;
};

DEV int64_t dev_table_find(const DevTable& v, uint64_t h, uint32_t model) {
  h = remap_hash(h);
  uint64_t s = probe_start(h, v.cap_mask);
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t i = (s + t) & v.cap_mask;
    uint64_t k = v.keys[i];
    if (k == 0) return -1;
    uint32_t m = v.meta[i];
    if ((m & META_OCC) && !(m & META_TOMB) && k == h &&
        (m & META_MODEL_MASK) == model)
      return (int64_t)i;
  }
  return -1;
}

// Lock-free probe-or-insert. Claims free slots via atomicCAS on the key
// word (deterministic probe order makes same-key racers converge on the
// same slot); tombstone resurrection via atomicCAS on meta; window-full
// falls back to stealing the lowest-stamp live slot (approx LRU).
DEV int64_t dev_table_put(const DevTable& v, uint64_t h, uint32_t model,
                          int32_t epoch) {
  h = remap_hash(h);
  uint64_t s = probe_start(h, v.cap_mask);
  int64_t first_tomb = -1;
  int64_t victim = -1;
  int32_t victim_stamp = 0;
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t i = (s + t) & v.cap_mask;
    uint64_t k = v.keys[i];
    if (k == 0) {
      uint64_t prev = atomicCAS((unsigned long long*)&v.keys[i], 0ull,
                                (unsigned long long)h);
      if (prev == 0) {  // claimed fresh slot (pods are zero-initialized)
        __threadfence();
        atomicExch(&v.meta[i], META_OCC | (model & META_MODEL_MASK));
        v.stamp[i] = epoch;
        return (int64_t)i;
      }
      k = prev;  // lost the race; re-evaluate this slot
    }
    if (k == h) {
      uint32_t m = v.meta[i];
      if (!(m & META_OCC)) return (int64_t)i;  // mid-insert by a racer
      if (!(m & META_TOMB)) {
        if ((m & META_MODEL_MASK) == model) {
          v.stamp[i] = epoch;
          return (int64_t)i;
        }
        continue;  // same hash, different model (astronomically rare)
      }
      // tombstone with our key: resurrect
      uint32_t want = META_OCC | (model & META_MODEL_MASK);
      atomicCAS(&v.meta[i], m, want);
      v.stamp[i] = epoch;
      return (int64_t)i;
    }
    uint32_t m = v.meta[i];
    if (m & META_TOMB) {
      if (first_tomb < 0) first_tomb = (int64_t)i;
    } else if (m & META_OCC) {
      int32_t st = v.stamp[i];
      if (victim < 0 || st < victim_stamp) {
        victim = (int64_t)i;
        victim_stamp = st;
      }
    }  // !OCC and k!=0: mid-insert elsewhere; skip
  }
  // Window exhausted: steal (approximate LRU eviction under pressure).
  int64_t i = first_tomb >= 0 ? first_tomb : victim;
  if (i < 0) i = (int64_t)s;
  for (int j = 0; j < v.pods_per_key; ++j) v.pods[i * v.pods_per_key + j] = 0;
  atomicExch((unsigned long long*)&v.keys[i], (unsigned long long)h);
  __threadfence();
  atomicExch(&v.meta[i], META_OCC | (model & META_MODEL_MASK));
  v.stamp[i] = epoch;
  return i;
}

DEV void dev_pod_set_add(const DevTable& v, int64_t slot, uint32_t entry,
                         int32_t epoch) {
  KVIDX_ASSERT(slot >= 0 && (uint64_t)slot <= v.cap_mask);
  KVIDX_ASSERT(entry != 0);
  uint32_t* p = v.pods + slot * v.pods_per_key;
  for (int j = 0; j < v.pods_per_key; ++j)
    if (p[j] == entry) return;
  for (int j = 0; j < v.pods_per_key; ++j) {
    uint32_t cur = p[j];
    if (cur == 0) {
      uint32_t prev = atomicCAS(&p[j], 0u, entry);
      if (prev == 0 || prev == entry) return;
    } else if (cur == entry) {
      return;
    }
  }
  atomicExch(&p[(uint32_t)epoch % v.pods_per_key], entry);  // full: overwrite
}

DEV void dev_emap_put(const DevTable& v, uint64_t h, uint32_t model,
                      uint64_t val) {
  h = remap_hash(h);
  uint64_t s = probe_start(h, v.cap_mask);
  int64_t first_tomb = -1;
  int64_t victim = -1;
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t i = (s + t) & v.cap_mask;
    uint64_t k = v.e_keys[i];
    if (k == 0) {
      uint64_t prev = atomicCAS((unsigned long long*)&v.e_keys[i], 0ull,
                                (unsigned long long)h);
      if (prev == 0) {
        v.e_vals[i] = val;
        __threadfence();
        atomicExch(&v.e_meta[i], META_OCC | (model & META_MODEL_MASK));
        return;
      }
      k = prev;
    }
    if (k == h) {
      uint32_t m = v.e_meta[i];
      if ((m & META_OCC) && (m & META_TOMB)) {
        v.e_vals[i] = val;
        __threadfence();
        atomicExch(&v.e_meta[i], META_OCC | (model & META_MODEL_MASK));
      } else {
        v.e_vals[i] = val;  // refresh value (idempotent for same chain)
      }
      return;
    }
    uint32_t m = v.e_meta[i];
    if (m & META_TOMB) {
      if (first_tomb < 0) first_tomb = (int64_t)i;
    } else if ((m & META_OCC) && victim < 0) {
      victim = (int64_t)i;
    }
  }
  int64_t i = first_tomb >= 0 ? first_tomb : victim;
  if (i < 0) i = (int64_t)s;
  atomicExch((unsigned long long*)&v.e_keys[i], (unsigned long long)h);
  v.e_vals[i] = val;
  __threadfence();
  atomicExch(&v.e_meta[i], META_OCC | (model & META_MODEL_MASK));
}

DEV int64_t dev_emap_find(const DevTable& v, uint64_t h, uint32_t model) {
  h = remap_hash(h);
  uint64_t s = probe_start(h, v.cap_mask);
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t i = (s + t) & v.cap_mask;
    uint64_t k = v.e_keys[i];
    if (k == 0) return -1;
    uint32_t m = v.e_meta[i];
    if ((m & META_OCC) && !(m & META_TOMB) && k == h &&
        (m & META_MODEL_MASK) == model)
      return (int64_t)i;
  }
  return -1;
}

// Probe one key and OR its visible pod entries into per-tier mask words.
// Returns: 0 absent, 1 present (raw pods nonzero), 2 present-but-empty.
template <typename OrFn>
DEV int dev_probe_collect(const DevTable& v, uint64_t h, uint32_t model,
                          const uint64_t* __restrict__ filter, int has_filter,
                          int num_pods, int W, int32_t epoch, OrFn&& or_bit) {
  int64_t slot = dev_table_find(v, h, model);
  if (slot < 0) return 0;
  v.stamp[slot] = epoch;  // LRU touch on read (non-atomic ok)
  const uint32_t* p = v.pods + slot * v.pods_per_key;
  int any_raw = 0;
  for (int j = 0; j < v.pods_per_key; ++j) {
    uint32_t e = p[j];
    if (e == 0) continue;
    any_raw = 1;
    uint32_t pid = pod_entry_id(e);
    uint32_t tier = pod_entry_tier(e);
    if (pid >= (uint32_t)num_pods || tier >= MAX_TIERS) continue;
    if (has_filter && !((filter[pid / 64] >> (pid % 64)) & 1)) continue;
    or_bit(tier, pid);
  }
  return any_raw ? 1 : 2;
}

// ---------------------------------------------------------------------
// Kernels
// ---------------------------------------------------------------------

__global__ void k_insert(DevTable v, const uint64_t* __restrict__ eh,
                         const uint64_t* __restrict__ rh, int64_t n,
                         uint32_t model, const uint32_t* __restrict__ entries,
                         int n_entries, int32_t epoch, int shard_id,
                         int num_shards, int emap_write) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  if (emap_write) dev_emap_put(v, eh[i], model, remap_hash(rh[i]));
  if (num_shards > 1 &&
      (int)(remap_hash(rh[i]) % (uint64_t)num_shards) != shard_id)
    return;  // engine map replicated; main table sharded by ownership
  int64_t slot = dev_table_put(v, rh[i], model, epoch);
  for (int j = 0; j < n_entries; ++j)
    dev_pod_set_add(v, slot, entries[j], epoch);
}

// Tombstone compaction (ROADMAP #9): thread per OLD slot rehashes live
// entries into a fresh bundle (lock-free via the usual CAS claims; old
// (key,model) pairs are unique so racers only contend on free slots).
__global__ void k_compact_main(DevTable ov, DevTable nv, int64_t cap) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= cap) return;
  uint32_t m = ov.meta[i];
  if (!(m & META_OCC) || (m & META_TOMB)) return;
  int64_t s2 = dev_table_put(nv, ov.keys[i], m & META_MODEL_MASK,
                             ov.stamp[i]);
  uint32_t* dst = nv.pods + s2 * nv.pods_per_key;
  const uint32_t* src = ov.pods + i * ov.pods_per_key;
  for (int j = 0; j < nv.pods_per_key; ++j) dst[j] = src[j];
}

__global__ void k_compact_emap(DevTable ov, DevTable nv, int64_t cap) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= cap) return;
  uint32_t m = ov.e_meta[i];
  if (!(m & META_OCC) || (m & META_TOMB)) return;
  dev_emap_put(nv, ov.e_keys[i], m & META_MODEL_MASK, ov.e_vals[i]);
}

__global__ void k_evict(DevTable v, const uint64_t* __restrict__ eh,
                        int64_t n, uint32_t model,
                        const uint32_t* __restrict__ entries, int n_entries) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  // emap entries retained on eviction (replica determinism under
  // sharding + chain continuity; see cpu_evict note).
  int64_t ei = dev_emap_find(v, eh[i], model);
  if (ei < 0) return;
  uint64_t req = v.e_vals[ei];
  int64_t slot = dev_table_find(v, req, model);
  if (slot < 0) return;
  uint32_t* p = v.pods + slot * v.pods_per_key;
  for (int j = 0; j < n_entries; ++j)
    for (int k = 0; k < v.pods_per_key; ++k)
      atomicCAS(&p[k], entries[j], 0u);
  bool empty = true;
  for (int k = 0; k < v.pods_per_key; ++k)
    if (p[k] != 0) { empty = false; break; }
  if (empty) atomicOr(&v.meta[slot], META_TOMB);
}

__global__ void k_get_request_keys(DevTable v,
                                   const uint64_t* __restrict__ eh, int64_t n,
                                   uint32_t model, uint8_t* __restrict__ found,
                                   uint64_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int64_t ei = dev_emap_find(v, eh[i], model);
  if (ei >= 0) {
    found[i] = 1;
    out[i] = v.e_vals[ei];
  }
}

// Generic lookup producing global found[] + per-tier masks [K, T, W].
__global__ void k_lookup_masks(DevTable v, const uint64_t* __restrict__ rh,
                               int64_t K, uint32_t model,
                               const uint64_t* __restrict__ filter,
                               int has_filter, int num_pods, int W,
                               int32_t epoch, int shard_id, int num_shards,
                               int n_tiers,
                               uint8_t* __restrict__ found,
                               unsigned long long* __restrict__ masks) {
  int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (k >= K) return;
  if (num_shards > 1 &&
      (int)(remap_hash(rh[k]) % (uint64_t)num_shards) != shard_id)
    return;  // unowned key: contributes zero masks on this shard
  unsigned long long* mk = masks + (size_t)k * n_tiers * W;
  int f = dev_probe_collect(
      v, rh[k], model, filter, has_filter, num_pods, W, epoch,
      [&](uint32_t tier, uint32_t pid) {
        if ((int)tier < n_tiers)  // interned => always true
          atomicOr(&mk[tier * W + pid / 64], 1ull << (pid % 64));
      });
  found[k] = (uint8_t)f;
}

// Fused probe + longest-prefix score. One workgroup per prompt; dynamic
// LDS holds the per-key per-tier masks; wave 0 then walks keys in order
// with the active pod set in registers (lane l = pod bit l).
__global__ void __launch_bounds__(256) k_fused_score(
    DevTable v, const uint64_t* __restrict__ hashes,
    const int32_t* __restrict__ offsets,  // [B+1]
    uint32_t model, const uint64_t* __restrict__ filter, int has_filter,
    const float* __restrict__ weights, int num_pods, int W, int32_t epoch,
    int n_tiers, float* __restrict__ scores) {
  // n_tiers = tiers actually registered (usually 1-2, <= MAX_TIERS):
  // sizing LDS by it instead of MAX_TIERS keeps 256-pod fleets (W=4)
  // at 16-32 KB/WG instead of 64 KB, preserving occupancy.
  extern __shared__ unsigned long long lds_masks[];  // [K, n_tiers, W]
  const int b = blockIdx.x;
  const int K = offsets[b + 1] - offsets[b];
  const uint64_t* h = hashes + offsets[b];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  // zero LDS
  for (int x = threadIdx.x; x < K * n_tiers * W; x += blockDim.x)
    lds_masks[x] = 0;
  __syncthreads();

  // phase 1: all probes in parallel (each is ~2 dependent HBM reads)
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    unsigned long long* mk = lds_masks + (size_t)k * n_tiers * W;
    dev_probe_collect(v, h[k], model, filter, has_filter, num_pods, W, epoch,
                      [&](uint32_t tier, uint32_t pid) {
                        if ((int)tier < n_tiers)  // interned => always true
                          atomicOr(&mk[tier * W + pid / 64],
                                   1ull << (pid % 64));
                      });
  }
  __syncthreads();

  // phase 2: the walk. Words (64-pod groups) are independent prefix
  // walks, so they are distributed across the workgroup's 4 waves
  // (wave v takes words v, v+4, ...); lane l owns pod w*64+l. The
  // per-word early break is exact (a word stops when ITS active set
  // empties - the old all-words break was only an optimization).
  for (int w = wave; w < W; w += 4) {
    float score = 0.f;
    int active = 0;
    for (int k = 0; k < K; ++k) {
      const unsigned long long* mk =
          lds_masks + (size_t)k * n_tiers * W + w;
      int cur = 0;
      float wmax = 0.f;
      for (int t = 0; t < n_tiers; ++t) {
        if ((mk[t * W] >> lane) & 1) {
          cur = 1;
          wmax = fmaxf(wmax, weights[t]);
        }
      }
      int act = (k == 0) ? cur : (active & cur);
      active = act;
      if (act) score += wmax;
      if (!__any(act)) break;  // this word's active set is empty
    }
    int pid = w * 64 + lane;
    if (pid < num_pods) scores[(size_t)b * num_pods + pid] = score;
  }
}

// Walk precomputed masks (e.g. after an RCCL all-reduce merge of shard
// masks) -> scores.  Per-pod scoring is independent across 64-pod words
// (the all-empty break is only an optimization), so huge fleets split
// across blockIdx.y word-groups of up to 16 words each.
// Grid = (B, ceil(W/16)), one wave per block.
__global__ void __launch_bounds__(64) k_score_from_masks(
    const unsigned long long* __restrict__ masks,  // [Ktot, T, W]
    const int32_t* __restrict__ offsets,           // [B+1]
    const float* __restrict__ weights, int T, int num_pods, int W,
    float* __restrict__ scores) {
  const int b = blockIdx.x;
  const int w_lo = blockIdx.y * 16;
  const int w_hi = min(W, w_lo + 16);
  const int WG = w_hi - w_lo;
  const int K = offsets[b + 1] - offsets[b];
  const int lane = threadIdx.x & 63;
  float score[16];
  int active[16];
  for (int w = 0; w < WG; ++w) {
    score[w] = 0.f;
    active[w] = 0;
  }
  for (int k = 0; k < K; ++k) {
    const unsigned long long* mk =
        masks + (size_t)(offsets[b] + k) * T * W;
    bool any = false;
    for (int w = 0; w < WG; ++w) {
      int cur = 0;
      float wmax = 0.f;
      for (int t = 0; t < T; ++t) {
        if ((mk[t * W + w_lo + w] >> lane) & 1) {
          cur = 1;
          wmax = fmaxf(wmax, weights[t]);
        }
      }
      int act = (k == 0) ? cur : (active[w] & cur);
      active[w] = act;
      if (act) score[w] += wmax;
      if (__any(act)) any = true;
    }
    if (!any) break;
  }
  for (int w = 0; w < WG; ++w) {
    int pid = (w_lo + w) * 64 + lane;
    if (pid < num_pods) scores[(size_t)b * num_pods + pid] = score[w];
  }
}

// Batched hash chains: one LANE per prompt (chains are serial per prompt,
// wave64-parallel across prompts; streaming CBOR->FNV keeps all state in
// registers).
__global__ void k_hash_chain(const int64_t* __restrict__ tokens,
                             const int64_t* __restrict__ tok_off,  // [B+1]
                             const uint64_t* __restrict__ parents,
                             const int64_t* __restrict__ chunk_off,  // [B+1]
                             int64_t B, int block_size,
                             uint64_t* __restrict__ out) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  uint64_t h = parents[b];
  const int64_t* t0 = tokens + tok_off[b];
  int64_t n_chunks = chunk_off[b + 1] - chunk_off[b];
  uint64_t* o = out + chunk_off[b];
  for (int64_t c = 0; c < n_chunks; ++c) {
    h = chunk_hash_fast(h, t0 + c * block_size, block_size);
    o[c] = h;
  }
}

// Transposed-layout hash chains (the read-path hot kernel): tokens are
// staged [token_pos][prompt] in int32, so at each chunk step all 64 lanes
// of a wave load one coalesced 256 B line per token position, and the
// next chunk's 16 independent loads prefetch under the current chunk's
// serial FNV ALU chain.  The row-major int64 variant (k_hash_chain)
// measured 2.5 ms per 512x512-chunk call on MI355X - pure HBM latency on
// strided per-lane loads; this layout removes that.
// out is [max_chunks][B] (transposed too; callers transpose back with a
// cheap torch op or consume the stride directly).
// ILP chains: each lane advances ILP independent prompt chains in one
// straight-line (branchless chunk_hash_fast) loop body, so the scheduler
// interleaves their serial FNV dependency chains - the chain is
// dependent-ALU-latency-bound at 1 wave/SIMD, and ILP-x recovers ~x of
// that latency.  Lane l of the grid owns prompts {l, l+L, .., l+(ILP-1)L}
// where L = total lanes, keeping every token load coalesced.
template <int BS, int ILP>
__global__ void k_hash_chain_tr(const int32_t* __restrict__ tokens_t,  // [T,B]
                                const uint64_t* __restrict__ parents,  // [B]
                                const int32_t* __restrict__ n_chunks,  // [B]
                                int64_t B, int64_t L, int max_chunks,
                                int row_major,
                                uint64_t* __restrict__ out) {  // [maxC,B] or [B,maxC]
  const int64_t lane = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (lane >= L) return;
  int64_t b[ILP];
  uint64_t h[ILP];
  int mc[ILP];
  bool live[ILP];
#pragma unroll
  for (int i = 0; i < ILP; ++i) {
    b[i] = lane + (int64_t)i * L;
    live[i] = b[i] < B;
    h[i] = live[i] ? parents[b[i]] : 0;
    mc[i] = live[i] ? n_chunks[b[i]] : 0;
  }
  for (int c = 0; c < max_chunks; ++c) {
    uint32_t tok[ILP][BS];
    // loads and hash bodies are UNCONDITIONAL (clamped addresses, dead
    // chains compute garbage that a select discards): keeps the whole
    // chunk step one basic block so the ILP chains actually interleave.
#pragma unroll
    for (int i = 0; i < ILP; ++i) {
      const int64_t bi = live[i] ? b[i] : 0;
#pragma unroll
      for (int j = 0; j < BS; ++j)
        tok[i][j] = (uint32_t)tokens_t[(int64_t)(c * BS + j) * B + bi];
    }
#pragma unroll
    for (int i = 0; i < ILP; ++i) {
      const bool active = live[i] && c < mc[i];
      const uint64_t h2 = chunk_hash_fast(h[i], tok[i], BS);
      h[i] = active ? h2 : h[i];
      // row_major writes land scattered (stride maxC per lane) but the
      // kernel is ALU-latency bound with idle bandwidth - writing the
      // layout the fused-score kernel consumes saves a 128 MB/call
      // transpose pass downstream.
      if (active)
        out[row_major ? (int64_t)b[i] * max_chunks + c
                      : (int64_t)c * B + b[i]] = h[i];
    }
  }
}

// Double-buffered single-chain variant: chunk c+1's coalesced loads are
// issued BEFORE chunk c's hash chain, so HBM latency hides under the
// serial ALU work (the plain variant exposes ~full load latency at each
// chunk boundary).  Token loop in chunk_hash_fast is force-unrolled so
// the buffers stay in named registers (no gpr_idx, no vmcnt(0) drain).
template <int BS>
__global__ void k_hash_chain_tr_pf(const int32_t* __restrict__ tokens_t,
                                   const uint64_t* __restrict__ parents,
                                   const int32_t* __restrict__ n_chunks,
                                   int64_t B, int64_t L, int max_chunks,
                                   int row_major,
                                   uint64_t* __restrict__ out) {
  const int64_t lane = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (lane >= L) return;
  uint64_t h = parents[lane];
  const int mc = n_chunks[lane];
  uint32_t cur[BS], nxt[BS];
#pragma unroll
  for (int j = 0; j < BS; ++j)
    cur[j] = (uint32_t)tokens_t[(int64_t)j * B + lane];
  for (int c = 0; c < max_chunks; ++c) {
    if (c + 1 < max_chunks) {
#pragma unroll
      for (int j = 0; j < BS; ++j)
        nxt[j] = (uint32_t)tokens_t[(int64_t)((c + 1) * BS + j) * B + lane];
    }
    const bool active = c < mc;
    const uint64_t h2 = chunk_hash_fast(h, cur, BS);
    h = active ? h2 : h;
    if (active)
      out[row_major ? (int64_t)lane * max_chunks + c
                    : (int64_t)c * B + lane] = h;
#pragma unroll
    for (int j = 0; j < BS; ++j) cur[j] = nxt[j];
  }
}

// Apply a batch of KV events fully on-device: one WAVE per pod-group
// (events of one pod processed serially -> per-pod ordering preserved,
// kvevents/pool.go:132-144); within a BlockStored event lane 0 streams the
// request-key chain (token_processor.go:94-123, stitched from the parent
// engine key via the engine map like pool.go:279-296) and all 64 lanes
// fan out the per-block inserts.
__global__ void k_apply_events(
    DevTable v, const int64_t* __restrict__ tokens,
    const int32_t* __restrict__ tok_off,     // [E+1]
    const uint64_t* __restrict__ ehashes,    // engine hashes, flat
    const int32_t* __restrict__ eh_off,      // [E+1]
    const uint64_t* __restrict__ parents,    // [E]
    const uint8_t* __restrict__ has_parent,  // [E]
    const uint8_t* __restrict__ ev_type,     // [E] 0=stored 1=removed
    const uint32_t* __restrict__ pod_entry,  // [E]
    const int32_t* __restrict__ grp_off,     // [G+1] event groups by pod
    const int32_t* __restrict__ model_of,    // [E] model id per event
    int64_t G, uint64_t init_hash, int block_size,
    int32_t epoch, int shard_id, int num_shards,
    uint64_t* __restrict__ req_scratch) {    // [total engine hashes]
  const int g = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (g >= G) return;
  const int e_begin = grp_off[g];
  const int e_end = grp_off[g + 1];

  for (int e = e_begin; e < e_end; ++e) {
    const int nh = eh_off[e + 1] - eh_off[e];
    const uint64_t* eh = ehashes + eh_off[e];
    const uint32_t model = (uint32_t)model_of[e];
    if (ev_type[e] == 1) {  // BlockRemoved
      for (int i = lane; i < nh; i += 64) {
        int64_t ei = dev_emap_find(v, eh[i], model);
        if (ei < 0) continue;
        uint64_t req = v.e_vals[ei];
        int64_t slot = dev_table_find(v, req, model);
        if (slot < 0) continue;  // emap retained (see cpu_evict note)
        uint32_t* p = v.pods + slot * v.pods_per_key;
        for (int k = 0; k < v.pods_per_key; ++k)
          atomicCAS(&p[k], pod_entry[e], 0u);
        bool empty = true;
        for (int k = 0; k < v.pods_per_key; ++k)
          if (p[k] != 0) { empty = false; break; }
        if (empty) atomicOr(&v.meta[slot], META_TOMB);
      }
      __builtin_amdgcn_wave_barrier();
      continue;
    }
    // BlockStored: lane 0 resolves parent + streams the request chain.
    const int n_tok = tok_off[e + 1] - tok_off[e];
    const int n_chunks = n_tok / block_size;
    // Engine-hash count must match the token-derived chain length; the
    // reference Add() (and this repo's CPU digest path) rejects the whole
    // event on mismatch - dropping here keeps CPU and GPU replicas
    // convergent on malformed streams. (Uniform per wave: n_chunks/nh
    // are scalar per event.)
    if (n_chunks != nh) continue;
    uint64_t* req = req_scratch + eh_off[e];
    if (lane == 0) {
      uint64_t parent = init_hash;
      if (has_parent[e]) {
        int64_t ei = dev_emap_find(v, parents[e], model);
        if (ei >= 0) parent = v.e_vals[ei];
      }
      const int64_t* t0 = tokens + tok_off[e];
      uint64_t h = parent;
      for (int c = 0; c < n_chunks; ++c) {
        h = chunk_hash_fast(h, t0 + (int64_t)c * block_size, block_size);
        req[c] = h;
      }
    }
    __threadfence_block();
    __builtin_amdgcn_wave_barrier();
    const int n_ins = n_chunks;
    for (int i = lane; i < n_ins; i += 64) {
      dev_emap_put(v, eh[i], model, remap_hash(req[i]));
      if (num_shards > 1 &&
          (int)(remap_hash(req[i]) % (uint64_t)num_shards) != shard_id)
        continue;
      int64_t slot = dev_table_put(v, req[i], model, epoch);
      dev_pod_set_add(v, slot, pod_entry[e], epoch);
    }
    __builtin_amdgcn_wave_barrier();
  }
}

// ---------------------------------------------------------------------
// Phase-split event application (the serial wave-per-group kernel above
// left 63 lanes idle during each chain; here chains run one LANE per
// pod-group and the per-block inserts fan out thread-per-block).
// The host guards ordering: a batch where some engine hash is both
// stored and removed falls back to the serial kernel.
// ---------------------------------------------------------------------

// Batch-local engine-hash -> global block index map (open addressing,
// zeroed tensors per call; value = idx+1 so 0 means empty).
__global__ void k_batch_bmap(const uint64_t* __restrict__ ehashes, int64_t n,
                             uint64_t* __restrict__ bkeys,
                             int32_t* __restrict__ bvals, int64_t bmask) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t h = remap_hash(ehashes[i]);
  uint64_t s = probe_start(h, (uint64_t)bmask);
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t j = (s + t) & (uint64_t)bmask;
    uint64_t prev = atomicCAS((unsigned long long*)&bkeys[j], 0ull,
                              (unsigned long long)h);
    if (prev == 0) {
      bvals[j] = (int32_t)(i + 1);  // consumers run in a later kernel
      return;
    }
    if (prev == h) return;  // duplicate hash: first writer's index stands
  }
}

DEV int64_t dev_bmap_find(const uint64_t* __restrict__ bkeys,
                          const int32_t* __restrict__ bvals, int64_t bmask,
                          uint64_t h) {
  h = remap_hash(h);
  uint64_t s = probe_start(h, (uint64_t)bmask);
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t j = (s + t) & (uint64_t)bmask;
    uint64_t k = bkeys[j];
    if (k == 0) return -1;
    if (k == h) {
      int32_t v = bvals[j];
      return v > 0 ? (int64_t)v - 1 : -1;
    }
  }
  return -1;
}

// Phase A: one lane per pod-group walks its events in order computing
// request-key chains into req_scratch.  Parent resolution: earlier block
// of the SAME group in this batch (bmap) first, then the persistent
// engine map (cross-batch), else the chain root.
__global__ void k_event_chains(
    DevTable v, const int64_t* __restrict__ tokens,
    const int32_t* __restrict__ tok_off, const uint64_t* __restrict__ ehashes,
    const int32_t* __restrict__ eh_off, const uint64_t* __restrict__ parents,
    const uint8_t* __restrict__ has_parent, const uint8_t* __restrict__ ev_type,
    const int32_t* __restrict__ grp_off, const int32_t* __restrict__ model_of,
    int64_t G, uint64_t init_hash, int block_size,
    const uint64_t* __restrict__ bkeys, const int32_t* __restrict__ bvals,
    int64_t bmask, uint64_t* __restrict__ req_scratch) {
  int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= G) return;
  const int e_begin = grp_off[g];
  const int e_end = grp_off[g + 1];
  KVIDX_ASSERT(e_begin >= 0 && e_begin <= e_end);
  const int grp_first_block = eh_off[e_begin];
  for (int e = e_begin; e < e_end; ++e) {
    if (ev_type[e] == 1) continue;  // removals have no chain
    const int nh = eh_off[e + 1] - eh_off[e];
    const int n_chunks = (tok_off[e + 1] - tok_off[e]) / block_size;
    if (n_chunks != nh) continue;  // drop mismatched event (see
                                   // k_apply_events; k_event_inserts
                                   // applies the same guard)
    uint64_t parent = init_hash;
    if (has_parent[e]) {
      int64_t bi = dev_bmap_find(bkeys, bvals, bmask, parents[e]);
      if (bi >= grp_first_block && bi < eh_off[e]) {
        parent = req_scratch[bi];  // earlier event of this group
      } else {
        int64_t ei = dev_emap_find(v, parents[e],
                                   (uint32_t)model_of[e]);
        if (ei >= 0) parent = v.e_vals[ei];
      }
    }
    const int64_t* t0 = tokens + tok_off[e];
    uint64_t h = parent;
    const int n_ins = min(n_chunks, nh);
    for (int c = 0; c < n_ins; ++c) {
      h = chunk_hash_fast(h, t0 + (int64_t)c * block_size, block_size);
      req_scratch[eh_off[e] + c] = h;
    }
  }
}

// Phase A variant for batches with NO batch-local parent references
// (the host checks parents against the batch's engine hashes): chains
// are then independent across EVENTS, not just pod groups, so one LANE
// per event runs them - 8x the parallelism of lane-per-group at the
// bench shape - and the tokens are staged TRANSPOSED [token_pos][event]
// in int32 like the read path's chain kernel, so each position is one
// coalesced 256 B line across the wave instead of 64 scattered 8 B
// reads (ROADMAP round-1 #2 follow-up; write path was 9.9M blocks/s
// with the serial chain phase dominating).
// Device-side staging for k_event_chains_ev: flat per-event int32
// tokens -> transposed [token_pos][event].  Host-side transpose of the
// ragged batch measured ~2-5 ms/batch in numpy (it briefly halved
// ingest); here it is one trivial bandwidth-bound kernel over ~2 MB.
__global__ void k_transpose_ev_tokens(const int32_t* __restrict__ flat,
                                      const int32_t* __restrict__ tok_off,
                                      int64_t E,
                                      int32_t* __restrict__ out) {  // [maxT,E]
  const int e = blockIdx.x;
  const int len = tok_off[e + 1] - tok_off[e];
  for (int i = blockIdx.y * blockDim.x + threadIdx.x; i < len;
       i += gridDim.y * blockDim.x)
    out[(int64_t)i * E + e] = flat[tok_off[e] + i];
}

__global__ void k_event_chains_ev(
    DevTable v, const int32_t* __restrict__ tokens_t,  // [maxT, E]
    const int32_t* __restrict__ tok_off, const uint64_t* __restrict__ ehashes,
    const int32_t* __restrict__ eh_off, const uint64_t* __restrict__ parents,
    const uint8_t* __restrict__ has_parent, const uint8_t* __restrict__ ev_type,
    const int32_t* __restrict__ model_of, int64_t E, uint64_t init_hash,
    int block_size, uint64_t* __restrict__ req_scratch) {
  const int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (e >= E) return;
  if (ev_type[e] == 1) return;  // removals have no chain
  const int nh = eh_off[e + 1] - eh_off[e];
  KVIDX_ASSERT(nh >= 0 && tok_off[e + 1] >= tok_off[e]);
  const int n_chunks = (tok_off[e + 1] - tok_off[e]) / block_size;
  if (n_chunks != nh) return;  // drop mismatched event (see k_apply_events)
  uint64_t parent = init_hash;
  if (has_parent[e]) {
    int64_t ei = dev_emap_find(v, parents[e], (uint32_t)model_of[e]);
    if (ei >= 0) parent = v.e_vals[ei];
  }
  uint64_t h = parent;
  uint32_t tok[64];  // block_size <= 64
  for (int c = 0; c < n_chunks; ++c) {
    for (int j = 0; j < block_size; ++j)
      tok[j] = (uint32_t)tokens_t[(int64_t)(c * block_size + j) * E + e];
    h = chunk_hash_fast(h, tok, block_size);
    req_scratch[eh_off[e] + c] = h;
  }
}

// Phase B: one thread per block applies its insert/removal.
__global__ void k_event_inserts(
    DevTable v, const uint64_t* __restrict__ ehashes,
    const int32_t* __restrict__ eh_off, const int32_t* __restrict__ ev_of,
    const uint8_t* __restrict__ ev_type, const int32_t* __restrict__ tok_off,
    const uint32_t* __restrict__ pod_entry,
    const int32_t* __restrict__ model_of, int64_t n_blocks,
    int block_size, int32_t epoch, int shard_id, int num_shards,
    const uint64_t* __restrict__ req_scratch) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n_blocks) return;
  const int e = ev_of[i];
  const uint32_t model = (uint32_t)model_of[e];
  if (ev_type[e] == 1) {  // BlockRemoved
    int64_t ei = dev_emap_find(v, ehashes[i], model);
    if (ei < 0) return;
    uint64_t req = v.e_vals[ei];
    int64_t slot = dev_table_find(v, req, model);
    if (slot < 0) return;  // emap retained (see cpu_evict note)
    uint32_t* p = v.pods + slot * v.pods_per_key;
    for (int k = 0; k < v.pods_per_key; ++k)
      atomicCAS(&p[k], pod_entry[e], 0u);
    bool empty = true;
    for (int k = 0; k < v.pods_per_key; ++k)
      if (p[k] != 0) { empty = false; break; }
    if (empty) atomicOr(&v.meta[slot], META_TOMB);
    return;
  }
  // BlockStored: dropped entirely when the engine-hash count mismatches
  // the token-derived chain (same guard as k_event_chains/k_apply_events)
  const int local = (int)(i - eh_off[e]);
  const int n_chunks = (tok_off[e + 1] - tok_off[e]) / block_size;
  if (n_chunks != eh_off[e + 1] - eh_off[e]) return;
  if (local >= n_chunks) return;
  const uint64_t req = req_scratch[i];
  dev_emap_put(v, ehashes[i], model, remap_hash(req));
  if (num_shards > 1 &&
      (int)(remap_hash(req) % (uint64_t)num_shards) != shard_id)
    return;
  int64_t slot = dev_table_put(v, req, model, epoch);
  dev_pod_set_add(v, slot, pod_entry[e], epoch);
}

// ---------------------------------------------------------------------
// Launchers
// ---------------------------------------------------------------------

static DevTable dev_view(at::Tensor& keys, at::Tensor& meta, at::Tensor& stamp,
                         at::Tensor& pods, at::Tensor& e_keys,
                         at::Tensor& e_meta, at::Tensor& e_vals,
                         int64_t pods_per_key) {
  TORCH_CHECK(keys.is_cuda(), "table must live on the GPU");
  int64_t cap = keys.numel();
  TORCH_CHECK((cap & (cap - 1)) == 0, "capacity must be a power of two");
  DevTable v;
  v.keys = reinterpret_cast<uint64_t*>(keys.data_ptr<int64_t>());
  v.meta = reinterpret_cast<uint32_t*>(meta.data_ptr<int32_t>());
  v.stamp = stamp.data_ptr<int32_t>();
  v.pods = reinterpret_cast<uint32_t*>(pods.data_ptr<int32_t>());
  v.e_keys = reinterpret_cast<uint64_t*>(e_keys.data_ptr<int64_t>());
  v.e_meta = reinterpret_cast<uint32_t*>(e_meta.data_ptr<int32_t>());
  v.e_vals = reinterpret_cast<uint64_t*>(e_vals.data_ptr<int64_t>());
  v.cap_mask = (uint64_t)cap - 1;
  v.pods_per_key = (int)pods_per_key;
  return v;
}

#define U64P(t) reinterpret_cast<const uint64_t*>((t).data_ptr<int64_t>())
#define STREAM at::cuda::getCurrentCUDAStream().stream()

void gpu_insert(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
                at::Tensor pods, at::Tensor e_keys, at::Tensor e_meta,
                at::Tensor e_vals, int64_t pods_per_key,
                at::Tensor engine_hashes, at::Tensor request_hashes,
                int64_t model_id, at::Tensor pod_entries, int64_t epoch,
                int64_t shard_id, int64_t num_shards, int64_t emap_write) {
  auto v = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                    pods_per_key);
  int64_t n = engine_hashes.numel();
  if (n == 0) return;
  int threads = 256;
  int blocks = (int)((n + threads - 1) / threads);
  hipLaunchKernelGGL(k_insert, dim3(blocks), dim3(threads), 0, STREAM, v,
                     U64P(engine_hashes), U64P(request_hashes), n,
                     (uint32_t)model_id,
                     reinterpret_cast<const uint32_t*>(
                         pod_entries.data_ptr<int32_t>()),
                     (int)pod_entries.numel(), (int32_t)epoch,
                     (int)shard_id, (int)num_shards, (int)emap_write);
}

void gpu_compact(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
                 at::Tensor pods, at::Tensor e_keys, at::Tensor e_meta,
                 at::Tensor e_vals, int64_t pods_per_key,
                 at::Tensor n_keys, at::Tensor n_meta, at::Tensor n_stamp,
                 at::Tensor n_pods, at::Tensor n_e_keys, at::Tensor n_e_meta,
                 at::Tensor n_e_vals) {
  auto ov = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                     pods_per_key);
  auto nv = dev_view(n_keys, n_meta, n_stamp, n_pods, n_e_keys, n_e_meta,
                     n_e_vals, pods_per_key);
  int64_t cap = keys.numel();
  int threads = 256;
  int blocks = (int)((cap + threads - 1) / threads);
  hipLaunchKernelGGL(k_compact_main, dim3(blocks), dim3(threads), 0, STREAM,
                     ov, nv, cap);
  hipLaunchKernelGGL(k_compact_emap, dim3(blocks), dim3(threads), 0, STREAM,
                     ov, nv, cap);
}

void gpu_evict(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
               at::Tensor pods, at::Tensor e_keys, at::Tensor e_meta,
               at::Tensor e_vals, int64_t pods_per_key,
               at::Tensor engine_hashes, int64_t model_id,
               at::Tensor pod_entries) {
  auto v = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                    pods_per_key);
  int64_t n = engine_hashes.numel();
  if (n == 0) return;
  int threads = 256;
  int blocks = (int)((n + threads - 1) / threads);
  hipLaunchKernelGGL(k_evict, dim3(blocks), dim3(threads), 0, STREAM, v,
                     U64P(engine_hashes), n, (uint32_t)model_id,
                     reinterpret_cast<const uint32_t*>(
                         pod_entries.data_ptr<int32_t>()),
                     (int)pod_entries.numel());
}

std::vector<at::Tensor> gpu_get_request_keys(
    at::Tensor keys, at::Tensor meta, at::Tensor stamp, at::Tensor pods,
    at::Tensor e_keys, at::Tensor e_meta, at::Tensor e_vals,
    int64_t pods_per_key, at::Tensor engine_hashes, int64_t model_id) {
  auto v = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                    pods_per_key);
  int64_t n = engine_hashes.numel();
  auto found = at::zeros({n}, engine_hashes.options().dtype(at::kByte));
  auto out = at::zeros({n}, engine_hashes.options());
  if (n == 0) return {found, out};
  int threads = 256;
  int blocks = (int)((n + threads - 1) / threads);
  hipLaunchKernelGGL(k_get_request_keys, dim3(blocks), dim3(threads), 0,
                     STREAM, v, U64P(engine_hashes), n, (uint32_t)model_id,
                     found.data_ptr<uint8_t>(),
                     reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>()));
  return {found, out};
}

std::vector<at::Tensor> gpu_lookup(at::Tensor keys, at::Tensor meta,
                                   at::Tensor stamp, at::Tensor pods,
                                   at::Tensor e_keys, at::Tensor e_meta,
                                   at::Tensor e_vals, int64_t pods_per_key,
                                   at::Tensor request_hashes, int64_t model_id,
                                   at::Tensor filter_words, int64_t num_pods,
                                   int64_t epoch, int64_t shard_id,
                                   int64_t num_shards, int64_t n_tiers) {
  auto v = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                    pods_per_key);
  int64_t K = request_hashes.numel();
  int64_t W = (num_pods + 63) / 64;
  bool has_filter = filter_words.numel() > 0;
  if (n_tiers < 1) n_tiers = 1;
  if (n_tiers > MAX_TIERS) n_tiers = MAX_TIERS;
  auto found = at::zeros({K}, request_hashes.options().dtype(at::kByte));
  // tier planes sized by the registered tier count (see cpu_lookup)
  auto masks = at::zeros({K, n_tiers, W}, request_hashes.options());
  if (K == 0) return {found, masks};
  int threads = 256;
  int blocks = (int)((K + threads - 1) / threads);
  hipLaunchKernelGGL(
      k_lookup_masks, dim3(blocks), dim3(threads), 0, STREAM, v,
      U64P(request_hashes), K, (uint32_t)model_id,
      has_filter ? U64P(filter_words) : nullptr, has_filter ? 1 : 0,
      (int)num_pods, (int)W, (int32_t)epoch, (int)shard_id, (int)num_shards,
      (int)n_tiers, found.data_ptr<uint8_t>(),
      reinterpret_cast<unsigned long long*>(masks.data_ptr<int64_t>()));
  return {found, masks};
}

at::Tensor gpu_fused_score(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
                           at::Tensor pods, at::Tensor e_keys,
                           at::Tensor e_meta, at::Tensor e_vals,
                           int64_t pods_per_key, at::Tensor hashes,
                           at::Tensor offsets, int64_t model_id,
                           at::Tensor filter_words, at::Tensor weights,
                           int64_t num_pods, int64_t epoch, int64_t max_k,
                           int64_t n_tiers) {
  auto v = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                    pods_per_key);
  int64_t B = offsets.numel() - 1;
  int64_t W = (num_pods + 63) / 64;
  TORCH_CHECK(W <= 16, "fused score supports up to 1024 pods");
  if (n_tiers < 1) n_tiers = 1;
  if (n_tiers > MAX_TIERS) n_tiers = MAX_TIERS;
  size_t lds = (size_t)max_k * n_tiers * W * sizeof(uint64_t);
  TORCH_CHECK(lds <= 64 * 1024,
              "fused score LDS overflow: reduce keys per prompt or pods");
  bool has_filter = filter_words.numel() > 0;
  auto scores = at::zeros({B, num_pods},
                          hashes.options().dtype(at::kFloat));
  if (B == 0) return scores;
  hipLaunchKernelGGL(
      k_fused_score, dim3((int)B), dim3(256), lds, STREAM, v, U64P(hashes),
      offsets.data_ptr<int32_t>(), (uint32_t)model_id,
      has_filter ? U64P(filter_words) : nullptr, has_filter ? 1 : 0,
      weights.data_ptr<float>(), (int)num_pods, (int)W, (int32_t)epoch,
      (int)n_tiers, scores.data_ptr<float>());
  return scores;
}

at::Tensor gpu_score_from_masks(at::Tensor masks, at::Tensor offsets,
                                at::Tensor weights, int64_t num_pods) {
  int64_t B = offsets.numel() - 1;
  int64_t W = (num_pods + 63) / 64;
  TORCH_CHECK(W <= 64, "score_from_masks supports up to 4096 pods");
  TORCH_CHECK(masks.dim() == 3, "masks must be [Ktot, T, W]");
  int64_t T = masks.size(1);  // tier planes, sized by the lookup
  auto scores = at::zeros({B, num_pods}, masks.options().dtype(at::kFloat));
  if (B == 0) return scores;
  int wgroups = (int)((W + 15) / 16);
  hipLaunchKernelGGL(
      k_score_from_masks, dim3((int)B, wgroups), dim3(64), 0, STREAM,
      reinterpret_cast<const unsigned long long*>(masks.data_ptr<int64_t>()),
      offsets.data_ptr<int32_t>(), weights.data_ptr<float>(), (int)T,
      (int)num_pods, (int)W, scores.data_ptr<float>());
  return scores;
}

std::vector<at::Tensor> gpu_hash_chain(at::Tensor tokens, at::Tensor tok_off,
                                       at::Tensor parents,
                                       int64_t block_size) {
  TORCH_CHECK(tokens.is_cuda());
  int64_t B = parents.numel();
  auto chunk_off_cpu = at::zeros({B + 1}, at::kLong);
  {
    auto off_cpu = tok_off.to(at::kCPU);
    const int64_t* offp = off_cpu.data_ptr<int64_t>();
    int64_t* cop = chunk_off_cpu.data_ptr<int64_t>();
    for (int64_t b = 0; b < B; ++b)
      cop[b + 1] = cop[b] + (offp[b + 1] - offp[b]) / block_size;
  }
  auto chunk_off = chunk_off_cpu.to(tokens.device());
  int64_t total = chunk_off_cpu.data_ptr<int64_t>()[B];
  auto out = at::empty({total}, tokens.options());
  if (B == 0) return {out, chunk_off};
  int threads = 256;
  int blocks = (int)((B + threads - 1) / threads);
  hipLaunchKernelGGL(k_hash_chain, dim3(blocks), dim3(threads), 0, STREAM,
                     tokens.data_ptr<int64_t>(), tok_off.data_ptr<int64_t>(),
                     U64P(parents), chunk_off.data_ptr<int64_t>(), B,
                     (int)block_size,
                     reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>()));
  return {out, chunk_off};
}

// tokens_t: int32 [T, B] (transposed); parents int64 [B]; n_chunks
// int32 [B].  Returns out int64 [max_chunks, B] (transposed hashes).
at::Tensor gpu_hash_chain_tr(at::Tensor tokens_t, at::Tensor parents,
                             at::Tensor n_chunks, int64_t block_size,
                             int64_t max_chunks, int64_t ilp,
                             int64_t row_major) {
  TORCH_CHECK(tokens_t.is_cuda() && tokens_t.dtype() == at::kInt);
  TORCH_CHECK(tokens_t.dim() == 2, "tokens_t must be [T, B]");
  int64_t B = tokens_t.size(1);
  TORCH_CHECK(parents.numel() == B && n_chunks.numel() == B);
  // row_major=1: out is [B, max_chunks] (what the fused-score kernel
  // consumes directly - no transpose pass); default [max_chunks, B].
  auto out = at::zeros(row_major ? std::initializer_list<int64_t>{B, max_chunks}
                                 : std::initializer_list<int64_t>{max_chunks, B},
                       parents.options());
  if (B == 0 || max_chunks == 0) return out;
  // ILP trades wave count for per-lane chain interleave: only pay the
  // wave reduction when there are waves to spare (tuned by
  // scripts/sweep_chain.py on MI355X).
  int64_t want_ilp = ilp;
  // Sweep on MI355X (profiles/r01_chain_sweep.md): ILP>1 always loses -
  // register pressure + extra loads outweigh the interleave; the
  // branchless body alone keeps the SIMD pipeline fed.  Default 1.
  if (want_ilp <= 0) want_ilp = 1;
  int64_t L = (B + want_ilp - 1) / want_ilp;  // lanes
  int threads = 256;
  int blocks = (int)((L + threads - 1) / threads);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(threads), 0, STREAM,
                       tokens_t.data_ptr<int32_t>(), U64P(parents),
                       n_chunks.data_ptr<int32_t>(), B, L, (int)max_chunks,
                       (int)row_major,
                       reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>()));
  };
  bool done = false;
  if (block_size == 16) {
    done = true;
    // Measured (profiles/r01_chain_sweep.md): the plain single-chain
    // kernel with the force-unrolled token loop wins at every batch
    // vs the explicit double-buffer prefetch variant - the prefetch's
    // register moves cost more than the load latency they hide once the
    // unroll removed the gpr_idx waits. (The original pf A/B runs also
    // had a dispatch bug that gave pf 1/8 the lanes, flattering it 8x;
    // the verdict only strengthens with the fix.)
    switch (want_ilp) {
      case 2: launch(k_hash_chain_tr<16, 2>); break;
      case 8:  // A/B: prefetch variant is one lane per prompt (no ILP)
        L = B;
        blocks = (int)((L + threads - 1) / threads);
        launch(k_hash_chain_tr_pf<16>);
        break;
      case 4: launch(k_hash_chain_tr<16, 4>); break;
      default: launch(k_hash_chain_tr<16, 1>); break;
    }
  }
  if (!done) {
    int fixed_ilp = block_size == 64 ? 2 : 4;
    L = (B + fixed_ilp - 1) / fixed_ilp;
    blocks = (int)((L + threads - 1) / threads);
    switch (block_size) {
      case 32: launch(k_hash_chain_tr<32, 4>); break;
      case 64: launch(k_hash_chain_tr<64, 2>); break;
      case 4:  launch(k_hash_chain_tr<4, 4>);  break;
      case 8:  launch(k_hash_chain_tr<8, 4>);  break;
      default:
        TORCH_CHECK(false, "hash_chain_tr supports block sizes 4/8/16/32/64;"
                           " use gpu_hash_chain for other sizes");
    }
  }
  return out;
}

void gpu_apply_events(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
                      at::Tensor pods, at::Tensor e_keys, at::Tensor e_meta,
                      at::Tensor e_vals, int64_t pods_per_key,
                      at::Tensor tokens, at::Tensor tok_off,
                      at::Tensor ehashes, at::Tensor eh_off,
                      at::Tensor parents, at::Tensor has_parent,
                      at::Tensor ev_type, at::Tensor pod_entry,
                      at::Tensor grp_off, at::Tensor model_of,
                      int64_t init_hash_bits, int64_t block_size,
                      int64_t epoch, int64_t shard_id, int64_t num_shards) {
  auto v = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                    pods_per_key);
  int64_t G = grp_off.numel() - 1;
  if (G == 0) return;
  TORCH_CHECK(model_of.numel() == ev_type.numel(),
              "model_of must have one id per event");
  auto req_scratch = at::empty({std::max<int64_t>(ehashes.numel(), 1)},
                               ehashes.options());
  const int waves_per_block = 4;
  int blocks = (int)((G + waves_per_block - 1) / waves_per_block);
  hipLaunchKernelGGL(
      k_apply_events, dim3(blocks), dim3(waves_per_block * 64), 0, STREAM, v,
      tokens.data_ptr<int64_t>(), tok_off.data_ptr<int32_t>(), U64P(ehashes),
      eh_off.data_ptr<int32_t>(), U64P(parents),
      has_parent.data_ptr<uint8_t>(), ev_type.data_ptr<uint8_t>(),
      reinterpret_cast<const uint32_t*>(pod_entry.data_ptr<int32_t>()),
      grp_off.data_ptr<int32_t>(), model_of.data_ptr<int32_t>(), G,
      (uint64_t)init_hash_bits, (int)block_size, (int32_t)epoch,
      (int)shard_id, (int)num_shards,
      reinterpret_cast<uint64_t*>(req_scratch.data_ptr<int64_t>()));
}

// Transposed-chain phase-split apply: NO batch-local parents (host
// verified), so chains run one lane per EVENT from transposed int32
// tokens; inserts thread-per-block as usual.  No bmap is needed -
// parents resolve via the persistent engine map or the chain root.
void gpu_apply_events_split_tr(
    at::Tensor keys, at::Tensor meta, at::Tensor stamp, at::Tensor pods,
    at::Tensor e_keys, at::Tensor e_meta, at::Tensor e_vals,
    int64_t pods_per_key, at::Tensor tokens_flat, at::Tensor tok_off,
    at::Tensor ehashes, at::Tensor eh_off, at::Tensor parents,
    at::Tensor has_parent, at::Tensor ev_type, at::Tensor pod_entry,
    at::Tensor ev_of, at::Tensor model_of, int64_t init_hash_bits,
    int64_t block_size, int64_t epoch, int64_t shard_id,
    int64_t num_shards, int64_t max_tokens) {
  auto v = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                    pods_per_key);
  TORCH_CHECK(tokens_flat.dtype() == at::kInt,
              "event tokens must be int32 (uint32 bit pattern)");
  TORCH_CHECK(block_size <= 64, "event chain kernel supports block_size<=64");
  int64_t E = ev_type.numel();
  int64_t n = ehashes.numel();
  if (E == 0 || n == 0) return;
  TORCH_CHECK(max_tokens > 0, "max_tokens must be positive");
  auto tokens_t = at::empty({max_tokens, E},
                            tokens_flat.options());
  auto req_scratch = at::zeros({n}, ehashes.options());
  int threads = 256;
  {
    int ygrid = (int)std::min<int64_t>((max_tokens + threads - 1) / threads,
                                       64);
    hipLaunchKernelGGL(k_transpose_ev_tokens, dim3((int)E, ygrid),
                       dim3(threads), 0, STREAM,
                       tokens_flat.data_ptr<int32_t>(),
                       tok_off.data_ptr<int32_t>(), E,
                       tokens_t.data_ptr<int32_t>());
  }
  TORCH_CHECK(model_of.numel() == E, "model_of must have one id per event");
  hipLaunchKernelGGL(k_event_chains_ev,
                     dim3((int)((E + threads - 1) / threads)), dim3(threads),
                     0, STREAM, v, tokens_t.data_ptr<int32_t>(),
                     tok_off.data_ptr<int32_t>(), U64P(ehashes),
                     eh_off.data_ptr<int32_t>(), U64P(parents),
                     has_parent.data_ptr<uint8_t>(),
                     ev_type.data_ptr<uint8_t>(),
                     model_of.data_ptr<int32_t>(), E,
                     (uint64_t)init_hash_bits, (int)block_size,
                     reinterpret_cast<uint64_t*>(
                         req_scratch.data_ptr<int64_t>()));
  hipLaunchKernelGGL(k_event_inserts,
                     dim3((int)((n + threads - 1) / threads)), dim3(threads),
                     0, STREAM, v, U64P(ehashes), eh_off.data_ptr<int32_t>(),
                     ev_of.data_ptr<int32_t>(), ev_type.data_ptr<uint8_t>(),
                     tok_off.data_ptr<int32_t>(),
                     reinterpret_cast<const uint32_t*>(
                         pod_entry.data_ptr<int32_t>()),
                     model_of.data_ptr<int32_t>(), n,
                     (int)block_size, (int32_t)epoch,
                     (int)shard_id, (int)num_shards,
                     reinterpret_cast<uint64_t*>(
                         req_scratch.data_ptr<int64_t>()));
}

// Phase-split apply: bmap build -> chains (lane per group) -> inserts
// (thread per block).  ev_of: int32 [total blocks] = owning event index.
void gpu_apply_events_split(
    at::Tensor keys, at::Tensor meta, at::Tensor stamp, at::Tensor pods,
    at::Tensor e_keys, at::Tensor e_meta, at::Tensor e_vals,
    int64_t pods_per_key, at::Tensor tokens, at::Tensor tok_off,
    at::Tensor ehashes, at::Tensor eh_off, at::Tensor parents,
    at::Tensor has_parent, at::Tensor ev_type, at::Tensor pod_entry,
    at::Tensor grp_off, at::Tensor ev_of, at::Tensor model_of,
    int64_t init_hash_bits, int64_t block_size, int64_t epoch,
    int64_t shard_id, int64_t num_shards) {
  auto v = dev_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                    pods_per_key);
  int64_t G = grp_off.numel() - 1;
  int64_t n = ehashes.numel();
  if (G == 0 || n == 0) return;
  int64_t bcap = 4;
  while (bcap < n * 2) bcap <<= 1;
  auto bkeys = at::zeros({bcap}, ehashes.options());
  auto bvals = at::zeros({bcap}, ehashes.options().dtype(at::kInt));
  auto req_scratch = at::zeros({n}, ehashes.options());
  int threads = 256;

  hipLaunchKernelGGL(k_batch_bmap,
                     dim3((int)((n + threads - 1) / threads)), dim3(threads),
                     0, STREAM, U64P(ehashes), n,
                     reinterpret_cast<uint64_t*>(bkeys.data_ptr<int64_t>()),
                     bvals.data_ptr<int32_t>(), bcap - 1);
  TORCH_CHECK(model_of.numel() == ev_type.numel(),
              "model_of must have one id per event");
  hipLaunchKernelGGL(k_event_chains,
                     dim3((int)((G + threads - 1) / threads)), dim3(threads),
                     0, STREAM, v, tokens.data_ptr<int64_t>(),
                     tok_off.data_ptr<int32_t>(), U64P(ehashes),
                     eh_off.data_ptr<int32_t>(), U64P(parents),
                     has_parent.data_ptr<uint8_t>(),
                     ev_type.data_ptr<uint8_t>(), grp_off.data_ptr<int32_t>(),
                     model_of.data_ptr<int32_t>(),
                     G, (uint64_t)init_hash_bits,
                     (int)block_size,
                     reinterpret_cast<uint64_t*>(bkeys.data_ptr<int64_t>()),
                     bvals.data_ptr<int32_t>(), bcap - 1,
                     reinterpret_cast<uint64_t*>(
                         req_scratch.data_ptr<int64_t>()));
  hipLaunchKernelGGL(k_event_inserts,
                     dim3((int)((n + threads - 1) / threads)), dim3(threads),
                     0, STREAM, v, U64P(ehashes), eh_off.data_ptr<int32_t>(),
                     ev_of.data_ptr<int32_t>(), ev_type.data_ptr<uint8_t>(),
                     tok_off.data_ptr<int32_t>(),
                     reinterpret_cast<const uint32_t*>(
                         pod_entry.data_ptr<int32_t>()),
                     model_of.data_ptr<int32_t>(), n,
                     (int)block_size, (int32_t)epoch,
                     (int)shard_id, (int)num_shards,
                     reinterpret_cast<uint64_t*>(
                         req_scratch.data_ptr<int64_t>()));
}

}  // namespace kvidx
