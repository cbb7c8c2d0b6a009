// kvidx_common.h - shared table layout + hashing for the KV-block index.
//
// The block->pod index is an open-addressing hash table designed to live in
// MI355X HBM3E (288 GB/GPU) and be probed by wave-cooperative HIP kernels;
// the exact same layout is operated on by the CPU reference implementation
// (cpu_ops.cpp) so CPU<->GPU differential tests are bit-exact.
//
// Behavioral parity notes (reference: /root/reference, Go):
//  - chunk hash = FNV-64a(canonical CBOR [parent, chunk, null])
//    (pkg/kvcache/kvblock/token_processor.go:94-112)
//  - dual keys: engine map (engine hash -> request hash) feeds eviction and
//    parent-chain stitching (pkg/kvcache/kvblock/in_memory.go:159-167)
//  - per-key pod set with bounded capacity (default 10,
//    in_memory.go:32-35); approximate LRU via epoch stamps.
//
// Layout (structure-of-arrays, all torch tensors):
//   keys  : int64 [C]    chunk hash; 0 = never-used (free). A real hash of
//                        0 is remapped to 1 (KVIDX_REMAP).
//   meta  : int32 [C]    bit31 OCC, bit30 TOMB, bits 0..15 model_id
//   stamp : int32 [C]    last-touch epoch (approximate LRU)
//   pods  : int32 [C*P]  pod entries; 0 = empty,
//                        else (tier<<24) | (pod_id+1)  (pod_id < 2^24-1)
//   e_keys/e_meta/e_vals: the engine->request map (same C, e_vals holds the
//                        request hash)

#pragma once

#include <cstdint>

// Debug asserts (ROADMAP r1 #10): compile with KVIDX_DEBUG=1 in the
// environment (setup.py adds -DKVIDX_DEBUG_ASSERTS) to trap on
// violated kernel/table invariants - a hipMemcheck-style guard with
// zero cost in release builds.
#ifdef KVIDX_DEBUG_ASSERTS
#if defined(__HIP_DEVICE_COMPILE__)
#define KVIDX_ASSERT(c) \
  do {                  \
    if (!(c)) __builtin_trap(); \
  } while (0)
#else
#include <cassert>
#define KVIDX_ASSERT(c) assert(c)
#endif
#else
#define KVIDX_ASSERT(c) ((void)0)
#endif

#if defined(__HIPCC__)
#define KVIDX_HD __host__ __device__ __forceinline__
#else
#define KVIDX_HD inline
#endif

namespace kvidx {

static constexpr uint64_t FNV64_OFFSET = 0xCBF29CE484222325ull;
static constexpr uint64_t FNV64_PRIME = 0x100000001B3ull;

static constexpr uint32_t META_OCC = 0x80000000u;
static constexpr uint32_t META_TOMB = 0x40000000u;
static constexpr uint32_t META_MODEL_MASK = 0xFFFFu;

static constexpr int MAX_TIERS = 4;
static constexpr int PROBE_MAX = 128;

KVIDX_HD uint64_t remap_hash(uint64_t h) { return h == 0 ? 1ull : h; }

KVIDX_HD uint64_t kvidx_checked_slot(uint64_t i, uint64_t cap_mask) {
  KVIDX_ASSERT(i <= cap_mask);
  return i;
}

KVIDX_HD uint64_t fnv1a_64_byte(uint64_t h, uint8_t b) {
  return (h ^ (uint64_t)b) * FNV64_PRIME;
}

// Canonical-CBOR unsigned integer append; returns new length.
KVIDX_HD int cbor_put_uint(uint8_t* buf, int len, uint64_t v, uint8_t major) {
  const uint8_t mt = major << 5;
  if (v < 24) {
    buf[len++] = mt | (uint8_t)v;
  } else if (v <= 0xFF) {
    buf[len++] = mt | 24;
    buf[len++] = (uint8_t)v;
  } else if (v <= 0xFFFF) {
    buf[len++] = mt | 25;
    buf[len++] = (uint8_t)(v >> 8);
    buf[len++] = (uint8_t)v;
  } else if (v <= 0xFFFFFFFFull) {
    buf[len++] = mt | 26;
    buf[len++] = (uint8_t)(v >> 24);
    buf[len++] = (uint8_t)(v >> 16);
    buf[len++] = (uint8_t)(v >> 8);
    buf[len++] = (uint8_t)v;
  } else {
    buf[len++] = mt | 27;
    for (int s = 56; s >= 0; s -= 8) buf[len++] = (uint8_t)(v >> s);
  }
  return len;
}

// Streaming canonical-CBOR unsigned integer fed straight into FNV-64a -
// no byte buffer, so GPU lanes keep everything in registers (no scratch).
KVIDX_HD uint64_t fnv_cbor_uint(uint64_t h, uint64_t v, uint8_t major) {
  const uint8_t mt = major << 5;
  if (v < 24) {
    h = fnv1a_64_byte(h, mt | (uint8_t)v);
  } else if (v <= 0xFF) {
    h = fnv1a_64_byte(h, mt | 24);
    h = fnv1a_64_byte(h, (uint8_t)v);
  } else if (v <= 0xFFFF) {
    h = fnv1a_64_byte(h, mt | 25);
    h = fnv1a_64_byte(h, (uint8_t)(v >> 8));
    h = fnv1a_64_byte(h, (uint8_t)v);
  } else if (v <= 0xFFFFFFFFull) {
    h = fnv1a_64_byte(h, mt | 26);
    for (int s = 24; s >= 0; s -= 8) h = fnv1a_64_byte(h, (uint8_t)(v >> s));
  } else {
    h = fnv1a_64_byte(h, mt | 27);
    for (int s = 56; s >= 0; s -= 8) h = fnv1a_64_byte(h, (uint8_t)(v >> s));
  }
  return h;
}

// Branchless (predicated) variants: identical output to fnv_cbor_uint but
// straight-line code - on the GPU this lets the scheduler interleave
// several independent chains' instruction streams (ILP hash kernels) and
// removes lane divergence on data-dependent encoding lengths.
KVIDX_HD uint64_t fnv_cbor_u32_branchless(uint64_t h, uint32_t v,
                                          uint8_t major) {
  const uint8_t mt = major << 5;
  const int sel = (v >= 24) + (v > 0xFF) + (v > 0xFFFF);  // 0..3
  const int n = sel == 3 ? 4 : sel;  // value bytes
  const uint8_t header = sel ? (uint8_t)(mt | (0x17 + sel))
                             : (uint8_t)(mt | v);
  h = fnv1a_64_byte(h, header);
#if defined(__HIP_DEVICE_COMPILE__)
#pragma unroll
#endif
  for (int k = 0; k < 4; ++k) {
    const uint8_t b = (uint8_t)(v >> (8 * (3 - k)));
    const uint64_t h2 = fnv1a_64_byte(h, b);
    h = (k >= 4 - n) ? h2 : h;
  }
  return h;
}

KVIDX_HD uint64_t fnv_cbor_u64_branchless(uint64_t h, uint64_t v,
                                          uint8_t major) {
  const uint8_t mt = major << 5;
  const int sel = (v >= 24) + (v > 0xFF) + (v > 0xFFFF) +
                  (v > 0xFFFFFFFFull);  // 0..4
  const int n = sel == 0 ? 0 : (1 << (sel - 1));  // 0,1,2,4,8 value bytes
  const uint8_t header = sel ? (uint8_t)(mt | (0x17 + sel))
                             : (uint8_t)(mt | v);
  h = fnv1a_64_byte(h, header);
#if defined(__HIP_DEVICE_COMPILE__)
#pragma unroll
#endif
  for (int k = 0; k < 8; ++k) {
    const uint8_t b = (uint8_t)(v >> (8 * (7 - k)));
    const uint64_t h2 = fnv1a_64_byte(h, b);
    h = (k >= 8 - n) ? h2 : h;
  }
  return h;
}

// Branchless chain link (bit-identical to chunk_hash; GPU hot path).
// The token loop is force-unrolled on device: a rolled loop makes hipcc
// read tok[] via gpr_idx (register-indexed) with a vmcnt(0) wait at the
// loop head, which both serializes the reads and drains any prefetched
// next-chunk loads (measured in the k_hash_chain_tr ISA).
template <typename TokT>
KVIDX_HD uint64_t chunk_hash_fast(uint64_t parent, const TokT* tokens,
                                  int n) {
  uint64_t h = FNV64_OFFSET;
  h = fnv1a_64_byte(h, 0x83);
  h = fnv_cbor_u64_branchless(h, parent, 0);
  h = fnv_cbor_u32_branchless(h, (uint32_t)n, 4);
#if defined(__HIP_DEVICE_COMPILE__)
#pragma unroll
#endif
  for (int i = 0; i < n; ++i)
    h = fnv_cbor_u32_branchless(h, (uint32_t)tokens[i], 0);
  h = fnv1a_64_byte(h, 0xF6);
  return h;
}

// One chain link: FNV-64a(CBOR([parent, tokens[0..n), null])), streamed.
template <typename TokT>
KVIDX_HD uint64_t chunk_hash(uint64_t parent, const TokT* tokens, int n) {
  uint64_t h = FNV64_OFFSET;
  h = fnv1a_64_byte(h, 0x83);                    // array(3)
  h = fnv_cbor_uint(h, parent, 0);
  h = fnv_cbor_uint(h, (uint64_t)n, 4);          // tokens array header
  for (int i = 0; i < n; ++i) h = fnv_cbor_uint(h, (uint64_t)(uint32_t)tokens[i], 0);
  h = fnv1a_64_byte(h, 0xF6);                    // null
  return h;
}

KVIDX_HD uint32_t make_pod_entry(uint32_t pod_id, uint32_t tier) {
  return ((tier & 0xFu) << 24) | ((pod_id + 1) & 0x00FFFFFFu);
}

KVIDX_HD uint32_t pod_entry_id(uint32_t e) { return (e & 0x00FFFFFFu) - 1; }
KVIDX_HD uint32_t pod_entry_tier(uint32_t e) { return (e >> 24) & 0xFu; }

// Probe start position (golden-ratio scramble so clustered hashes spread).
KVIDX_HD uint64_t probe_start(uint64_t h, uint64_t cap_mask) {
  return (h * 0x9E3779B97F4A7C15ull) & cap_mask;
}

}  // namespace kvidx
