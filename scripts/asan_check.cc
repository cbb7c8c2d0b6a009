// Standalone ASan/UBSan harness for the shared table/hashing header
// (ops/csrc/kvidx_common.h) - the code every CPU op and GPU kernel
// builds on.  torch-free so it compiles with plain g++ sanitizers:
//
//   make asan   (scripts via Makefile; exits nonzero on any finding)
//
// Exercises: CBOR shortest-form encoding across every width boundary,
// branchless-vs-reference chain equality on random streams, pod-entry
// packing round trips, and remap/probe helpers - under
// -fsanitize=address,undefined so buffer overflows, shift UB and
// integer UB in the hashing core cannot hide.
#include <cassert>
#include <cstdio>
#include <cstdlib>
#include <random>
#include <vector>

#include "../llmd_kvcache_amd/ops/csrc/kvidx_common.h"

using namespace kvidx;

int main() {
  std::mt19937_64 rng(12345);

  // 1. CBOR width boundaries through both encode paths
  const uint64_t edges[] = {0,    1,    23,   24,   255,  256,
                            65535, 65536, 0xFFFFFFFFull, 0x100000000ull,
                            0xFFFFFFFFFFFFFFFFull};
  for (uint64_t v : edges) {
    for (uint8_t major : {(uint8_t)0, (uint8_t)4}) {
      uint64_t a = fnv_cbor_uint(FNV64_OFFSET, v, major);
      uint64_t b = fnv_cbor_u64_branchless(FNV64_OFFSET, v, major);
      if (a != b) {
        fprintf(stderr, "cbor mismatch v=%llu major=%u\n",
                (unsigned long long)v, major);
        return 1;
      }
      if (v <= 0xFFFFFFFFull) {
        uint64_t c = fnv_cbor_u32_branchless(FNV64_OFFSET, (uint32_t)v,
                                             major);
        if (a != c) {
          fprintf(stderr, "cbor32 mismatch v=%llu\n",
                  (unsigned long long)v);
          return 1;
        }
      }
    }
  }

  // 2. chain equality: branchless fast path vs reference, random
  //    streams at every supported block size
  for (int bs : {4, 8, 16, 32, 64}) {
    for (int iter = 0; iter < 2000; ++iter) {
      std::vector<uint32_t> toks(bs);
      for (auto& t : toks) t = (uint32_t)rng();
      uint64_t parent = rng();
      uint64_t a = chunk_hash(parent, toks.data(), bs);
      uint64_t b = chunk_hash_fast(parent, toks.data(), bs);
      if (a != b) {
        fprintf(stderr, "chain mismatch bs=%d iter=%d\n", bs, iter);
        return 1;
      }
    }
  }

  // 3. pod entry packing round trips (incl. max ids)
  for (int iter = 0; iter < 100000; ++iter) {
    uint32_t pid = (uint32_t)(rng() % (0x00FFFFFEu));
    uint32_t tier = (uint32_t)(rng() % MAX_TIERS);
    uint32_t e = make_pod_entry(pid, tier);
    if (pod_entry_id(e) != pid || pod_entry_tier(e) != tier) {
      fprintf(stderr, "pod entry round-trip failed\n");
      return 1;
    }
  }

  // 4. remap + probe helpers over the full hash range
  for (int iter = 0; iter < 100000; ++iter) {
    uint64_t h = rng();
    uint64_t cap_mask = (1ull << (6 + iter % 20)) - 1;
    uint64_t s = probe_start(remap_hash(h), cap_mask);
    if (s > cap_mask) {
      fprintf(stderr, "probe_start out of range\n");
      return 1;
    }
  }
  assert(remap_hash(0) == 1);

  printf("asan_check OK\n");
  return 0;
}
