// cpu_ops.cpp - CPU reference implementation of the kvidx table ops.
//
// Operates on the exact same tensor layout as the HIP kernels
// (kvidx_common.h) so CPU<->GPU differential tests are bit-exact, and
// serves as the fast CPU backend (the pure-Python in-memory index is the
// behavioral reference; this is the performance path on CPU-only hosts).
// Thread-safety is provided by the Python-side caller (one writer at a
// time); the GPU path uses atomics instead (hip_ops.hip).

#include <torch/extension.h>

#include <cstring>
#include <vector>

#include "kvidx_common.h"

namespace kvidx {

struct TableView {
  uint64_t* keys;
  uint32_t* meta;
  int32_t* stamp;
  uint32_t* pods;
  uint64_t* e_keys;
  uint32_t* e_meta;
  uint64_t* e_vals;
  uint64_t cap_mask;
  int64_t capacity;
  int pods_per_key;
};

static TableView make_view(at::Tensor& keys, at::Tensor& meta,
                           at::Tensor& stamp, at::Tensor& pods,
                           at::Tensor& e_keys, at::Tensor& e_meta,
                           at::Tensor& e_vals, int pods_per_key) {
  TORCH_CHECK(keys.is_contiguous() && meta.is_contiguous() &&
                  pods.is_contiguous(),
              "table tensors must be contiguous");
  int64_t cap = keys.numel();
  TORCH_CHECK((cap & (cap - 1)) == 0, "capacity must be a power of two");
  TableView v;
  v.keys = reinterpret_cast<uint64_t*>(keys.data_ptr<int64_t>());
  v.meta = reinterpret_cast<uint32_t*>(meta.data_ptr<int32_t>());
  v.stamp = stamp.data_ptr<int32_t>();
  v.pods = reinterpret_cast<uint32_t*>(pods.data_ptr<int32_t>());
  v.e_keys = reinterpret_cast<uint64_t*>(e_keys.data_ptr<int64_t>());
  v.e_meta = reinterpret_cast<uint32_t*>(e_meta.data_ptr<int32_t>());
  v.e_vals = reinterpret_cast<uint64_t*>(e_vals.data_ptr<int64_t>());
  v.cap_mask = (uint64_t)cap - 1;
  v.capacity = cap;
  v.pods_per_key = pods_per_key;
  return v;
}

// Probe for an existing live key. Returns slot or -1.
static int64_t table_find(const TableView& v, uint64_t h, uint32_t model) {
  h = remap_hash(h);
  uint64_t s = probe_start(h, v.cap_mask);
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t i = (s + t) & v.cap_mask;
    uint64_t k = v.keys[i];
    if (k == 0) return -1;  // never-used: chain ends
    uint32_t m = v.meta[i];
    if ((m & META_OCC) && !(m & META_TOMB) && k == h &&
        (m & META_MODEL_MASK) == model)
      return (int64_t)i;
  }
  return -1;
}

// Probe-or-insert. Window-full policy: reuse first tombstone, else
// overwrite the lowest-stamp live slot (approximate LRU eviction, the
// CPU analog of the reference's LRU-under-pressure, in_memory.go:32-35).
static int64_t table_put(const TableView& v, uint64_t h, uint32_t model,
                         int32_t epoch) {
  h = remap_hash(h);
  uint64_t s = probe_start(h, v.cap_mask);
  int64_t first_tomb = -1;
  int64_t victim = -1;
  int32_t victim_stamp = 0;
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t i = (s + t) & v.cap_mask;
    uint64_t k = v.keys[i];
    uint32_t m = v.meta[i];
    if (k == 0) {  // free: claim
      if (first_tomb >= 0) { i = (uint64_t)first_tomb; }
      v.keys[i] = h;
      std::memset(v.pods + i * v.pods_per_key, 0,
                  sizeof(uint32_t) * v.pods_per_key);
      v.meta[i] = META_OCC | (model & META_MODEL_MASK);
      v.stamp[i] = epoch;
      return (int64_t)i;
    }
    if ((m & META_OCC) && !(m & META_TOMB)) {
      if (k == h && (m & META_MODEL_MASK) == model) {
        v.stamp[i] = epoch;
        return (int64_t)i;
      }
      if (victim < 0 || v.stamp[i] < victim_stamp) {
        victim = (int64_t)i;
        victim_stamp = v.stamp[i];
      }
    } else if ((m & META_TOMB) && first_tomb < 0) {
      first_tomb = (int64_t)i;
    }
  }
  int64_t i = first_tomb >= 0 ? first_tomb : victim;
  if (i < 0) i = (int64_t)s;
  v.keys[i] = h;
  std::memset(v.pods + i * v.pods_per_key, 0,
              sizeof(uint32_t) * v.pods_per_key);
  v.meta[i] = META_OCC | (model & META_MODEL_MASK);
  v.stamp[i] = epoch;
  return i;
}

static void pod_set_add(const TableView& v, int64_t slot, uint32_t entry,
                        int32_t epoch) {
  uint32_t* p = v.pods + slot * v.pods_per_key;
  for (int j = 0; j < v.pods_per_key; ++j)
    if (p[j] == entry) return;
  for (int j = 0; j < v.pods_per_key; ++j)
    if (p[j] == 0) {
      p[j] = entry;
      return;
    }
  p[(uint32_t)epoch % v.pods_per_key] = entry;  // full: pseudo-LRU overwrite
}

// ---- engine map (same probing, value payload) ------------------------

static int64_t emap_put(const TableView& v, uint64_t h, uint32_t model,
                        uint64_t val, int32_t epoch) {
  h = remap_hash(h);
  uint64_t s = probe_start(h, v.cap_mask);
  int64_t first_tomb = -1;
  int64_t victim = -1;
  int32_t victim_stamp = 0;
  for (int t = 0; t < PROBE_MAX; ++t) {
    uint64_t i = (s + t) & v.cap_mask;
    uint64_t k = v.e_keys[i];
    uint32_t m = v.e_meta[i];
    if (k == 0) {
      if (first_tomb >= 0) i = (uint64_t)first_tomb;
      v.e_keys[i] = h;
      v.e_vals[i] = val;
      v.e_meta[i] = META_OCC | (model & META_MODEL_MASK);
      return (int64_t)i;
    }
    if ((m & META_OCC) && !(m & META_TOMB)) {
      if (k == h && (m & META_MODEL_MASK) == model) {
        v.e_vals[i] = val;
        return (int64_t)i;
      }
      if (victim < 0) victim = (int64_t)i;  // first as victim (no stamps)
      (void)victim_stamp;
    } else if ((m & META_TOMB) && first_tomb < 0) {
      first_tomb = (int64_t)i;
    }
  }
  int64_t i = first_tomb >= 0 ? first_tomb : victim;
  if (i < 0) i = (int64_t)s;
  v.e_keys[i] = h;
  v.e_vals[i] = val;
  v.e_meta[i] = META_OCC | (model & META_MODEL_MASK);
  (void)epoch;
  return i;
}

static int64_t emap_find(const TableView& v, uint64_t h, uint32_t model) {
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

// ---- public ops ------------------------------------------------------

std::vector<uint64_t> tokens_to_chunk_hashes(std::vector<uint64_t> tokens,
                                             uint64_t parent,
                                             int64_t block_size) {
  TORCH_CHECK(block_size > 0 && block_size <= 256,
              "block_size must be in (0, 256]");
  std::vector<uint32_t> toks(tokens.begin(), tokens.end());
  int64_t n_chunks = (int64_t)toks.size() / block_size;
  std::vector<uint64_t> out;
  out.reserve(n_chunks);
  uint64_t h = parent;
  for (int64_t c = 0; c < n_chunks; ++c) {
    h = chunk_hash(h, toks.data() + c * block_size, (int)block_size);
    out.push_back(h);
  }
  return out;
}

// Branchless-variant twin of tokens_to_chunk_hashes - exists so tests
// can verify chunk_hash_fast (the GPU hot path) == chunk_hash on every
// CBOR length-boundary case without a GPU.
std::vector<uint64_t> tokens_to_chunk_hashes_fast(
    std::vector<uint64_t> tokens, uint64_t parent, int64_t block_size) {
  std::vector<uint32_t> toks(tokens.begin(), tokens.end());
  int64_t n_chunks = (int64_t)toks.size() / block_size;
  std::vector<uint64_t> out;
  out.reserve(n_chunks);
  uint64_t h = parent;
  for (int64_t c = 0; c < n_chunks; ++c) {
    h = chunk_hash_fast(h, toks.data() + c * block_size, (int)block_size);
    out.push_back(h);
  }
  return out;
}

// Batched chain hashing: tokens (int64 [total]), offsets (int64 [B+1]),
// parents (int64 [B], u64 bits) -> per-prompt chunk hashes, flat, with
// chunk-count offsets returned alongside.
std::vector<at::Tensor> hash_chain_batch(at::Tensor tokens, at::Tensor offsets,
                                         at::Tensor parents,
                                         int64_t block_size) {
  TORCH_CHECK(tokens.dtype() == at::kLong && offsets.dtype() == at::kLong &&
              parents.dtype() == at::kLong);
  auto tok = tokens.contiguous();
  auto off = offsets.contiguous();
  auto par = parents.contiguous();
  int64_t B = par.numel();
  const int64_t* offp = off.data_ptr<int64_t>();
  const int64_t* tokp = tok.data_ptr<int64_t>();
  const uint64_t* parp = reinterpret_cast<uint64_t*>(par.data_ptr<int64_t>());

  auto chunk_off = at::zeros({B + 1}, tokens.options());
  int64_t* cop = chunk_off.data_ptr<int64_t>();
  for (int64_t b = 0; b < B; ++b)
    cop[b + 1] = cop[b] + (offp[b + 1] - offp[b]) / block_size;
  auto out = at::empty({cop[B]}, tokens.options());
  uint64_t* outp = reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>());

  at::parallel_for(0, B, 1, [&](int64_t begin, int64_t end) {
    std::vector<uint32_t> buf(block_size);
    for (int64_t b = begin; b < end; ++b) {
      uint64_t h = parp[b];
      int64_t n_chunks = cop[b + 1] - cop[b];
      const int64_t* t0 = tokp + offp[b];
      for (int64_t c = 0; c < n_chunks; ++c) {
        for (int64_t j = 0; j < block_size; ++j)
          buf[j] = (uint32_t)t0[c * block_size + j];
        h = chunk_hash(h, buf.data(), (int)block_size);
        outp[cop[b] + c] = h;
      }
    }
  });
  return {out, chunk_off};
}

// Tombstone compaction (ROADMAP #9): rehash every live entry of `old`
// into the freshly-zeroed `nw` bundle (optionally a different power-of-
// two capacity). Stamps (approximate-LRU epochs) and pod rows are
// carried over verbatim; tombstones and dead probe windows disappear.
void cpu_compact(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
                 at::Tensor pods, at::Tensor e_keys, at::Tensor e_meta,
                 at::Tensor e_vals, int64_t pods_per_key,
                 at::Tensor n_keys, at::Tensor n_meta, at::Tensor n_stamp,
                 at::Tensor n_pods, at::Tensor n_e_keys, at::Tensor n_e_meta,
                 at::Tensor n_e_vals) {
  auto ov = make_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                      (int)pods_per_key);
  auto nv = make_view(n_keys, n_meta, n_stamp, n_pods, n_e_keys, n_e_meta,
                      n_e_vals, (int)pods_per_key);
  int64_t cap = keys.numel();
  for (int64_t i = 0; i < cap; ++i) {
    uint32_t m = ov.meta[i];
    if (!(m & META_OCC) || (m & META_TOMB)) continue;
    int64_t s2 = table_put(nv, (uint64_t)ov.keys[i], m & META_MODEL_MASK,
                           ov.stamp[i]);
    std::memcpy(nv.pods + s2 * nv.pods_per_key,
                ov.pods + i * ov.pods_per_key,
                sizeof(uint32_t) * nv.pods_per_key);
  }
  for (int64_t i = 0; i < cap; ++i) {
    uint32_t m = ov.e_meta[i];
    if (!(m & META_OCC) || (m & META_TOMB)) continue;
    emap_put(nv, (uint64_t)ov.e_keys[i], m & META_MODEL_MASK,
             (uint64_t)ov.e_vals[i], 0);
  }
}

// shard_id/num_shards: the main table stores only keys it owns
// (request_hash % num_shards == shard_id); the engine->request map is
// replicated on every shard so parent-chain stitching never needs a
// cross-rank lookup (see parallel/sharded.py).
void cpu_insert(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
                at::Tensor pods, at::Tensor e_keys, at::Tensor e_meta,
                at::Tensor e_vals, int64_t pods_per_key,
                at::Tensor engine_hashes, at::Tensor request_hashes,
                int64_t model_id, at::Tensor pod_entries, int64_t epoch,
                int64_t shard_id, int64_t num_shards, int64_t emap_write) {
  auto v = make_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                     (int)pods_per_key);
  auto eh = engine_hashes.contiguous();
  auto rh = request_hashes.contiguous();
  auto pe = pod_entries.contiguous();
  const uint64_t* ehp = reinterpret_cast<uint64_t*>(eh.data_ptr<int64_t>());
  const uint64_t* rhp = reinterpret_cast<uint64_t*>(rh.data_ptr<int64_t>());
  const uint32_t* pep = reinterpret_cast<uint32_t*>(pe.data_ptr<int32_t>());
  int64_t n = eh.numel();
  int64_t m = pe.numel();
  TORCH_CHECK(rh.numel() == n, "engine/request key length mismatch");
  for (int64_t i = 0; i < n; ++i) {
    if (emap_write)
      emap_put(v, ehp[i], (uint32_t)model_id, remap_hash(rhp[i]),
               (int32_t)epoch);
    if (num_shards > 1 &&
        (int64_t)(remap_hash(rhp[i]) % (uint64_t)num_shards) != shard_id)
      continue;
    int64_t slot = table_put(v, rhp[i], (uint32_t)model_id, (int32_t)epoch);
    for (int64_t j = 0; j < m; ++j)
      pod_set_add(v, slot, pep[j], (int32_t)epoch);
  }
}

// Walk precomputed (possibly all-reduce-merged) masks -> scores; CPU twin
// of k_score_from_masks for gloo-backed multi-process tests.
at::Tensor cpu_score_from_masks(at::Tensor masks, at::Tensor offsets,
                                at::Tensor weights, int64_t num_pods) {
  auto m = masks.contiguous();
  auto off = offsets.contiguous();
  TORCH_CHECK(m.dim() == 3, "masks must be [Ktot, T, W]");
  const int64_t T = m.size(1);  // tier planes (sized by caller)
  const uint64_t* mp = reinterpret_cast<uint64_t*>(m.data_ptr<int64_t>());
  const int32_t* op = off.data_ptr<int32_t>();
  const float* wts = weights.data_ptr<float>();
  int64_t B = off.numel() - 1;
  int64_t W = (num_pods + 63) / 64;
  TORCH_CHECK(W <= 64, "num_pods > 4096 not supported");
  auto scores = at::zeros({B, num_pods}, at::kFloat);
  float* sp = scores.data_ptr<float>();
  at::parallel_for(0, B, 1, [&](int64_t begin, int64_t end) {
    for (int64_t b = begin; b < end; ++b) {
      uint64_t active[64];
      float* out = sp + b * num_pods;
      int64_t K = op[b + 1] - op[b];
      for (int64_t k = 0; k < K; ++k) {
        const uint64_t* mk = mp + (size_t)(op[b] + k) * T * W;
        bool any = false;
        for (int64_t w = 0; w < W; ++w) {
          uint64_t cur = 0;
          for (int t = 0; t < T; ++t) cur |= mk[t * W + w];
          uint64_t act = (k == 0) ? cur : (active[w] & cur);
          active[w] = act;
          if (act) any = true;
          uint64_t bits = act;
          while (bits) {
            int bit = __builtin_ctzll(bits);
            bits &= bits - 1;
            float wmax = 0.f;
            for (int t = 0; t < T; ++t)
              if ((mk[t * W + w] >> bit) & 1) wmax = std::max(wmax, wts[t]);
            out[w * 64 + bit] += wmax;
          }
        }
        if (!any) break;
      }
    }
  });
  return scores;
}

void cpu_evict(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
               at::Tensor pods, at::Tensor e_keys, at::Tensor e_meta,
               at::Tensor e_vals, int64_t pods_per_key,
               at::Tensor engine_hashes, int64_t model_id,
               at::Tensor pod_entries) {
  auto v = make_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                     (int)pods_per_key);
  auto eh = engine_hashes.contiguous();
  auto pe = pod_entries.contiguous();
  const uint64_t* ehp = reinterpret_cast<uint64_t*>(eh.data_ptr<int64_t>());
  const uint32_t* pep = reinterpret_cast<uint32_t*>(pe.data_ptr<int32_t>());
  // Engine->request mappings are deliberately RETAINED on eviction
  // (unlike in_memory.go:252-255): with the sharded deployment the emap
  // is replicated while the main table is ownership-filtered, so only
  // the owning shard can observe "pod set now empty" - dropping the
  // mapping locally would diverge the replicas and break parent-chain
  // stitching after removals.  A retained mapping also keeps child
  // chains content-consistent (the reference restarts them from the
  // root); stale entries are reclaimed by the emap's own slot-steal.
  for (int64_t i = 0; i < eh.numel(); ++i) {
    int64_t ei = emap_find(v, ehp[i], (uint32_t)model_id);
    if (ei < 0) continue;
    uint64_t req = v.e_vals[ei];
    int64_t slot = table_find(v, req, (uint32_t)model_id);
    if (slot < 0) continue;
    uint32_t* p = v.pods + slot * v.pods_per_key;
    for (int64_t j = 0; j < pe.numel(); ++j)
      for (int k = 0; k < v.pods_per_key; ++k)
        if (p[k] == pep[j]) p[k] = 0;
    bool empty = true;
    for (int k = 0; k < v.pods_per_key; ++k)
      if (p[k] != 0) { empty = false; break; }
    if (empty) v.meta[slot] |= META_TOMB;
  }
}

// found: 0 absent, 1 present-with-visible-pods, 2 present-but-empty
// (chain-cut marker, in_memory.go:118-121).
std::vector<at::Tensor> cpu_lookup(at::Tensor keys, at::Tensor meta,
                                   at::Tensor stamp, at::Tensor pods,
                                   at::Tensor e_keys, at::Tensor e_meta,
                                   at::Tensor e_vals, int64_t pods_per_key,
                                   at::Tensor request_hashes, int64_t model_id,
                                   at::Tensor filter_words, int64_t num_pods,
                                   int64_t epoch, int64_t shard_id,
                                   int64_t num_shards,
                                   int64_t n_tiers) {
  auto v = make_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                     (int)pods_per_key);
  auto rh = request_hashes.contiguous();
  const uint64_t* rhp = reinterpret_cast<uint64_t*>(rh.data_ptr<int64_t>());
  int64_t K = rh.numel();
  int64_t W = (num_pods + 63) / 64;
  bool has_filter = filter_words.numel() > 0;
  const uint64_t* fw = has_filter ? reinterpret_cast<uint64_t*>(
                                        filter_words.data_ptr<int64_t>())
                                  : nullptr;

  if (n_tiers < 1) n_tiers = 1;
  if (n_tiers > MAX_TIERS) n_tiers = MAX_TIERS;
  auto found = at::zeros({K}, at::kByte);
  // masks sized by the REGISTERED tier count (usually 1-2): halves-to-
  // quarters the fallback path's mask traffic and the sharded
  // all_reduce bytes vs a fixed MAX_TIERS=4 plane count.
  auto masks = at::zeros({K, n_tiers, W}, at::kLong);
  uint8_t* fp = found.data_ptr<uint8_t>();
  uint64_t* mp = reinterpret_cast<uint64_t*>(masks.data_ptr<int64_t>());

  for (int64_t k = 0; k < K; ++k) {
    if (num_shards > 1 &&
        (int64_t)(remap_hash(rhp[k]) % (uint64_t)num_shards) != shard_id)
      continue;  // unowned key: another shard's mask contribution
    int64_t slot = table_find(v, rhp[k], (uint32_t)model_id);
    if (slot < 0) continue;
    v.stamp[slot] = (int32_t)epoch;  // LRU touch on read
    const uint32_t* p = v.pods + slot * v.pods_per_key;
    bool any_raw = false, any_visible = false;
    for (int j = 0; j < v.pods_per_key; ++j) {
      uint32_t e = p[j];
      if (e == 0) continue;
      any_raw = true;
      uint32_t pid = pod_entry_id(e);
      uint32_t tier = pod_entry_tier(e);
      if (pid >= (uint32_t)num_pods || tier >= (uint32_t)n_tiers) continue;
      if (has_filter && !((fw[pid / 64] >> (pid % 64)) & 1)) continue;
      mp[(k * n_tiers + tier) * W + pid / 64] |= 1ull << (pid % 64);
      any_visible = true;
    }
    (void)any_visible;
    fp[k] = any_raw ? 1 : 2;  // 2 = present-but-empty (chain cut)
  }
  return {found, masks};
}

// Fused probe + longest-prefix score (the read-path hot loop, scoring
// parity with kvblock_scorer.go:108-151).
at::Tensor cpu_fused_score(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
                           at::Tensor pods, at::Tensor e_keys,
                           at::Tensor e_meta, at::Tensor e_vals,
                           int64_t pods_per_key, at::Tensor hashes,
                           at::Tensor counts, int64_t model_id,
                           at::Tensor filter_words, at::Tensor weights,
                           int64_t num_pods, int64_t epoch) {
  auto v = make_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                     (int)pods_per_key);
  auto h = hashes.contiguous();
  auto cnt = counts.contiguous();
  const uint64_t* hp = reinterpret_cast<uint64_t*>(h.data_ptr<int64_t>());
  const int32_t* cp = cnt.data_ptr<int32_t>();
  int64_t B = cnt.numel();
  int64_t W = (num_pods + 63) / 64;
  bool has_filter = filter_words.numel() > 0;
  const uint64_t* fw = has_filter ? reinterpret_cast<uint64_t*>(
                                        filter_words.data_ptr<int64_t>())
                                  : nullptr;
  const float* wts = weights.data_ptr<float>();

  TORCH_CHECK(W <= 64, "num_pods > 4096 not supported");
  auto scores = at::zeros({B, num_pods}, at::kFloat);
  float* sp = scores.data_ptr<float>();

  std::vector<int64_t> bases(B + 1, 0);
  for (int64_t b = 0; b < B; ++b) bases[b + 1] = bases[b] + cp[b];

  at::parallel_for(0, B, 1, [&](int64_t begin, int64_t end) {
    for (int64_t b = begin; b < end; ++b) {
      uint64_t tier_masks[MAX_TIERS * 64];
      uint64_t active[64];
      uint64_t cur[64];
      int64_t base = bases[b];
      int64_t K = cp[b];
      float* out = sp + b * num_pods;
      bool first = true;
      for (int64_t k = 0; k < K; ++k) {
        for (int64_t x = 0; x < MAX_TIERS * W; ++x) tier_masks[x] = 0;
        for (int64_t w = 0; w < W; ++w) cur[w] = 0;
        int64_t slot = table_find(v, hp[base + k], (uint32_t)model_id);
        if (slot >= 0) {
          v.stamp[slot] = (int32_t)epoch;
          const uint32_t* p = v.pods + slot * v.pods_per_key;
          for (int j = 0; j < v.pods_per_key; ++j) {
            uint32_t e = p[j];
            if (e == 0) continue;
            uint32_t pid = pod_entry_id(e);
            uint32_t tier = pod_entry_tier(e);
            if (pid >= (uint32_t)num_pods || tier >= MAX_TIERS) continue;
            if (has_filter && !((fw[pid / 64] >> (pid % 64)) & 1)) continue;
            tier_masks[tier * W + pid / 64] |= 1ull << (pid % 64);
            cur[pid / 64] |= 1ull << (pid % 64);
          }
        }
        if (first) {
          for (int64_t w = 0; w < W; ++w) active[w] = cur[w];
          first = false;
        } else {
          for (int64_t w = 0; w < W; ++w) active[w] &= cur[w];
        }
        bool any = false;
        for (int64_t w = 0; w < W; ++w) {
          uint64_t a = active[w];
          if (!a) continue;
          any = true;
          uint64_t bits = a;
          while (bits) {
            int bit = __builtin_ctzll(bits);
            bits &= bits - 1;
            int64_t pid = w * 64 + bit;
            float wmax = 0.f;
            for (int t = 0; t < MAX_TIERS; ++t)
              if ((tier_masks[t * W + w] >> bit) & 1)
                wmax = std::max(wmax, wts[t]);
            out[pid] += wmax;
          }
        }
        if (!any) break;
      }
    }
  });
  return scores;
}

std::vector<at::Tensor> cpu_get_request_keys(
    at::Tensor keys, at::Tensor meta, at::Tensor stamp, at::Tensor pods,
    at::Tensor e_keys, at::Tensor e_meta, at::Tensor e_vals,
    int64_t pods_per_key, at::Tensor engine_hashes, int64_t model_id) {
  auto v = make_view(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                     (int)pods_per_key);
  auto eh = engine_hashes.contiguous();
  const uint64_t* ehp = reinterpret_cast<uint64_t*>(eh.data_ptr<int64_t>());
  int64_t n = eh.numel();
  auto found = at::zeros({n}, at::kByte);
  auto out = at::zeros({n}, at::kLong);
  uint8_t* fp = found.data_ptr<uint8_t>();
  uint64_t* op = reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>());
  for (int64_t i = 0; i < n; ++i) {
    int64_t ei = emap_find(v, ehp[i], (uint32_t)model_id);
    if (ei >= 0) {
      fp[i] = 1;
      op[i] = v.e_vals[ei];
    }
  }
  return {found, out};
}

}  // namespace kvidx
