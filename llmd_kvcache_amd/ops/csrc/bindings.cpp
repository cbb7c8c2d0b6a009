// bindings.cpp - pybind11 module _kvidx_C: CPU ops always, HIP ops when
// built with KVIDX_WITH_HIP (the default on this ROCm image; hipcc
// cross-compiles gfx950 without a GPU present).

#include <torch/extension.h>

namespace kvidx {

// cpu_ops.cpp
std::vector<uint64_t> tokens_to_chunk_hashes(std::vector<uint64_t> tokens,
                                             uint64_t parent,
                                             int64_t block_size);
std::vector<at::Tensor> hash_chain_batch(at::Tensor, at::Tensor, at::Tensor,
                                         int64_t);
std::vector<uint64_t> tokens_to_chunk_hashes_fast(std::vector<uint64_t>,
                                                  uint64_t, int64_t);
void cpu_insert(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                at::Tensor, at::Tensor, int64_t, at::Tensor, at::Tensor,
                int64_t, at::Tensor, int64_t, int64_t, int64_t, int64_t);
at::Tensor cpu_score_from_masks(at::Tensor, at::Tensor, at::Tensor, int64_t);
void cpu_compact(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                 at::Tensor, at::Tensor, int64_t, at::Tensor, at::Tensor,
                 at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor);
void cpu_evict(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
               at::Tensor, at::Tensor, int64_t, at::Tensor, int64_t,
               at::Tensor);
std::vector<at::Tensor> cpu_lookup(at::Tensor, at::Tensor, at::Tensor,
                                   at::Tensor, at::Tensor, at::Tensor,
                                   at::Tensor, int64_t, at::Tensor, int64_t,
                                   at::Tensor, int64_t, int64_t, int64_t,
                                   int64_t, int64_t);
at::Tensor cpu_fused_score(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                           at::Tensor, at::Tensor, at::Tensor, int64_t,
                           at::Tensor, at::Tensor, int64_t, at::Tensor,
                           at::Tensor, int64_t, int64_t);
std::vector<at::Tensor> cpu_get_request_keys(at::Tensor, at::Tensor,
                                             at::Tensor, at::Tensor,
                                             at::Tensor, at::Tensor,
                                             at::Tensor, int64_t, at::Tensor,
                                             int64_t);

#ifdef KVIDX_WITH_HIP
// hip_ops.hip
void gpu_insert(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                at::Tensor, at::Tensor, int64_t, at::Tensor, at::Tensor,
                int64_t, at::Tensor, int64_t, int64_t, int64_t, int64_t);
void gpu_compact(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                 at::Tensor, at::Tensor, int64_t, at::Tensor, at::Tensor,
                 at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor);
void gpu_evict(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
               at::Tensor, at::Tensor, int64_t, at::Tensor, int64_t,
               at::Tensor);
std::vector<at::Tensor> gpu_get_request_keys(at::Tensor, at::Tensor,
                                             at::Tensor, at::Tensor,
                                             at::Tensor, at::Tensor,
                                             at::Tensor, int64_t, at::Tensor,
                                             int64_t);
std::vector<at::Tensor> gpu_lookup(at::Tensor, at::Tensor, at::Tensor,
                                   at::Tensor, at::Tensor, at::Tensor,
                                   at::Tensor, int64_t, at::Tensor, int64_t,
                                   at::Tensor, int64_t, int64_t, int64_t,
                                   int64_t, int64_t);
at::Tensor gpu_fused_score(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                           at::Tensor, at::Tensor, at::Tensor, int64_t,
                           at::Tensor, at::Tensor, int64_t, at::Tensor,
                           at::Tensor, int64_t, int64_t, int64_t, int64_t);
at::Tensor gpu_score_from_masks(at::Tensor, at::Tensor, at::Tensor, int64_t);
std::vector<at::Tensor> gpu_hash_chain(at::Tensor, at::Tensor, at::Tensor,
                                       int64_t);
at::Tensor gpu_hash_chain_tr(at::Tensor, at::Tensor, at::Tensor, int64_t,
                             int64_t, int64_t, int64_t);
void gpu_apply_events(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor, at::Tensor, at::Tensor, int64_t, at::Tensor,
                      at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor, int64_t, int64_t, int64_t, int64_t,
                      int64_t);
void gpu_apply_events_split(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                            at::Tensor, at::Tensor, at::Tensor, int64_t,
                            at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                            at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                            at::Tensor, at::Tensor, at::Tensor, int64_t,
                            int64_t, int64_t, int64_t, int64_t);
void gpu_apply_events_split_tr(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                            at::Tensor, at::Tensor, at::Tensor, int64_t,
                            at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                            at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                            at::Tensor, at::Tensor, int64_t, int64_t,
                            int64_t, int64_t, int64_t, int64_t);
#endif

namespace wire {
void register_wirefront(pybind11::module_& m);  // wirefront.cpp
}

}  // namespace kvidx

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native KV-block index ops (CPU reference + gfx950 HIP)";
  m.def("tokens_to_chunk_hashes", &kvidx::tokens_to_chunk_hashes);
  m.def("hash_chain_batch", &kvidx::hash_chain_batch);
  m.def("tokens_to_chunk_hashes_fast", &kvidx::tokens_to_chunk_hashes_fast);
  m.def("cpu_insert", &kvidx::cpu_insert);
  m.def("cpu_compact", &kvidx::cpu_compact);
  m.def("cpu_evict", &kvidx::cpu_evict);
  m.def("cpu_lookup", &kvidx::cpu_lookup);
  m.def("cpu_fused_score", &kvidx::cpu_fused_score);
  m.def("cpu_score_from_masks", &kvidx::cpu_score_from_masks);
  m.def("cpu_get_request_keys", &kvidx::cpu_get_request_keys);
#ifdef KVIDX_WITH_HIP
  m.attr("HAS_HIP") = true;
  m.def("gpu_insert", &kvidx::gpu_insert);
  m.def("gpu_compact", &kvidx::gpu_compact);
  m.def("gpu_evict", &kvidx::gpu_evict);
  m.def("gpu_get_request_keys", &kvidx::gpu_get_request_keys);
  m.def("gpu_lookup", &kvidx::gpu_lookup);
  m.def("gpu_fused_score", &kvidx::gpu_fused_score);
  m.def("gpu_score_from_masks", &kvidx::gpu_score_from_masks);
  m.def("gpu_hash_chain", &kvidx::gpu_hash_chain);
  m.def("gpu_hash_chain_tr", &kvidx::gpu_hash_chain_tr,
        py::arg("tokens_t"), py::arg("parents"), py::arg("n_chunks"),
        py::arg("block_size"), py::arg("max_chunks"), py::arg("ilp") = 0,
        py::arg("row_major") = 0);
  m.def("gpu_apply_events", &kvidx::gpu_apply_events);
  m.def("gpu_apply_events_split", &kvidx::gpu_apply_events_split);
  m.def("gpu_apply_events_split_tr", &kvidx::gpu_apply_events_split_tr);
#else
  m.attr("HAS_HIP") = false;
#endif
  kvidx::wire::register_wirefront(m);
}
