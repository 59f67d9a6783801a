// _sail_kernels: pybind module assembling all HIP kernels (gfx950-only).
#include <torch/extension.h>

// strings.hip
torch::Tensor like_mask(torch::Tensor offsets, torch::Tensor bytes, py::bytes pattern);
torch::Tensor string_hash64(torch::Tensor offsets, torch::Tensor bytes);
std::vector<torch::Tensor> substr_fixed(torch::Tensor offsets, torch::Tensor bytes,
                                        int64_t start, int64_t len);
// hash_agg.hip
torch::Tensor grouped_acc(torch::Tensor gid, c10::optional<torch::Tensor> mask,
                          std::vector<c10::optional<torch::Tensor>> vals,
                          std::vector<int64_t> ops, int64_t G);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "sail_amd MI355X (gfx950) kernels";
  m.def("like_mask", &like_mask, "SQL LIKE mask over string column");
  m.def("string_hash64", &string_hash64, "FNV-1a 64 hash per string row");
  m.def("substr_fixed", &substr_fixed, "fixed-length substring extraction");
  m.def("grouped_acc", &grouped_acc,
        "fused grouped accumulation (sum/count/min/max) for small group counts");
}
