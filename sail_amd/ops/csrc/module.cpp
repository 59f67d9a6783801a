// _sail_kernels: pybind module assembling all HIP kernels (gfx950-only).
#include <torch/extension.h>

// strings.hip
torch::Tensor like_mask(torch::Tensor offsets, torch::Tensor bytes, py::bytes pattern);
torch::Tensor string_hash64(torch::Tensor offsets, torch::Tensor bytes);
std::vector<torch::Tensor> substr_fixed(torch::Tensor offsets, torch::Tensor bytes,
                                        int64_t start, int64_t len);
torch::Tensor string_hash64_seeded(torch::Tensor offsets, torch::Tensor bytes,
                                   int64_t seed, int64_t mult);
torch::Tensor str_pairs_equal(torch::Tensor offs_a, torch::Tensor bytes_a,
                              torch::Tensor ia, torch::Tensor offs_b,
                              torch::Tensor bytes_b, torch::Tensor ib);
// hash_agg.hip
torch::Tensor grouped_acc(torch::Tensor gid, c10::optional<torch::Tensor> mask,
                          std::vector<c10::optional<torch::Tensor>> vals,
                          std::vector<int64_t> ops, int64_t G);
// hash_join.hip
std::vector<torch::Tensor> hj_build(torch::Tensor keys);
torch::Tensor hj_probe_unique(torch::Tensor tkeys, torch::Tensor tvals, torch::Tensor keys);
std::vector<torch::Tensor> hj_build_chain(torch::Tensor keys);
torch::Tensor hj_probe_count(torch::Tensor tkeys, torch::Tensor theads,
                             torch::Tensor next, torch::Tensor keys);
std::vector<torch::Tensor> hj_probe_fill(torch::Tensor tkeys, torch::Tensor theads,
                                         torch::Tensor next, torch::Tensor keys,
                                         torch::Tensor offsets, int64_t total);
std::vector<torch::Tensor> hg_group(torch::Tensor keys);
// parquet_decode.hip
torch::Tensor pq_rle_decode(torch::Tensor buf, torch::Tensor pages, int64_t total);
torch::Tensor pq_plain_copy(torch::Tensor buf, torch::Tensor pages, int64_t total,
                            int64_t width);
void pq_copy_bytes(torch::Tensor buf, torch::Tensor pages, torch::Tensor out);
torch::Tensor pq_flba_i64(torch::Tensor buf, torch::Tensor pages, int64_t total,
                          int64_t width);
std::vector<torch::Tensor> pq_delta_decode(torch::Tensor buf, torch::Tensor pages,
                                           int64_t total);
std::vector<torch::Tensor> pq_bytearray_walk(torch::Tensor buf, torch::Tensor pages,
                                             int64_t total);
void pq_segscan(torch::Tensor data, torch::Tensor pages);
// mfma_reduce.hip
torch::Tensor mfma_sum_f64(torch::Tensor x, bool use_mfma);
torch::Tensor pq_gather_strings(torch::Tensor buf, torch::Tensor src_pos,
                                torch::Tensor lengths, torch::Tensor out_offsets,
                                int64_t total_bytes);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "sail_amd MI355X (gfx950) kernels";
  m.def("like_mask", &like_mask, "SQL LIKE mask over string column");
  m.def("string_hash64", &string_hash64, "FNV-1a 64 hash per string row");
  m.def("substr_fixed", &substr_fixed, "fixed-length substring extraction");
  m.def("string_hash64_seeded", &string_hash64_seeded,
        "seeded 64-bit string hash (second family for exact codes)");
  m.def("str_pairs_equal", &str_pairs_equal,
        "byte-exact equality of (row_a, row_b) string pairs");
  m.def("grouped_acc", &grouped_acc,
        "fused grouped accumulation (sum/count/min/max) for small group counts");
  m.def("hj_build", &hj_build, "hash join build (unique keys)");
  m.def("hj_probe_unique", &hj_probe_unique, "hash join probe, first match");
  m.def("hj_build_chain", &hj_build_chain, "hash join build with chains");
  m.def("hj_probe_count", &hj_probe_count, "hash join probe, count matches");
  m.def("hj_probe_fill", &hj_probe_fill, "hash join probe, emit match pairs");
  m.def("hg_group", &hg_group, "hash group-id assignment (sort-free group_ids)");
  m.def("pq_rle_decode", &pq_rle_decode, "parquet RLE/bit-packed hybrid decode");
  m.def("pq_plain_copy", &pq_plain_copy, "parquet PLAIN fixed-width page copy");
  m.def("pq_copy_bytes", &pq_copy_bytes, "parquet raw page-region byte copy");
  m.def("pq_flba_i64", &pq_flba_i64, "parquet FLBA big-endian -> int64");
  m.def("pq_delta_decode", &pq_delta_decode, "parquet DELTA_BINARY_PACKED decode");
  m.def("pq_bytearray_walk", &pq_bytearray_walk, "parquet PLAIN byte_array walk");
  m.def("pq_segscan", &pq_segscan, "per-page inclusive int64 scan (+page base)");
  m.def("mfma_sum_f64", &mfma_sum_f64,
        "f64 sum via v_mfma_f64_16x16x4 (measurement harness vs VALU)");
  m.def("pq_gather_strings", &pq_gather_strings, "gather byte_array payloads");
}
