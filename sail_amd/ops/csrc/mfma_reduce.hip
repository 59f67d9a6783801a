// MFMA-assisted f64 sum reduction — the measurement the north star asks
// for (MFMA for aggregate reductions). CDNA4 keeps the f64 matrix op
// v_mfma_f64_16x16x4_f64 (D[16x16] += A[16x4]*B[4x16]); with B = ones the
// accumulator's column 0 collects row sums of streamed A tiles, turning a
// column sum into matrix ops.
//
// Expectation going in (documented either way, profiles/README.md): a
// whole-column sum reads each element once — HBM-bandwidth-bound at
// ~6.3 TB/s regardless of the FLOP engine. One VALU v_add_f64 per element
// already saturates that; MFMA's 2048 FLOP/instr cannot add bandwidth.
// The kernel exists to MEASURE that, not to assume it.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define SAIL_CHECK(x) TORCH_CHECK(x, #x)

namespace {

typedef double v4d __attribute__((ext_vector_type(4)));

// one wave per block-slice; each lane streams elements as the A fragment
// (lane l holds A[l%16, l/16]), B = 1.0 broadcast.
__global__ void mfma_sum_f64_kernel(const double* __restrict__ x, long n,
                                    double* __restrict__ out) {
#if defined(__gfx950__) || defined(__gfx90a__) || defined(__gfx942__)
  v4d acc = {0.0, 0.0, 0.0, 0.0};
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = idx; i < n; i += stride) {
    double a = x[i];
    acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, 1.0, acc, 0, 0, 0);
  }
  // acc lanes hold D[16x16] fragments; every element of D is a partial sum
  // of DISTINCT inputs only in column 0 rows — but with B==1 every column
  // j of D receives the same row sums, so D holds 16 copies. Take lane
  // fragment sum / 16 after a wave reduction.
  double mine = acc[0] + acc[1] + acc[2] + acc[3];
  for (int off = 32; off; off >>= 1)
    mine += __shfl_down(mine, off, 64);
  if ((threadIdx.x & 63) == 0)
    atomicAdd(out, mine / 16.0);
#else
  // non-MFMA fallback: plain strided add (same contract)
  double mine = 0.0;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = idx; i < n; i += stride) mine += x[i];
  for (int off = 32; off; off >>= 1)
    mine += __shfl_down(mine, off, 64);
  if ((threadIdx.x & 63) == 0) atomicAdd(out, mine);
#endif
}

// VALU baseline with identical structure (for the A/B measurement)
__global__ void valu_sum_f64_kernel(const double* __restrict__ x, long n,
                                    double* __restrict__ out) {
  double mine = 0.0;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = idx; i < n; i += stride) mine += x[i];
  for (int off = 32; off; off >>= 1)
    mine += __shfl_down(mine, off, 64);
  if ((threadIdx.x & 63) == 0) atomicAdd(out, mine);
}

}  // namespace

torch::Tensor mfma_sum_f64(torch::Tensor x, bool use_mfma) {
  SAIL_CHECK(x.is_cuda());
  SAIL_CHECK(x.scalar_type() == torch::kFloat64);
  auto out = torch::zeros({1}, x.options());
  long n = x.numel();
  if (n == 0) return out;
  int blocks = (int)std::min<long>((n + 255) / 256, 4096);
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  if (use_mfma) {
    hipLaunchKernelGGL(mfma_sum_f64_kernel, dim3(blocks), dim3(256), 0,
                       stream, x.data_ptr<double>(), n, out.data_ptr<double>());
  } else {
    hipLaunchKernelGGL(valu_sum_f64_kernel, dim3(blocks), dim3(256), 0,
                       stream, x.data_ptr<double>(), n, out.data_ptr<double>());
  }
  return out;
}
