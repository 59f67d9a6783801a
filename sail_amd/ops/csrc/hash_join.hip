// Hash join build/probe kernels for MI355X (gfx950).
//
// Open-addressing (linear probing) table in HBM keyed by int64, values are
// build-side row indices. MurmurHash3 finalizer spreads keys; table size is
// a power of two at ~50% max load. Device-scope atomicCAS makes the build
// correct across XCDs (guide Guideline 16).
//
//  * hj_build:        insert all keys; returns dup flag (any key seen twice)
//  * hj_probe_unique: first-match probe -> build row idx or -1 per probe row
//                     (exact joins for unique-key builds; semi/anti/existence
//                     joins for any build since only existence matters)
//  * hj_probe_count + hj_probe_fill: two-phase multi-match probe for
//                     duplicate-key builds (counts -> exclusive scan by the
//                     caller -> fill), replacing sort+searchsorted entirely.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <cstdint>

namespace {

constexpr int kBlock = 256;
constexpr long long kEmpty = 0x8000000000000001ll;  // reserved empty marker

inline int grid_for(int64_t n, int per_thread = 8) {
  int64_t blocks = (n + (int64_t)kBlock * per_thread - 1) / ((int64_t)kBlock * per_thread);
  return (int)std::max<int64_t>(1, std::min<int64_t>(blocks, 4096));
}

__device__ inline uint64_t mix64(uint64_t k) {
  k ^= k >> 33;
  k *= 0xff51afd7ed558ccdull;
  k ^= k >> 33;
  k *= 0xc4ceb9fe1a85ec53ull;
  k ^= k >> 33;
  return k;
}

__global__ void hj_build_kernel(const int64_t* __restrict__ keys, int64_t n,
                                int64_t* __restrict__ tkeys,
                                int64_t* __restrict__ tvals, int64_t tmask,
                                int* __restrict__ dup_flag) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    long long k = keys[i];
    if (k == kEmpty) k = kEmpty + 1;  // remap the reserved marker
    uint64_t h = mix64((uint64_t)k) & tmask;
    while (true) {
      long long prev = atomicCAS((unsigned long long*)&tkeys[h],
                                 (unsigned long long)kEmpty, (unsigned long long)k);
      if (prev == kEmpty) {
        tvals[h] = i;
        break;
      }
      if (prev == k) {
        *dup_flag = 1;  // duplicate key: keep first, flag for caller
        break;
      }
      h = (h + 1) & tmask;
    }
  }
}

__global__ void hj_probe_unique_kernel(const int64_t* __restrict__ keys, int64_t n,
                                       const int64_t* __restrict__ tkeys,
                                       const int64_t* __restrict__ tvals,
                                       int64_t tmask,
                                       int64_t* __restrict__ out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    long long k = keys[i];
    if (k == kEmpty) k = kEmpty + 1;
    uint64_t h = mix64((uint64_t)k) & tmask;
    int64_t r = -1;
    while (true) {
      long long tk = tkeys[h];
      if (tk == kEmpty) break;
      if (tk == k) {
        r = tvals[h];
        break;
      }
      h = (h + 1) & tmask;
    }
    out[i] = r;
  }
}

// duplicate-key build: chain layout. tvals holds the FIRST build row for the
// key; `next[row]` links further rows with the same key (built by a second
// pass in insertion order — order within a key is arbitrary, as in any hash
// join).
__global__ void hj_chain_kernel(const int64_t* __restrict__ keys, int64_t n,
                                int64_t* __restrict__ tkeys,
                                int64_t* __restrict__ theads, int64_t tmask,
                                int64_t* __restrict__ next) {
  // push-front chains: head[slot] <- row with next[row] = old head
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    long long k = keys[i];
    if (k == kEmpty) k = kEmpty + 1;
    uint64_t h = mix64((uint64_t)k) & tmask;
    while (true) {
      long long prev = atomicCAS((unsigned long long*)&tkeys[h],
                                 (unsigned long long)kEmpty, (unsigned long long)k);
      if (prev == kEmpty || prev == k) {
        // claim slot for this key (first claimer) or found the key's slot
        long long old = atomicExch((unsigned long long*)&theads[h],
                                   (unsigned long long)i);
        next[i] = old;  // old == -1 for first
        break;
      }
      h = (h + 1) & tmask;
    }
  }
}

__global__ void hj_probe_count_kernel(const int64_t* __restrict__ keys, int64_t n,
                                      const int64_t* __restrict__ tkeys,
                                      const int64_t* __restrict__ theads,
                                      const int64_t* __restrict__ next,
                                      int64_t tmask,
                                      int32_t* __restrict__ counts) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    long long k = keys[i];
    if (k == kEmpty) k = kEmpty + 1;
    uint64_t h = mix64((uint64_t)k) & tmask;
    int c = 0;
    while (true) {
      long long tk = tkeys[h];
      if (tk == kEmpty) break;
      if (tk == k) {
        for (int64_t r = theads[h]; r >= 0; r = next[r]) ++c;
        break;
      }
      h = (h + 1) & tmask;
    }
    counts[i] = c;
  }
}

__global__ void hj_probe_fill_kernel(const int64_t* __restrict__ keys, int64_t n,
                                     const int64_t* __restrict__ tkeys,
                                     const int64_t* __restrict__ theads,
                                     const int64_t* __restrict__ next,
                                     int64_t tmask,
                                     const int64_t* __restrict__ offsets,
                                     int64_t* __restrict__ out_probe,
                                     int64_t* __restrict__ out_build) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    long long k = keys[i];
    if (k == kEmpty) k = kEmpty + 1;
    uint64_t h = mix64((uint64_t)k) & tmask;
    int64_t o = offsets[i];
    while (true) {
      long long tk = tkeys[h];
      if (tk == kEmpty) break;
      if (tk == k) {
        for (int64_t r = theads[h]; r >= 0; r = next[r]) {
          out_probe[o] = i;
          out_build[o] = r;
          ++o;
        }
        break;
      }
      h = (h + 1) & tmask;
    }
  }
}

int64_t table_size_for(int64_t n) {
  int64_t sz = 64;
  while (sz < 2 * n) sz <<= 1;
  return sz;
}

// ------------------------------------------------------------------
// hash GROUP-ID assignment: the sort-free replacement for sparse-domain
// group_ids (unique-sort + searchsorted + representative scatter were
// ~600 ms of ClickBench's group-by-URL/WatchID queries). Claim pass CASes
// each distinct key into the table recording a representative row; the
// scan pass hands out dense ids in slot order; gid is then a plain gather.
// Group NUMBERING is arbitrary (hash aggregation is unordered — ORDER BY
// decides the final order, as on every other path).
// ------------------------------------------------------------------
__global__ void hg_claim_kernel(const int64_t* __restrict__ keys, int64_t n,
                                int64_t* __restrict__ tkeys,
                                int64_t* __restrict__ trow, int64_t tmask,
                                int64_t* __restrict__ slot_of_row) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    long long k = keys[i];
    if (k == kEmpty) k = kEmpty + 1;
    uint64_t h = mix64((uint64_t)k) & tmask;
    while (true) {
      long long prev = atomicCAS((unsigned long long*)&tkeys[h],
                                 (unsigned long long)kEmpty, (unsigned long long)k);
      if (prev == kEmpty) {
        trow[h] = i;  // claim winner records the representative row
        slot_of_row[i] = h;
        break;
      }
      if (prev == k) {
        slot_of_row[i] = h;
        break;
      }
      h = (h + 1) & tmask;
    }
  }
}

__global__ void hg_scan_kernel(const int64_t* __restrict__ tkeys,
                               const int64_t* __restrict__ trow, int64_t sz,
                               int32_t* __restrict__ slot_ids,
                               int64_t* __restrict__ rep,
                               int* __restrict__ counter) {
  for (int64_t s = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; s < sz;
       s += gridDim.x * (int64_t)blockDim.x) {
    if (tkeys[s] != (long long)kEmpty) {
      int id = atomicAdd(counter, 1);
      slot_ids[s] = id;
      rep[id] = trow[s];
    }
  }
}

}  // namespace

std::vector<torch::Tensor> hj_build(torch::Tensor keys) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  int64_t n = keys.numel();
  int64_t sz = table_size_for(std::max<int64_t>(n, 1));
  auto tkeys = torch::full({sz}, kEmpty, keys.options());
  auto tvals = torch::empty({sz}, keys.options());
  auto dup = torch::zeros({1}, keys.options().dtype(torch::kInt32));
  if (n) {
    hipStream_t stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hj_build_kernel, dim3(grid_for(n)), dim3(kBlock), 0, stream,
                       keys.data_ptr<int64_t>(), n, tkeys.data_ptr<int64_t>(),
                       tvals.data_ptr<int64_t>(), sz - 1, dup.data_ptr<int32_t>());
  }
  return {tkeys, tvals, dup};
}

torch::Tensor hj_probe_unique(torch::Tensor tkeys, torch::Tensor tvals,
                              torch::Tensor keys) {
  int64_t n = keys.numel();
  auto out = torch::empty({n}, keys.options());
  if (n) {
    hipStream_t stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hj_probe_unique_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, keys.data_ptr<int64_t>(), n,
                       tkeys.data_ptr<int64_t>(), tvals.data_ptr<int64_t>(),
                       tkeys.numel() - 1, out.data_ptr<int64_t>());
  }
  return out;
}

std::vector<torch::Tensor> hj_build_chain(torch::Tensor keys) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  int64_t n = keys.numel();
  int64_t sz = table_size_for(std::max<int64_t>(n, 1));
  auto tkeys = torch::full({sz}, kEmpty, keys.options());
  auto theads = torch::full({sz}, -1, keys.options());
  auto next = torch::empty({std::max<int64_t>(n, 1)}, keys.options());
  if (n) {
    hipStream_t stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hj_chain_kernel, dim3(grid_for(n)), dim3(kBlock), 0, stream,
                       keys.data_ptr<int64_t>(), n, tkeys.data_ptr<int64_t>(),
                       theads.data_ptr<int64_t>(), sz - 1, next.data_ptr<int64_t>());
  }
  return {tkeys, theads, next};
}

torch::Tensor hj_probe_count(torch::Tensor tkeys, torch::Tensor theads,
                             torch::Tensor next, torch::Tensor keys) {
  int64_t n = keys.numel();
  auto counts = torch::zeros({n}, keys.options().dtype(torch::kInt32));
  if (n) {
    hipStream_t stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hj_probe_count_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, keys.data_ptr<int64_t>(), n,
                       tkeys.data_ptr<int64_t>(), theads.data_ptr<int64_t>(),
                       next.data_ptr<int64_t>(), tkeys.numel() - 1,
                       counts.data_ptr<int32_t>());
  }
  return counts;
}

std::vector<torch::Tensor> hj_probe_fill(torch::Tensor tkeys, torch::Tensor theads,
                                         torch::Tensor next, torch::Tensor keys,
                                         torch::Tensor offsets, int64_t total) {
  int64_t n = keys.numel();
  auto out_probe = torch::empty({total}, keys.options());
  auto out_build = torch::empty({total}, keys.options());
  if (n && total) {
    hipStream_t stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hj_probe_fill_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, keys.data_ptr<int64_t>(), n,
                       tkeys.data_ptr<int64_t>(), theads.data_ptr<int64_t>(),
                       next.data_ptr<int64_t>(), tkeys.numel() - 1,
                       offsets.data_ptr<int64_t>(), out_probe.data_ptr<int64_t>(),
                       out_build.data_ptr<int64_t>());
  }
  return {out_probe, out_build};
}


std::vector<torch::Tensor> hg_group(torch::Tensor keys) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  int64_t n = keys.numel();
  int64_t sz = table_size_for(std::max<int64_t>(n, 1));
  auto tkeys = torch::full({sz}, kEmpty, keys.options());
  auto trow = torch::empty({sz}, keys.options());
  auto slot_of_row = torch::empty({std::max<int64_t>(n, 1)}, keys.options());
  auto slot_ids = torch::empty({sz}, keys.options().dtype(torch::kInt32));
  auto rep = torch::empty({std::max<int64_t>(n, 1)}, keys.options());
  auto counter = torch::zeros({1}, keys.options().dtype(torch::kInt32));
  if (n) {
    hipStream_t stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_claim_kernel, dim3(grid_for(n)), dim3(kBlock), 0, stream,
                       keys.data_ptr<int64_t>(), n, tkeys.data_ptr<int64_t>(),
                       trow.data_ptr<int64_t>(), sz - 1,
                       slot_of_row.data_ptr<int64_t>());
    hipLaunchKernelGGL(hg_scan_kernel, dim3(grid_for(sz)), dim3(kBlock), 0, stream,
                       tkeys.data_ptr<int64_t>(), trow.data_ptr<int64_t>(), sz,
                       slot_ids.data_ptr<int32_t>(), rep.data_ptr<int64_t>(),
                       counter.data_ptr<int32_t>());
  }
  // gid = slot_ids[slot_of_row]; ng = counter (host sync on read)
  auto gid = slot_ids.index_select(0, slot_of_row.slice(0, 0, n)).to(torch::kInt64);
  return {gid, rep, counter};
}
