// String kernels for MI355X (gfx950).
//
// Device-side operations over Arrow-layout string columns
// (offsets int64[n+1], bytes uint8[total]):
//   like_mask      — SQL LIKE ('%', '_') per row -> bool mask
//   string_hash64  — FNV-1a 64-bit per row (grouping/join keys for raw
//                    strings; dictionary-encoded columns never need this)
//   substr_fixed   — substring(start,len) into a fixed-pitch buffer
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - one thread per row, grid-stride; block = 256 (4 waves);
//    memory-bound — strings are short (<64B), threads in a wave read
//    adjacent rows so the wave touches a contiguous byte range (L2-friendly).
//  - grid capped at 2048 blocks per Guideline 11.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <cstdint>
#include <vector>

#define SAIL_CHECK(x) TORCH_CHECK(x, #x)

namespace {

constexpr int kBlock = 256;

inline int grid_for(int64_t n) {
  int64_t blocks = (n + kBlock - 1) / kBlock;
  return (int)std::min<int64_t>(blocks, 2048);
}

__device__ inline bool like_match(const uint8_t* s, int64_t slen,
                                  const uint8_t* p, int plen) {
  // iterative wildcard match with single backtrack point ('%')
  int64_t si = 0, ss = 0;
  int pi = 0, star = -1;
  while (si < slen) {
    if (pi < plen && (p[pi] == '_' || p[pi] == s[si])) {
      ++si;
      ++pi;
    } else if (pi < plen && p[pi] == '%') {
      star = pi++;
      ss = si;
    } else if (star >= 0) {
      pi = star + 1;
      si = ++ss;
    } else {
      return false;
    }
  }
  while (pi < plen && p[pi] == '%') ++pi;
  return pi == plen;
}

__global__ void like_mask_kernel(const int64_t* __restrict__ offsets,
                                 const uint8_t* __restrict__ bytes,
                                 const uint8_t* __restrict__ pattern, int plen,
                                 bool* __restrict__ out, int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    int64_t lo = offsets[i];
    out[i] = like_match(bytes + lo, offsets[i + 1] - lo, pattern, plen);
  }
}

// SWAR first-char scan: 8 candidate positions tested per 8-byte load
// (guide: byte-serial inner loops are the measured cost of LIKE over 150M
// comments — the bytes are L1-resident after the first touch, the ops are
// not free).
__device__ inline int64_t find_from(const uint8_t* s, int64_t len,
                                    const uint8_t* needle, int nlen,
                                    int64_t start) {
  if (nlen == 0) return start <= len ? start : -1;
  uint8_t c0 = needle[0];
  uint64_t pat = 0x0101010101010101ull * (uint64_t)c0;
  int64_t last = len - nlen;
  int64_t j = start;
  for (; j + 8 <= len; j += 8) {
    uint64_t w;
    __builtin_memcpy(&w, s + j, 8);
    uint64_t x = w ^ pat;
    uint64_t z = (x - 0x0101010101010101ull) & ~x & 0x8080808080808080ull;
    while (z) {
      int pos = (__ffsll((unsigned long long)z) - 1) / 8;  // byte index
      int64_t cand = j + pos;
      z &= z - 1ull;  // clear lowest set bit's byte marker
      if (cand > last) return -1;
      bool ok = true;
      for (int k = 1; k < nlen; ++k) {
        if (s[cand + k] != needle[k]) { ok = false; break; }
      }
      if (ok) return cand;
    }
    if (j + 8 > last) break;
  }
  for (; j <= last; ++j) {
    if (s[j] != c0) continue;
    bool ok = true;
    for (int k = 1; k < nlen; ++k) {
      if (s[j + k] != needle[k]) { ok = false; break; }
    }
    if (ok) return j;
  }
  return -1;
}

__global__ void contains_kernel(const int64_t* __restrict__ offsets,
                                const uint8_t* __restrict__ bytes,
                                const uint8_t* __restrict__ needle, int nlen,
                                bool* __restrict__ out, int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    int64_t lo = offsets[i];
    int64_t len = offsets[i + 1] - lo;
    out[i] = find_from(bytes + lo, len, needle, nlen, 0) >= 0;
  }
}

// '%s1%s2%...%': sequential floating-segment search (q13's
// '%special%requests%'), each segment via the SWAR scanner.
__global__ void contains_chain_kernel(const int64_t* __restrict__ offsets,
                                      const uint8_t* __restrict__ bytes,
                                      const uint8_t* __restrict__ needles,
                                      const int* __restrict__ seg_off,
                                      int nseg,
                                      bool* __restrict__ out, int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    int64_t lo = offsets[i];
    int64_t len = offsets[i + 1] - lo;
    const uint8_t* s = bytes + lo;
    int64_t at = 0;
    bool ok = true;
    for (int g = 0; g < nseg && ok; ++g) {
      int nl = seg_off[g + 1] - seg_off[g];
      int64_t f = find_from(s, len, needles + seg_off[g], nl, at);
      if (f < 0) ok = false;
      else at = f + nl;
    }
    out[i] = ok;
  }
}

__global__ void string_hash64_kernel(const int64_t* __restrict__ offsets,
                                     const uint8_t* __restrict__ bytes,
                                     int64_t* __restrict__ out, int64_t n) {
  // FNV-1a is byte-serial, but the LOADS need not be: pull 8 bytes per
  // unaligned dword2 load and fold them from the register — 8x fewer
  // memory ops per row while producing the identical byte-order hash
  // (must match fnv_key_tensor / the CPU reference bit-for-bit).
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    int64_t lo = offsets[i], hi = offsets[i + 1];
    uint64_t h = 14695981039346656037ull;
    int64_t j = lo;
    for (; j + 8 <= hi; j += 8) {
      uint64_t w;
      __builtin_memcpy(&w, bytes + j, 8);  // unaligned load, little-endian
#pragma unroll
      for (int b = 0; b < 8; ++b) {
        h = (h ^ (w & 0xffull)) * 1099511628211ull;
        w >>= 8;
      }
    }
    for (; j < hi; ++j) {
      h = (h ^ bytes[j]) * 1099511628211ull;
    }
    // mix in length; keep sign bit clear so sentinels (<0) stay distinct
    h ^= (uint64_t)(hi - lo) * 0x9E3779B97F4A7C15ull;
    out[i] = (int64_t)(h >> 1);
  }
}

__global__ void string_hash64_seeded_kernel(const int64_t* __restrict__ offsets,
                                            const uint8_t* __restrict__ bytes,
                                            int64_t* __restrict__ out, int64_t n,
                                            uint64_t seed, uint64_t mult) {
  // second, independent hash family for the 128-bit exact-string-code
  // scheme (joins.exact_string_codes): different multiplier, not just a
  // different seed — an FNV seed change is a function of (h1, len) and
  // would NOT be independent
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    int64_t lo = offsets[i], hi = offsets[i + 1];
    uint64_t h = seed;
    int64_t j = lo;
    for (; j + 8 <= hi; j += 8) {
      uint64_t w;
      __builtin_memcpy(&w, bytes + j, 8);
#pragma unroll
      for (int b = 0; b < 8; ++b) {
        h = (h ^ (w & 0xffull)) * mult;
        w >>= 8;
      }
    }
    for (; j < hi; ++j) h = (h ^ bytes[j]) * mult;
    h ^= (uint64_t)(hi - lo) * 0x2545F4914F6CDD1Dull;
    out[i] = (int64_t)h;
  }
}

__global__ void str_pairs_equal_kernel(const int64_t* __restrict__ offs_a,
                                       const uint8_t* __restrict__ bytes_a,
                                       const int64_t* __restrict__ ia,
                                       const int64_t* __restrict__ offs_b,
                                       const uint8_t* __restrict__ bytes_b,
                                       const int64_t* __restrict__ ib,
                                       uint8_t* __restrict__ out, int64_t n) {
  // byte-exact equality per (row_a, row_b) pair — the verification pass
  // that turns hashed string keys into exact ones
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    int64_t a = ia[i], b = ib[i];
    int64_t la = offs_a[a + 1] - offs_a[a];
    int64_t lb = offs_b[b + 1] - offs_b[b];
    if (la != lb) { out[i] = 0; continue; }
    const uint8_t* pa = bytes_a + offs_a[a];
    const uint8_t* pb = bytes_b + offs_b[b];
    int64_t j = 0;
    bool eq = true;
    for (; j + 8 <= la; j += 8) {
      uint64_t wa, wb;
      __builtin_memcpy(&wa, pa + j, 8);
      __builtin_memcpy(&wb, pb + j, 8);
      if (wa != wb) { eq = false; break; }
    }
    if (eq)
      for (; j < la; ++j)
        if (pa[j] != pb[j]) { eq = false; break; }
    out[i] = eq ? 1 : 0;
  }
}

__global__ void substr_fixed_kernel(const int64_t* __restrict__ offsets,
                                    const uint8_t* __restrict__ bytes,
                                    int start, int len,
                                    uint8_t* __restrict__ out_bytes,
                                    int32_t* __restrict__ out_lens, int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (int64_t)blockDim.x) {
    int64_t lo = offsets[i];
    int64_t slen = offsets[i + 1] - lo;
    int64_t b = lo + start;
    int m = 0;
    for (; m < len && start + m < slen; ++m) {
      out_bytes[i * len + m] = bytes[b + m];
    }
    for (int k = m; k < len; ++k) out_bytes[i * len + k] = 0;
    out_lens[i] = m;
  }
}

}  // namespace

torch::Tensor like_mask(torch::Tensor offsets, torch::Tensor bytes, py::bytes pattern) {
  SAIL_CHECK(offsets.is_cuda() && bytes.is_cuda());
  std::string pat(pattern);
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, offsets.options().dtype(torch::kBool));
  if (n == 0) return out;
  auto patT = torch::empty({(int64_t)pat.size()},
                           torch::TensorOptions().dtype(torch::kUInt8));
  std::memcpy(patT.data_ptr(), pat.data(), pat.size());
  patT = patT.to(offsets.device());
  // fast path: pure containment '%abc%' (no '_' inside)
  bool pure_contains = pat.size() >= 2 && pat.front() == '%' && pat.back() == '%' &&
                       pat.find('_') == std::string::npos &&
                       pat.find('%', 1) == pat.size() - 1;
  // '%s1%s2%...%': floating-segment chain (no '_' anywhere)
  bool chain = !pure_contains && pat.size() >= 2 && pat.front() == '%' &&
               pat.back() == '%' && pat.find('_') == std::string::npos;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  if (pure_contains) {
    hipLaunchKernelGGL(contains_kernel, dim3(grid_for(n)), dim3(kBlock), 0, stream,
                       offsets.data_ptr<int64_t>(), bytes.data_ptr<uint8_t>(),
                       patT.data_ptr<uint8_t>() + 1, (int)pat.size() - 2,
                       out.data_ptr<bool>(), n);
  } else if (chain) {
    std::string needles;
    std::vector<int> seg_off = {0};
    size_t p = 1;
    while (p < pat.size()) {
      size_t q = pat.find('%', p);
      if (q == std::string::npos) q = pat.size();
      if (q > p) {
        needles.append(pat, p, q - p);
        seg_off.push_back((int)needles.size());
      }
      p = q + 1;
    }
    int nseg = (int)seg_off.size() - 1;
    auto nT = torch::empty({(int64_t)std::max<size_t>(needles.size(), 1)},
                           torch::TensorOptions().dtype(torch::kUInt8));
    if (!needles.empty()) std::memcpy(nT.data_ptr(), needles.data(), needles.size());
    nT = nT.to(offsets.device());
    auto oT = torch::empty({(int64_t)seg_off.size()},
                           torch::TensorOptions().dtype(torch::kInt32));
    std::memcpy(oT.data_ptr(), seg_off.data(), seg_off.size() * sizeof(int));
    oT = oT.to(offsets.device());
    hipLaunchKernelGGL(contains_chain_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, offsets.data_ptr<int64_t>(),
                       bytes.data_ptr<uint8_t>(), nT.data_ptr<uint8_t>(),
                       oT.data_ptr<int>(), nseg, out.data_ptr<bool>(), n);
  } else {
    hipLaunchKernelGGL(like_mask_kernel, dim3(grid_for(n)), dim3(kBlock), 0, stream,
                       offsets.data_ptr<int64_t>(), bytes.data_ptr<uint8_t>(),
                       patT.data_ptr<uint8_t>(), (int)pat.size(),
                       out.data_ptr<bool>(), n);
  }
  return out;
}

torch::Tensor string_hash64(torch::Tensor offsets, torch::Tensor bytes) {
  SAIL_CHECK(offsets.is_cuda() && bytes.is_cuda());
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, offsets.options().dtype(torch::kInt64));
  if (n == 0) return out;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(string_hash64_kernel, dim3(grid_for(n)), dim3(kBlock), 0, stream,
                     offsets.data_ptr<int64_t>(), bytes.data_ptr<uint8_t>(),
                     out.data_ptr<int64_t>(), n);
  return out;
}

torch::Tensor string_hash64_seeded(torch::Tensor offsets, torch::Tensor bytes,
                                   int64_t seed, int64_t mult) {
  SAIL_CHECK(offsets.is_cuda() && bytes.is_cuda());
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, offsets.options().dtype(torch::kInt64));
  if (n == 0) return out;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(string_hash64_seeded_kernel, dim3(grid_for(n)), dim3(kBlock),
                     0, stream, offsets.data_ptr<int64_t>(),
                     bytes.data_ptr<uint8_t>(), out.data_ptr<int64_t>(), n,
                     (uint64_t)seed, (uint64_t)mult);
  return out;
}

torch::Tensor str_pairs_equal(torch::Tensor offs_a, torch::Tensor bytes_a,
                              torch::Tensor ia, torch::Tensor offs_b,
                              torch::Tensor bytes_b, torch::Tensor ib) {
  SAIL_CHECK(offs_a.is_cuda() && bytes_a.is_cuda() && ia.is_cuda());
  int64_t n = ia.numel();
  auto out = torch::empty({n}, bytes_a.options().dtype(torch::kUInt8));
  if (n == 0) return out;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(str_pairs_equal_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                     stream, offs_a.data_ptr<int64_t>(),
                     bytes_a.data_ptr<uint8_t>(), ia.data_ptr<int64_t>(),
                     offs_b.data_ptr<int64_t>(), bytes_b.data_ptr<uint8_t>(),
                     ib.data_ptr<int64_t>(), out.data_ptr<uint8_t>(), n);
  return out;
}

std::vector<torch::Tensor> substr_fixed(torch::Tensor offsets, torch::Tensor bytes,
                                        int64_t start, int64_t len) {
  SAIL_CHECK(offsets.is_cuda() && bytes.is_cuda());
  int64_t n = offsets.numel() - 1;
  auto out_bytes = torch::empty({n * len}, bytes.options());
  auto out_lens = torch::empty({n}, offsets.options().dtype(torch::kInt32));
  if (n) {
    hipStream_t stream = c10::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(substr_fixed_kernel, dim3(grid_for(n)), dim3(kBlock), 0, stream,
                       offsets.data_ptr<int64_t>(), bytes.data_ptr<uint8_t>(),
                       (int)start, (int)len, out_bytes.data_ptr<uint8_t>(),
                       out_lens.data_ptr<int32_t>(), n);
  }
  return {out_bytes, out_lens};
}
