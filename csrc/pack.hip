// K2 — int4/int8 code packing (and unpack, for tests/CPU parity).
// codes int32 [n, k] in [0, 2^bits) -> qweight uint32 [n, k*bits/32],
// little-endian nibbles/bytes (layout doc: lfmq/ops/ref.py).  Bandwidth-
// trivial; correctness-critical — the same layout every dequant kernel
// decodes.  One output word per thread.
#include "common.h"

namespace {

template <int BITS>
__global__ void pack_kernel(const i32* __restrict__ codes,
                            u32* __restrict__ packed, u64 words, int k,
                            int kw) {
  constexpr int PER = 32 / BITS;
  for (u64 idx = (u64)blockIdx.x * blockDim.x + threadIdx.x; idx < words;
       idx += (u64)gridDim.x * blockDim.x) {
    const u64 row = idx / kw;
    const int w = idx % kw;
    const i32* src = codes + row * k + (u64)w * PER;
    u32 out = 0;
#pragma unroll
    for (int i = 0; i < PER; ++i) {
      const int kk = w * PER + i;
      const u32 c = kk < k ? (u32)src[i] & ((1u << BITS) - 1) : 0u;
      out |= c << (i * BITS);
    }
    packed[idx] = out;
  }
}

template <int BITS>
__global__ void unpack_kernel(const u32* __restrict__ packed,
                              i32* __restrict__ codes, u64 total, int k,
                              int kw) {
  constexpr int PER = 32 / BITS;
  for (u64 idx = (u64)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (u64)gridDim.x * blockDim.x) {
    const u64 row = idx / k;
    const int kk = idx % k;
    const u32 word = packed[row * kw + kk / PER];
    codes[idx] = (i32)((word >> ((kk % PER) * BITS)) & ((1u << BITS) - 1));
  }
}

u32 grid_for(u64 total) {
  u64 b = (total + 255) / 256;
  return (u32)(b > 2048 ? 2048 : (b ? b : 1));
}

}  // namespace

hipError_t lfmq_pack(const void* codes, void* packed, long n, int k, int bits,
                     hipStream_t stream) {
  const int per = 32 / bits;
  const int kw = (k + per - 1) / per;
  const u64 words = (u64)n * kw;
  if (bits == 4)
    hipLaunchKernelGGL(pack_kernel<4>, dim3(grid_for(words)), dim3(256), 0,
                       stream, (const i32*)codes, (u32*)packed, words, k, kw);
  else if (bits == 8)
    hipLaunchKernelGGL(pack_kernel<8>, dim3(grid_for(words)), dim3(256), 0,
                       stream, (const i32*)codes, (u32*)packed, words, k, kw);
  else
    return hipErrorInvalidValue;
  return hipGetLastError();
}

hipError_t lfmq_unpack(const void* packed, void* codes, long n, int k, int bits,
                       hipStream_t stream) {
  const int per = 32 / bits;
  const int kw = (k + per - 1) / per;
  const u64 total = (u64)n * k;
  if (bits == 4)
    hipLaunchKernelGGL(unpack_kernel<4>, dim3(grid_for(total)), dim3(256), 0,
                       stream, (const u32*)packed, (i32*)codes, total, k, kw);
  else if (bits == 8)
    hipLaunchKernelGGL(unpack_kernel<8>, dim3(grid_for(total)), dim3(256), 0,
                       stream, (const u32*)packed, (i32*)codes, total, k, kw);
  else
    return hipErrorInvalidValue;
  return hipGetLastError();
}
