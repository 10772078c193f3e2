// K7 — fused rotary embedding (NeoX half-split, Llama-3 θ=500000),
// applied in place to q [b,s,hq,d] and k [b,s,hkv,d] bf16 in ONE launch.
// inv_freq [d/2] fp32 is precomputed on host (Appendix B: keep trig
// tables off the device hot path; only the per-token sincos runs here).
#include "common.h"

namespace {

__global__ void rope_kernel(u32* __restrict__ q, u32* __restrict__ k,
                            const i32* __restrict__ positions,
                            const float* __restrict__ inv_freq, int bs, int s,
                            int hq, int hkv, int d) {
  const int dp = d / 4;  // dword-pairs per head half
  const u64 q_pairs = (u64)bs * hq * dp;
  const u64 total = q_pairs + (u64)bs * hkv * dp;
  for (u64 idx = (u64)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (u64)gridDim.x * blockDim.x) {
    u32* t;
    int nh;
    u64 rel;
    if (idx < q_pairs) {
      t = q; nh = hq; rel = idx;
    } else {
      t = k; nh = hkv; rel = idx - q_pairs;
    }
    const int j = rel % dp;
    const u64 th = rel / dp;            // (token, head) flat
    const int tok = th / nh;
    const int pos = positions[tok % s];
    // dword j of the first half pairs with dword j + d/4 (elements
    // (2j, 2j+1) with (2j + d/2, 2j+1 + d/2))
    u32* base = t + th * (d / 2);
    const u32 a = base[j], b = base[j + dp];
    float a0, a1, b0, b1;
    bf16x2_to_f32(a, a0, a1);
    bf16x2_to_f32(b, b0, b1);
    float ang0 = (float)pos * inv_freq[2 * j];
    float ang1 = (float)pos * inv_freq[2 * j + 1];
    float c0, s0, c1, s1;
    __sincosf(ang0, &s0, &c0);
    __sincosf(ang1, &s1, &c1);
    base[j] = pack_bf16x2(a0 * c0 - b0 * s0, a1 * c1 - b1 * s1);
    base[j + dp] = pack_bf16x2(a0 * s0 + b0 * c0, a1 * s1 + b1 * c1);
  }
}

}  // namespace

hipError_t lfmq_rope(void* q, void* k, const void* positions,
                     const void* inv_freq, int bs, int s, int hq, int hkv,
                     int d, hipStream_t stream) {
  if (d % 4) return hipErrorInvalidValue;
  u64 total = (u64)bs * (hq + hkv) * (d / 4);
  u32 blocks = (u32)((total + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  if (blocks == 0) blocks = 1;
  hipLaunchKernelGGL(rope_kernel, dim3(blocks), dim3(256), 0, stream, (u32*)q,
                     (u32*)k, (const i32*)positions, (const float*)inv_freq,
                     bs, s, hq, hkv, d);
  return hipGetLastError();
}
