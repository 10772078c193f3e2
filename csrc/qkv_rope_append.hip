// Fused decode epilogue of the qkv projection (K7 + K10 in one pass):
// reads the raw fused qkv GEMV output [b*s, (hq+2hkv)*d] bf16 and in one
// launch (a) applies NeoX rotary to q -> q_out [b, s, hq, d] (== the
// [b, hq, 1, d] attention layout at s == 1, so decode never transposes),
// (b) applies rotary to k and scatters it into k_cache at its position,
// (c) scatters v into v_cache.  Replaces rope + kv_append + two reshape
// copies on the decode path — these tensors are tiny, so the win is
// launch/latency count, the thing the decode step is bound by.
#include "common.h"

namespace {

__global__ void qkv_rope_append_kernel(
    const u32* __restrict__ qkv, u32* __restrict__ q_out,
    u32* __restrict__ kc, u32* __restrict__ vc,
    const i32* __restrict__ positions, const float* __restrict__ inv_freq,
    int b, int s, int hq, int hkv, int d, int tmax, int pos0,
    const i32* __restrict__ pos_dev) {
  if (pos_dev != nullptr) pos0 = *pos_dev;  // hipGraph mode
  const int dp = d / 4;          // dword-pairs per rotated half
  const int dw = d / 2;          // dwords per head
  const int qkv_dw = (hq + 2 * hkv) * dw;
  const u64 q_pairs = (u64)b * s * hq * dp;
  const u64 k_pairs = (u64)b * s * hkv * dp;
  const u64 v_dwords = (u64)b * s * hkv * dw;
  const u64 total = q_pairs + k_pairs + v_dwords;

  for (u64 idx = (u64)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (u64)gridDim.x * blockDim.x) {
    if (idx < q_pairs + k_pairs) {
      const bool is_q = idx < q_pairs;
      const u64 rel = is_q ? idx : idx - q_pairs;
      const int nh = is_q ? hq : hkv;
      const int j = rel % dp;
      const int hh = (rel / dp) % nh;
      const int tok = rel / ((u64)dp * nh);
      const int si = tok % s;
      const int bb = tok / s;
      const int pos = positions[si];
      const u32* src = qkv + (u64)tok * qkv_dw +
                       (is_q ? hh * dw : (hq + hh) * dw);
      const u32 a = src[j], bv = src[j + dp];
      float a0, a1, b0, b1;
      bf16x2_to_f32(a, a0, a1);
      bf16x2_to_f32(bv, b0, b1);
      const float ang0 = (float)pos * inv_freq[2 * j];
      const float ang1 = (float)pos * inv_freq[2 * j + 1];
      float c0, s0, c1, s1;
      __sincosf(ang0, &s0, &c0);
      __sincosf(ang1, &s1, &c1);
      const u32 lo = pack_bf16x2(a0 * c0 - b0 * s0, a1 * c1 - b1 * s1);
      const u32 hi = pack_bf16x2(a0 * s0 + b0 * c0, a1 * s1 + b1 * c1);
      if (is_q) {
        u32* dst = q_out + ((u64)tok * hq + hh) * dw;
        dst[j] = lo;
        dst[j + dp] = hi;
      } else {
        u32* dst = kc + (((u64)bb * hkv + hh) * tmax + pos0 + si) * dw;
        dst[j] = lo;
        dst[j + dp] = hi;
      }
    } else {
      const u64 rel = idx - q_pairs - k_pairs;
      const int j = rel % dw;
      const int hh = (rel / dw) % hkv;
      const int tok = rel / ((u64)dw * hkv);
      const int si = tok % s;
      const int bb = tok / s;
      const u32 v = qkv[(u64)tok * qkv_dw + (hq + hkv + hh) * dw + j];
      vc[(((u64)bb * hkv + hh) * tmax + pos0 + si) * dw + j] = v;
    }
  }
}

}  // namespace

hipError_t lfmq_qkv_rope_append(const void* qkv, void* q_out, void* kc,
                                void* vc, const void* positions,
                                const void* inv_freq, int b, int s, int hq,
                                int hkv, int d, int tmax, int pos0,
                                const void* pos_dev, hipStream_t stream) {
  if (d % 4) return hipErrorInvalidValue;
  const u64 total = (u64)b * s * ((hq + hkv) * (d / 4) + hkv * (d / 2));
  u32 blocks = (u32)((total + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  if (blocks == 0) blocks = 1;
  hipLaunchKernelGGL(qkv_rope_append_kernel, dim3(blocks), dim3(256), 0,
                     stream, (const u32*)qkv, (u32*)q_out, (u32*)kc, (u32*)vc,
                     (const i32*)positions, (const float*)inv_freq, b, s, hq,
                     hkv, d, tmax, pos0, (const i32*)pos_dev);
  return hipGetLastError();
}
