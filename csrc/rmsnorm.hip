// K6 — fused RMSNorm: y = x · rsqrt(mean(x²)+ε) · w, optional fused
// residual add in the same pass (one read, one write — the op is
// HBM-bound).
// One workgroup per row; bf16 loaded 8-wide (G13: scalar bf16 is 2×
// slower); wave shuffle + LDS cross-wave reduction.
#include "common.h"

namespace {

constexpr int TPB = 256;

// MODE: 0 = plain, 1 = fused residual-in (y = norm(x+r)),
//       2 = fused residual-in + write new residual (r_out = x+r)
template <int MODE>
__global__ __launch_bounds__(TPB) void rmsnorm_kernel(
    const u32* __restrict__ x, const u32* __restrict__ res_in,
    const u32* __restrict__ w, u32* __restrict__ y, u32* __restrict__ res_out,
    int hidden, float eps) {
  const int row = blockIdx.x;
  const int dw = hidden / 2;  // dwords per row (hidden even)
  const u32* xr = x + (u64)row * dw;
  const u32* rr = MODE ? res_in + (u64)row * dw : nullptr;
  u32* yr = y + (u64)row * dw;
  u32* ro = MODE == 2 ? res_out + (u64)row * dw : nullptr;

  __shared__ float red[TPB / LFMQ_WAVE];
  __shared__ float inv_rms_sh;

  float ss = 0.f;
  // pass 1: sum of squares (values kept in registers when hidden small
  // enough; re-read from L1 otherwise — rows are <= 16 KB, L1-resident)
  for (int i = threadIdx.x; i < dw; i += TPB) {
    float a, b;
    bf16x2_to_f32(xr[i], a, b);
    if (MODE) {
      float c, d;
      bf16x2_to_f32(rr[i], c, d);
      a += c;
      b += d;
    }
    ss = fmaf(a, a, ss);
    ss = fmaf(b, b, ss);
  }
  ss = wave_reduce_sum(ss);
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  if (lane == 0) red[wave] = ss;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
#pragma unroll
    for (int i = 0; i < TPB / LFMQ_WAVE; ++i) t += red[i];
    inv_rms_sh = rsqrtf(t / (float)hidden + eps);
  }
  __syncthreads();
  const float inv = inv_rms_sh;

  for (int i = threadIdx.x; i < dw; i += TPB) {
    float a, b, wa, wb;
    bf16x2_to_f32(xr[i], a, b);
    bf16x2_to_f32(w[i], wa, wb);
    if (MODE) {
      float c, d;
      bf16x2_to_f32(rr[i], c, d);
      a += c;
      b += d;
      if (MODE == 2) ro[i] = pack_bf16x2(a, b);
    }
    yr[i] = pack_bf16x2(a * inv * wa, b * inv * wb);
  }
}

}  // namespace

hipError_t lfmq_rmsnorm(const void* x, const void* residual, const void* w,
                        void* y, void* res_out, int rows, int hidden,
                        float eps, hipStream_t stream) {
  if (hidden % 2) return hipErrorInvalidValue;
  dim3 grid(rows), block(TPB);
  if (residual == nullptr)
    hipLaunchKernelGGL((rmsnorm_kernel<0>), grid, block, 0, stream,
                       (const u32*)x, nullptr, (const u32*)w, (u32*)y, nullptr,
                       hidden, eps);
  else if (res_out == nullptr)
    hipLaunchKernelGGL((rmsnorm_kernel<1>), grid, block, 0, stream,
                       (const u32*)x, (const u32*)residual, (const u32*)w,
                       (u32*)y, nullptr, hidden, eps);
  else
    hipLaunchKernelGGL((rmsnorm_kernel<2>), grid, block, 0, stream,
                       (const u32*)x, (const u32*)residual, (const u32*)w,
                       (u32*)y, (u32*)res_out, hidden, eps);
  return hipGetLastError();
}
