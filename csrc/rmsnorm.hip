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
  float t = 0.f;
#pragma unroll
  for (int i = 0; i < TPB / LFMQ_WAVE; ++i) t += red[i];
  const float inv = rsqrtf(t / (float)hidden + eps);

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

// Single-WAVE variant for the decode shape (few rows): no LDS, no
// barriers — 64 lanes, values held in registers between the reduce and
// the scale pass (hidden <= 8192: <= 32 dwords/lane).  The 256-thread
// two-barrier version measured 6.1 us/call on [1,4096]; the decode step
// runs 64 of these per token.
template <int MODE, int DWPL>  // dwords per lane
__global__ __launch_bounds__(LFMQ_WAVE) void rmsnorm_wave_kernel(
    const u32* __restrict__ x, const u32* __restrict__ res_in,
    const u32* __restrict__ w, u32* __restrict__ y, u32* __restrict__ res_out,
    int hidden, float eps) {
  const int row = blockIdx.x;
  const int dw = hidden / 2;
  const int lane = threadIdx.x;
  const u32* xr = x + (u64)row * dw;
  const u32* rr = MODE ? res_in + (u64)row * dw : nullptr;
  u32* yr = y + (u64)row * dw;
  u32* ro = MODE == 2 ? res_out + (u64)row * dw : nullptr;

  float va[DWPL * 2];
  float ss = 0.f;
#pragma unroll
  for (int i = 0; i < DWPL; ++i) {
    const int idx = lane + i * LFMQ_WAVE;
    float a = 0.f, b = 0.f;
    if (idx < dw) {
      bf16x2_to_f32(xr[idx], a, b);
      if (MODE) {
        float c, d;
        bf16x2_to_f32(rr[idx], c, d);
        a += c;
        b += d;
        if (MODE == 2) ro[idx] = pack_bf16x2(a, b);
      }
    }
    va[2 * i] = a;
    va[2 * i + 1] = b;
    ss = fmaf(a, a, ss);
    ss = fmaf(b, b, ss);
  }
  ss = wave_reduce_sum(ss);
  const float inv = rsqrtf(ss / (float)hidden + eps);
#pragma unroll
  for (int i = 0; i < DWPL; ++i) {
    const int idx = lane + i * LFMQ_WAVE;
    if (idx < dw) {
      float wa, wb;
      bf16x2_to_f32(w[idx], wa, wb);
      yr[idx] = pack_bf16x2(va[2 * i] * inv * wa, va[2 * i + 1] * inv * wb);
    }
  }
}

template <int MODE>
bool launch_wave(const u32* x, const u32* r, const u32* w, u32* y, u32* ro,
                 int rows, int hidden, float eps, hipStream_t stream) {
  const int dw = hidden / 2;
#define WCASE(D)                                                           \
  if (dw <= LFMQ_WAVE * D) {                                               \
    hipLaunchKernelGGL((rmsnorm_wave_kernel<MODE, D>), dim3(rows),         \
                       dim3(LFMQ_WAVE), 0, stream, x, r, w, y, ro, hidden, \
                       eps);                                               \
    return true;                                                           \
  }
  WCASE(8)
#undef WCASE
  return false;
}

}  // namespace

hipError_t lfmq_rmsnorm(const void* x, const void* residual, const void* w,
                        void* y, void* res_out, int rows, int hidden,
                        float eps, hipStream_t stream) {
  if (hidden % 2) return hipErrorInvalidValue;
  if (rows <= 16 && hidden <= 1024) {  // tiny rows: wave-per-row path
    bool ok;
    if (residual == nullptr)
      ok = launch_wave<0>((const u32*)x, nullptr, (const u32*)w, (u32*)y,
                          nullptr, rows, hidden, eps, stream);
    else if (res_out == nullptr)
      ok = launch_wave<1>((const u32*)x, (const u32*)residual, (const u32*)w,
                          (u32*)y, nullptr, rows, hidden, eps, stream);
    else
      ok = launch_wave<2>((const u32*)x, (const u32*)residual, (const u32*)w,
                          (u32*)y, (u32*)res_out, rows, hidden, eps, stream);
    if (ok) return hipGetLastError();
  }
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
