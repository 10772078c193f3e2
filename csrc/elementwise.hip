// K9/K10/K11 — silu·mul, kv-cache append, residual add.  All HBM-bound:
// bf16 processed 2..8-wide per thread (G13), grid-stride, capped grids.
#include "common.h"

namespace {

// silu(gate) ⊙ up.  gate/up are the two halves of one fused gate_up
// projection output [rows, 2*inter] (stride_dw between a row's gate and
// its up part = inter/2 dwords), writing [rows, inter].
__global__ void silu_mul_kernel(const u32* __restrict__ gu, u32* __restrict__ y,
                                u64 rows, int inter_dw) {
  const u64 total = rows * inter_dw;
  for (u64 idx = (u64)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (u64)gridDim.x * blockDim.x) {
    const u64 row = idx / inter_dw;
    const int j = idx % inter_dw;
    const u32 gd = gu[row * 2 * inter_dw + j];
    const u32 ud = gu[row * 2 * inter_dw + inter_dw + j];
    float g0, g1, u0, u1;
    bf16x2_to_f32(gd, g0, g1);
    bf16x2_to_f32(ud, u0, u1);
    // silu(x) = x / (1 + e^-x); __expf is plenty at bf16 output precision
    const float s0 = g0 / (1.f + __expf(-g0));
    const float s1 = g1 / (1.f + __expf(-g1));
    y[idx] = pack_bf16x2(s0 * u0, s1 * u1);
  }
}

// separate-tensor variant (gate and up not interleaved)
__global__ void silu_mul2_kernel(const u32* __restrict__ g,
                                 const u32* __restrict__ u, u32* __restrict__ y,
                                 u64 total_dw) {
  for (u64 idx = (u64)blockIdx.x * blockDim.x + threadIdx.x; idx < total_dw;
       idx += (u64)gridDim.x * blockDim.x) {
    float g0, g1, u0, u1;
    bf16x2_to_f32(g[idx], g0, g1);
    bf16x2_to_f32(u[idx], u0, u1);
    y[idx] = pack_bf16x2(g0 / (1.f + __expf(-g0)) * u0,
                         g1 / (1.f + __expf(-g1)) * u1);
  }
}

__global__ void residual_add_kernel(const u32* __restrict__ a,
                                    const u32* __restrict__ b,
                                    u32* __restrict__ y, u64 total_dw) {
  for (u64 idx = (u64)blockIdx.x * blockDim.x + threadIdx.x; idx < total_dw;
       idx += (u64)gridDim.x * blockDim.x) {
    float a0, a1, b0, b1;
    bf16x2_to_f32(a[idx], a0, a1);
    bf16x2_to_f32(b[idx], b0, b1);
    y[idx] = pack_bf16x2(a0 + b0, a1 + b1);
  }
}

// scatter new k,v [b, s, hkv, d] (as produced by the qkv split, token-major)
// into caches [b, hkv, T_max, d] at positions [pos0, pos0+s)
__global__ void kv_append_kernel(u32* __restrict__ kc, u32* __restrict__ vc,
                                 const u32* __restrict__ kn,
                                 const u32* __restrict__ vn, int b, int s,
                                 int hkv, int d, int tmax, int pos0) {
  const int dw = d / 2;
  const u64 total = (u64)b * s * hkv * dw;
  for (u64 idx = (u64)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (u64)gridDim.x * blockDim.x) {
    const int j = idx % dw;
    u64 r = idx / dw;
    const int h = r % hkv;
    r /= hkv;
    const int si = r % s;
    const int bb = r / s;
    const u64 src = (((u64)bb * s + si) * hkv + h) * dw + j;
    const u64 dst = (((u64)bb * hkv + h) * tmax + pos0 + si) * dw + j;
    kc[dst] = kn[src];
    vc[dst] = vn[src];
  }
}

u32 grid_for(u64 total) {
  u64 b = (total + 255) / 256;
  if (b > 2048) b = 2048;
  if (b == 0) b = 1;
  return (u32)b;
}

}  // namespace

hipError_t lfmq_silu_mul_fused(const void* gate_up, void* y, long rows,
                               int inter, hipStream_t stream) {
  if (inter % 2) return hipErrorInvalidValue;
  u64 total = (u64)rows * (inter / 2);
  hipLaunchKernelGGL(silu_mul_kernel, dim3(grid_for(total)), dim3(256), 0,
                     stream, (const u32*)gate_up, (u32*)y, (u64)rows, inter / 2);
  return hipGetLastError();
}

hipError_t lfmq_silu_mul(const void* g, const void* u, void* y, long total,
                         hipStream_t stream) {
  if (total % 2) return hipErrorInvalidValue;
  hipLaunchKernelGGL(silu_mul2_kernel, dim3(grid_for(total / 2)), dim3(256), 0,
                     stream, (const u32*)g, (const u32*)u, (u32*)y,
                     (u64)(total / 2));
  return hipGetLastError();
}

hipError_t lfmq_residual_add(const void* a, const void* b, void* y, long total,
                             hipStream_t stream) {
  if (total % 2) return hipErrorInvalidValue;
  hipLaunchKernelGGL(residual_add_kernel, dim3(grid_for(total / 2)), dim3(256),
                     0, stream, (const u32*)a, (const u32*)b, (u32*)y,
                     (u64)(total / 2));
  return hipGetLastError();
}

hipError_t lfmq_kv_append(void* kc, void* vc, const void* kn, const void* vn,
                          int b, int s, int hkv, int d, int tmax, int pos0,
                          hipStream_t stream) {
  if (d % 2) return hipErrorInvalidValue;
  u64 total = (u64)b * s * hkv * (d / 2);
  hipLaunchKernelGGL(kv_append_kernel, dim3(grid_for(total)), dim3(256), 0,
                     stream, (u32*)kc, (u32*)vc, (const u32*)kn,
                     (const u32*)vn, b, s, hkv, d, tmax, pos0);
  return hipGetLastError();
}
