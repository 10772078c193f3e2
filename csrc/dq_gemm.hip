// K3b/K4b — W4A16 / W8A16 dequant-fused GEMM for prefill/eval (m large).
//
// y[m,n] = x[m,k] · dequant(W)[n,k]^T on MFMA (mfma_f32_16x16x32_bf16):
// 128x128 output tile, BK=64, 8 waves (2M x 4N, 64x32 per wave).
// B (weights) are dequantized int4/int8 -> bf16 IN THE STAGING PASS and
// written to LDS with the T2 XOR swizzle (byte ^= (row&7)<<4) so the
// 16-lane ds_read_b128 fragment reads are conflict-free; activations
// stage through the same swizzled layout.  bf16 activations through LDS
// per the north star (BASELINE.json:5).  Double-buffered issue-early /
// write-late schedule (guide §6 G15 / T14): one barrier per k-tile, the
// next tile's global loads in flight across the MFMAs — measured
// 386-400 TF/s with the dequant fused into staging.
// Constraints: k % 64 == 0, group_size % 64 == 0.
#include "common.h"

namespace {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int TPB = 512;
typedef s16x8 frag_ab;

__device__ __forceinline__ u32 swz(u32 byte_off, u32 row) {
  return byte_off ^ ((row & 7u) << 4);
}

template <int BITS>
__global__ __launch_bounds__(TPB, 1) void dq_gemm_kernel(
    const u32* __restrict__ qweight, const u32* __restrict__ scales_u16,
    const u32* __restrict__ qzeros, const u32* __restrict__ x,
    u32* __restrict__ y, const u32* __restrict__ bias, int m, int n, int k,
    int g) {
  // LDS: 2 x (A [BM][BK] + B [BN][BK]) bf16 (double-buffered, 64 KiB)
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* ldsA = smem;
  char* ldsB = smem + BM * BK * 2;

  const u32 nbx = (n + BN - 1) / BN;
  const u32 bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int m0 = (bid / nbx) * BM;
  const int n0 = (bid % nbx) * BN;

  const int tid = threadIdx.x;
  const int wave = tid / LFMQ_WAVE;
  const int lane = tid % LFMQ_WAVE;
  const int wm = wave / 4, wn = wave % 4;  // 2 x 4 wave grid

  const int G = k / g;
  const int zper = 32 / BITS;
  const int zw = (G + zper - 1) / zper;

  // A staging coords: thread -> (row r, 16B slot) : 128 rows x 4 slots
  const int ar = tid / 4;
  const int aslot = tid % 4;
  const int arow_g = min(m0 + ar, m - 1);  // clamped: OOB rows unstored
  auto x_rsrc = make_rsrc(x, (u32)min((u64)m * k * 2, (u64)0xFFFFFFF0));

  // B staging coords: thread -> (col c, 8B pair) : 128 cols x 4 pairs
  const int bc = tid / 4;
  const int bpair = tid % 4;
  const int bcol_g = min(n0 + bc, n - 1);

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ksteps = k / BK;

  // Double-buffered T14 schedule (guide §6 G15): per k-tile, ONE barrier;
  // the next tile's global loads are ISSUED before the MFMAs and the
  // LDS writes land after them into the other buffer (legal: that
  // buffer's last readers passed this iteration's barrier).
  char* ldsA2[2] = {ldsA, ldsA + (BM + BN) * BK * 2};
  char* ldsB2[2] = {ldsB, ldsB + (BM + BN) * BK * 2};

  auto stage_a_load = [&](int kt, u32x4 va[2]) {
#pragma unroll
    for (int s2 = 0; s2 < 2; ++s2) {
      const int slot = aslot + 4 * s2;
      const u32 gbyte = (u32)(((u64)arow_g * k + kt * BK + slot * 8) * 2);
      va[s2] = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, gbyte, 0, 0);
    }
  };
  auto stage_a_write = [&](int buf, const u32x4 va[2]) {
#pragma unroll
    for (int s2 = 0; s2 < 2; ++s2) {
      const int slot = aslot + 4 * s2;
      *(u32x4*)(ldsA2[buf] + swz((u32)(ar * BK * 2 + slot * 16), ar)) = va[s2];
    }
  };
  auto stage_b = [&](int kt, int buf) {
    const int kbase = kt * BK + bpair * 16;
    const int gi = kbase / g;
    const float s = bf16_bits_to_f32(
        ((const unsigned short*)scales_u16)[(u64)bcol_g * G + gi]);
    const u32 zword = qzeros[(u64)bcol_g * zw + gi / zper];
    const float zb = BITS == 4 ? (float)(unpack4(zword, gi & 7) + 128u)
                               : (float)unpack8(zword, gi & 3);
    u32 out[8];
    if (BITS == 4) {
      const u32* wrow = qweight + (u64)bcol_g * (k / 8);
      const u32 w0 = wrow[kbase / 8], w1 = wrow[kbase / 8 + 1];
      DequantPairs4 d0 = dequant4_pairs(w0), d1 = dequant4_pairs(w1);
      const u32 ps[8] = {d0.p01, d0.p23, d0.p45, d0.p67,
                         d1.p01, d1.p23, d1.p45, d1.p67};
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float lo, hi;
        bf16x2_to_f32(ps[i], lo, hi);
        out[i] = pack_bf16x2(fmaf(s, lo, -s * zb), fmaf(s, hi, -s * zb));
      }
    } else {
      const u32* wrow = qweight + (u64)bcol_g * (k / 4);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const u32 wv = wrow[kbase / 4 + i];
        const float c0 = (float)(wv & 0xFFu), c1 = (float)((wv >> 8) & 0xFFu);
        const float c2 = (float)((wv >> 16) & 0xFFu), c3 = (float)(wv >> 24);
        out[2 * i] = pack_bf16x2(s * (c0 - zb), s * (c1 - zb));
        out[2 * i + 1] = pack_bf16x2(s * (c2 - zb), s * (c3 - zb));
      }
    }
    char* dst0 = ldsB2[buf] + swz((u32)(bc * BK * 2 + bpair * 32), bc);
    *(u32x4*)dst0 = u32x4{out[0], out[1], out[2], out[3]};
    *(u32x4*)(ldsB2[buf] + swz((u32)(bc * BK * 2 + bpair * 32 + 16), bc)) =
        u32x4{out[4], out[5], out[6], out[7]};
  };
  auto mfma_tile = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int k8 = ks * 32 + (lane >> 4) * 8;
      frag_ab a[4], b[2];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wm * 64 + i * 16 + (lane & 15);
        a[i] = *(frag_ab*)(ldsA2[buf] + swz((u32)(row * BK * 2 + k8 * 2), row));
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int col = wn * 32 + j * 16 + (lane & 15);
        b[j] = *(frag_ab*)(ldsB2[buf] + swz((u32)(col * BK * 2 + k8 * 2), col));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
  };

  {
    u32x4 va[2];
    stage_a_load(0, va);
    stage_a_write(0, va);
    stage_b(0, 0);
  }
  int cur = 0;
  for (int kt = 0; kt < ksteps; ++kt) {
    __syncthreads();  // buf[cur] staged; buf[cur^1]'s last readers done
    u32x4 va[2];
    if (kt + 1 < ksteps) stage_a_load(kt + 1, va);  // issue early
    mfma_tile(cur);
    if (kt + 1 < ksteps) {                          // write late
      stage_a_write(cur ^ 1, va);
      stage_b(kt + 1, cur ^ 1);
    }
    cur ^= 1;
  }

  // ---- epilogue: C[m0+..][n0+..] (C/D map: col=l&15, row=(l>>4)*4+r) --
  const int ccol = n0 + wn * 32 + (lane & 15);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int col = ccol + j * 16;
      if (col >= n) continue;
      float badd = 0.f;
      if (bias) badd = bf16_bits_to_f32(((const unsigned short*)bias)[col]);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm * 64 + i * 16 + (lane >> 4) * 4 + r;
        if (row < m)
          ((unsigned short*)y)[(u64)row * n + col] =
              (unsigned short)f32_to_bf16_bits_rne(acc[i][j][r] + badd);
      }
    }
  }
}

}  // namespace

hipError_t lfmq_dq_gemm(const void* qweight, const void* scales,
                        const void* qzeros, const void* x, void* y,
                        const void* bias, int m, int n, int k, int g, int bits,
                        hipStream_t stream) {
  if (k % BK || g % BK || g > k) return hipErrorInvalidValue;
  const u32 nbx = ceil_div_u32(n, BN), nby = ceil_div_u32(m, BM);
  dim3 grid(nbx * nby), block(TPB);
  const size_t lds = 2 * (BM + BN) * BK * 2;  // double-buffered
  if (bits == 4)
    hipLaunchKernelGGL((dq_gemm_kernel<4>), grid, block, lds, stream,
                       (const u32*)qweight, (const u32*)scales,
                       (const u32*)qzeros, (const u32*)x, (u32*)y,
                       (const u32*)bias, m, n, k, g);
  else if (bits == 8)
    hipLaunchKernelGGL((dq_gemm_kernel<8>), grid, block, lds, stream,
                       (const u32*)qweight, (const u32*)scales,
                       (const u32*)qzeros, (const u32*)x, (u32*)y,
                       (const u32*)bias, m, n, k, g);
  else
    return hipErrorInvalidValue;
  return hipGetLastError();
}
