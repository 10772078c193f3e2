// K3a/K4a — W4A16 / W8A16 dequant-fused GEMV for decode (m <= 16).
//
// The decode hot kernel: y[m,n] = x[m,k] · W[n,k]^T,
// W int4/int8 group-quantized (layout: lfmq/ops/ref.py).  Memory-bound on
// the packed weight stream — design targets the HBM3E roofline:
//   - one wave per output column n; lanes split k (lane l owns 32
//     consecutive k per 2048-wide k-iteration) => every wave-instruction
//     reads 1 KiB of contiguous packed weight (64 lanes x dwordx4),
//     perfectly coalesced, via hardware-bounds-checked buffer loads
//     (OOB returns 0, so ragged k needs no branches).
//   - weights are loaded non-temporal (aux=2): each line is read once by
//     one CU (MI355X_MICROARCH.md "nt-weights").
//   - int4 dequant off the critical path: v_perm assembles bf16 pairs
//     (128+c) directly from nibbles (csrc/common.h dequant4_pairs), dot
//     products via v_dot2c_f32_bf16, and the (zero+128)·scale bias is
//     folded per group:  s·Σ(c−z)x = s·(Σ(128+c)x − (128+z)·Σx).
//     ~2.3 VALU per weight => VALU stays well under the HBM pace.
//   - activations x are tiny (m·k bf16) and L1/L2-resident; they are
//     re-read per column from cache, not staged (GEMV row of the
//     "glds vs register staging" table: LDS round trip is pure overhead).
// group_size must be a multiple of 32 (one lane-chunk never spans groups).
#include "common.h"

namespace {

constexpr int WAVES_PER_WG = 4;
constexpr u32 ONE2_BF16 = 0x3F803F80u;  // bf16x2 (1.0, 1.0)

template <int M, int BITS>
__global__ __launch_bounds__(WAVES_PER_WG * LFMQ_WAVE) void dq_gemv_kernel(
    const u32* __restrict__ qweight, const u32* __restrict__ scales_u16,
    const u32* __restrict__ qzeros, const u32* __restrict__ x,
    u32* __restrict__ y, const u32* __restrict__ bias, int m, int n, int k,
    int g) {
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  const int col = blockIdx.x * WAVES_PER_WG + wave;
  if (col >= n) return;

  const int kw_row_bytes = k * BITS / 8;          // packed bytes per row
  const int G = k / g;
  const int zw = (G + (32 / BITS) - 1) / (32 / BITS);  // zeros words per row

  auto w_rsrc = make_rsrc((const char*)qweight + (u64)col * kw_row_bytes,
                          kw_row_bytes);
  auto x_rsrc = make_rsrc(x, (u32)(m * k * 2));
  const unsigned short* srow = (const unsigned short*)scales_u16 + (u64)col * G;
  const u32* zrow = qzeros + (u64)col * zw;

  float acc[M];
#pragma unroll
  for (int mm = 0; mm < M; ++mm) acc[mm] = 0.f;

  const int kiters = (k + 2047) / 2048;
  for (int it = 0; it < kiters; ++it) {
    const int k0 = it * 2048 + lane * 32;  // this lane's first k
    // group metadata (one group per lane-chunk: g % 32 == 0)
    float s = 0.f, zb = 0.f;
    if (k0 < k) {
      const int gi = k0 / g;
      s = bf16_bits_to_f32(srow[gi]);
      const u32 zword = zrow[gi / (32 / BITS)];
      const u32 z = BITS == 4 ? unpack4(zword, gi & 7) : unpack8(zword, gi & 3);
      zb = BITS == 4 ? (float)(z + 128u) : (float)z;
    }

    if (BITS == 4) {
      const u32 wbyte = (u32)(it * 1024 + lane * 16);
      u32x4 w = __builtin_amdgcn_raw_buffer_load_b128(w_rsrc, wbyte, 0, /*nt*/ 2);
      DequantPairs4 dq[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) dq[j] = dequant4_pairs(w[j]);
#pragma unroll
      for (int mm = 0; mm < M; ++mm) {
        if (mm >= m) break;
        const u32 xbyte = (u32)(mm * k * 2 + it * 4096 + lane * 64);
        float cx = 0.f, xs = 0.f;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          u32x4 xq = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 16 * j, 0, 0);
          cx = dot2_bf16(dq[j].p01, xq[0], cx);
          cx = dot2_bf16(dq[j].p23, xq[1], cx);
          cx = dot2_bf16(dq[j].p45, xq[2], cx);
          cx = dot2_bf16(dq[j].p67, xq[3], cx);
          xs = dot2_bf16(ONE2_BF16, xq[0], xs);
          xs = dot2_bf16(ONE2_BF16, xq[1], xs);
          xs = dot2_bf16(ONE2_BF16, xq[2], xs);
          xs = dot2_bf16(ONE2_BF16, xq[3], xs);
        }
        acc[mm] = fmaf(s, cx, acc[mm]);
        acc[mm] = fmaf(-s * zb, xs, acc[mm]);
      }
    } else {  // BITS == 8: byte codes via v_cvt_f32_ubyte
      const u32 wbyte = (u32)(it * 2048 + lane * 32);
      u32x4 w0 = __builtin_amdgcn_raw_buffer_load_b128(w_rsrc, wbyte, 0, 2);
      u32x4 w1 = __builtin_amdgcn_raw_buffer_load_b128(w_rsrc, wbyte + 16, 0, 2);
#pragma unroll
      for (int mm = 0; mm < M; ++mm) {
        if (mm >= m) break;
        const u32 xbyte = (u32)(mm * k * 2 + it * 4096 + lane * 64);
        float cx = 0.f, xs = 0.f;
#pragma unroll
        for (int half = 0; half < 2; ++half) {
          u32x4 wv = half ? w1 : w0;
          u32x4 xa = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 32 * half, 0, 0);
          u32x4 xb = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 32 * half + 16, 0, 0);
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            u32 xd0 = (j < 2) ? xa[2 * j] : xb[2 * (j - 2)];
            u32 xd1 = (j < 2) ? xa[2 * j + 1] : xb[2 * (j - 2) + 1];
            float x0, x1, x2, x3;
            bf16x2_to_f32(xd0, x0, x1);
            bf16x2_to_f32(xd1, x2, x3);
            // (float)(byte) lowers to v_cvt_f32_ubyte0..3 on gfx950
            cx = fmaf((float)(wv[j] & 0xFFu), x0, cx);
            cx = fmaf((float)((wv[j] >> 8) & 0xFFu), x1, cx);
            cx = fmaf((float)((wv[j] >> 16) & 0xFFu), x2, cx);
            cx = fmaf((float)(wv[j] >> 24), x3, cx);
            xs += x0 + x1 + x2 + x3;
          }
        }
        acc[mm] = fmaf(s, cx, acc[mm]);
        acc[mm] = fmaf(-s * zb, xs, acc[mm]);
      }
    }
  }

  // wave-reduce each row's partial and store
  float b = 0.f;
  if (bias != nullptr && lane == 0) {
    const unsigned short* brow = (const unsigned short*)bias;
    b = bf16_bits_to_f32(brow[col]);
  }
#pragma unroll
  for (int mm = 0; mm < M; ++mm) {
    if (mm >= m) break;
    float v = wave_reduce_sum(acc[mm]);
    if (lane == 0) {
      unsigned short* yrow = (unsigned short*)y;
      yrow[(u64)mm * n + col] = (unsigned short)f32_to_bf16_bits_rne(v + b);
    }
  }
}

template <int BITS>
hipError_t launch_bits(const void* qweight, const void* scales, const void* qzeros,
                       const void* x, void* y, const void* bias, int m, int n,
                       int k, int g, hipStream_t stream) {
  dim3 grid(ceil_div_u32(n, WAVES_PER_WG));
  dim3 block(WAVES_PER_WG * LFMQ_WAVE);
#define LFMQ_CASE(MM)                                                          \
  hipLaunchKernelGGL((dq_gemv_kernel<MM, BITS>), grid, block, 0, stream,       \
                     (const u32*)qweight, (const u32*)scales,                  \
                     (const u32*)qzeros, (const u32*)x, (u32*)y,               \
                     (const u32*)bias, m, n, k, g)
  if (m <= 1) LFMQ_CASE(1);
  else if (m <= 2) LFMQ_CASE(2);
  else if (m <= 4) LFMQ_CASE(4);
  else if (m <= 8) LFMQ_CASE(8);
  else LFMQ_CASE(16);
#undef LFMQ_CASE
  return hipGetLastError();
}

}  // namespace

hipError_t lfmq_dq_gemv(const void* qweight, const void* scales,
                        const void* qzeros, const void* x, void* y,
                        const void* bias, int m, int n, int k, int g, int bits,
                        hipStream_t stream) {
  if (m > 16 || k % 8 != 0 || g % 32 != 0 || g > k) return hipErrorInvalidValue;
  if (bits == 4)
    return launch_bits<4>(qweight, scales, qzeros, x, y, bias, m, n, k, g, stream);
  if (bits == 8)
    return launch_bits<8>(qweight, scales, qzeros, x, y, bias, m, n, k, g, stream);
  return hipErrorInvalidValue;
}
