// K5 — grouped dequant-GEMV over E experts with ragged per-expert slot
// counts (Mixtral MoE).  One launch covers every
// expert: grid (n-tiles, E); workgroup (4 waves, 4 columns each) streams
// expert e's packed rows once and loops its slot range in chunks of 8.
// Token indirection: slot s reads activation row src_tok[s] (null =
// identity, used for the w2 pass whose input is already slot-major).
// GATEUP: W = [gate|up] fused rows, silu(gate)·up epilogue (out n/2).
// Offsets/ids live on device so the MoE decode path stays hipGraph-
// capturable.  Same packed layout + (128+c) dequant as dq_gemv.hip.
#include "common.h"

namespace {

constexpr int WAVES_PER_WG = 4;
constexpr int MCHUNK = 8;
constexpr u32 ONE2_BF16 = 0x3F803F80u;

template <int BITS, bool GATEUP>
__global__ __launch_bounds__(WAVES_PER_WG * LFMQ_WAVE) void grouped_kernel(
    const u32* __restrict__ qweight, const u32* __restrict__ scales_u16,
    const u32* __restrict__ qzeros, const u32* __restrict__ x,
    u32* __restrict__ y, const i32* __restrict__ src_tok,
    const i32* __restrict__ offsets, int n, int k, int g, int x_rows) {
  const int e = blockIdx.y;
  const int s_begin = offsets[e];
  const int s_end = offsets[e + 1];
  if (s_begin >= s_end) return;

  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  const int cbase = (blockIdx.x * WAVES_PER_WG + wave) * (GATEUP ? 2 : 4);
  int col[4];
#pragma unroll
  for (int c = 0; c < 4; ++c)
    col[c] = GATEUP ? ((c & 1) + cbase + (c >> 1) * (n / 2)) : cbase + c;
  if (col[0] >= n) return;

  const int kw_row_bytes = k * BITS / 8;
  const int G = k / g;
  const int per_z = 32 / BITS;
  const int zw = (G + per_z - 1) / per_z;
  const u64 expert_w_words = (u64)n * k * BITS / 32;

  __amdgpu_buffer_rsrc_t w_rsrc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int cc = col[c] < n ? col[c] : n - 1;
    w_rsrc[c] = make_rsrc_uniform(
        (const char*)(qweight + (u64)e * expert_w_words) + (u64)cc * kw_row_bytes,
        kw_row_bytes);
  }
  auto x_rsrc = make_rsrc(x, (u32)((u64)x_rows * k * 2));
  const unsigned short* srow =
      (const unsigned short*)scales_u16 + (u64)e * n * G;
  const u32* zrow = qzeros + (u64)e * n * zw;

  const int kiters = (k + 2047) / 2048;
  for (int s0 = s_begin; s0 < s_end; s0 += MCHUNK) {
    const int ms = min(MCHUNK, s_end - s0);
    float acc[4][MCHUNK];
#pragma unroll
    for (int c = 0; c < 4; ++c)
#pragma unroll
      for (int mm = 0; mm < MCHUNK; ++mm) acc[c][mm] = 0.f;

    for (int it = 0; it < kiters; ++it) {
      const int k0 = it * 2048 + lane * 32;
      const int gi = k0 < k ? k0 / g : 0;
      float s[4], sz[4];
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        const int cc = col[c] < n ? col[c] : n - 1;
        const float sv =
            k0 < k ? bf16_bits_to_f32(srow[(u64)cc * G + gi]) : 0.f;
        const u32 zword = zrow[(u64)cc * zw + gi / per_z];
        const u32 z = BITS == 4 ? unpack4(zword, gi % 8) : unpack8(zword, gi % 4);
        s[c] = sv;
        sz[c] = sv * (BITS == 4 ? (float)(z + 128u) : (float)z);
      }
      u32x4 w[4];
      const u32 wbyte4 = (u32)(it * 1024 + lane * 16);
#pragma unroll
      for (int c = 0; c < 4; ++c)
        w[c] = BITS == 4
                   ? __builtin_amdgcn_raw_buffer_load_b128(w_rsrc[c], wbyte4, 0, 2)
                   : u32x4{0, 0, 0, 0};
      for (int mm = 0; mm < MCHUNK; ++mm) {
        if (mm >= ms) break;
        const int slot = s0 + mm;
        const int row = src_tok ? src_tok[slot] : slot;
        const u32 xbyte = (u32)((u64)row * k * 2 + it * 4096 + lane * 64);
        u32x4 xq0 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte, 0, 0);
        u32x4 xq1 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 16, 0, 0);
        u32x4 xq2 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 32, 0, 0);
        u32x4 xq3 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 48, 0, 0);
        float xs = 0.f;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
          xs = dot2_bf16(ONE2_BF16, xq[0], xs);
          xs = dot2_bf16(ONE2_BF16, xq[1], xs);
          xs = dot2_bf16(ONE2_BF16, xq[2], xs);
          xs = dot2_bf16(ONE2_BF16, xq[3], xs);
        }
        const float live = k0 < k ? 1.f : 0.f;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          float cx = 0.f;
          if (BITS == 4) {
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
              const DequantPairs4 dq = dequant4_pairs(w[c][j]);
              cx = dot2_bf16(dq.p01, xq[0], cx);
              cx = dot2_bf16(dq.p23, xq[1], cx);
              cx = dot2_bf16(dq.p45, xq[2], cx);
              cx = dot2_bf16(dq.p67, xq[3], cx);
            }
          } else {  // BITS == 8: two halves of 16B
            const u32 wbyte8 = (u32)(it * 2048 + lane * 32);
            u32x4 w0 = __builtin_amdgcn_raw_buffer_load_b128(w_rsrc[c], wbyte8, 0, 2);
            u32x4 w1 = __builtin_amdgcn_raw_buffer_load_b128(w_rsrc[c], wbyte8 + 16, 0, 2);
            float xv[32];
            const u32x4 xa[4] = {xq0, xq1, xq2, xq3};
#pragma unroll
            for (int h = 0; h < 4; ++h)
#pragma unroll
              for (int j = 0; j < 4; ++j)
                bf16x2_to_f32(xa[h][j], xv[h * 8 + 2 * j], xv[h * 8 + 2 * j + 1]);
#pragma unroll
            for (int half = 0; half < 2; ++half) {
              const u32x4 wv = half ? w1 : w0;
#pragma unroll
              for (int j = 0; j < 4; ++j) {
                const int t4 = half * 16 + j * 4;
                cx = fmaf((float)(wv[j] & 0xFFu), xv[t4], cx);
                cx = fmaf((float)((wv[j] >> 8) & 0xFFu), xv[t4 + 1], cx);
                cx = fmaf((float)((wv[j] >> 16) & 0xFFu), xv[t4 + 2], cx);
                cx = fmaf((float)(wv[j] >> 24), xv[t4 + 3], cx);
              }
            }
          }
          acc[c][mm] = fmaf(s[c] * live, cx, acc[c][mm]);
          acc[c][mm] = fmaf(-sz[c] * live, xs, acc[c][mm]);
        }
      }
    }

    // epilogue per slot
    for (int mm = 0; mm < MCHUNK; ++mm) {
      if (mm >= ms) break;
      const int slot = s0 + mm;
      float v[4];
#pragma unroll
      for (int c = 0; c < 4; ++c) v[c] = wave_reduce_sum(acc[c][mm]);
      if (lane == 0) {
        unsigned short* yrow = (unsigned short*)y;
        if (GATEUP) {
          const int inter = n / 2;
#pragma unroll
          for (int p = 0; p < 2; ++p) {
            const int oc = cbase + p;
            if (oc < inter) {
              const float gt = v[p];
              const float sil = gt / (1.f + __expf(-gt));
              yrow[(u64)slot * inter + oc] =
                  (unsigned short)f32_to_bf16_bits_rne(sil * v[p + 2]);
            }
          }
        } else {
#pragma unroll
          for (int c = 0; c < 4; ++c)
            if (col[c] < n)
              yrow[(u64)slot * n + col[c]] =
                  (unsigned short)f32_to_bf16_bits_rne(v[c]);
        }
      }
    }
  }
}

}  // namespace

hipError_t lfmq_grouped_dq_gemv(const void* qweight, const void* scales,
                                const void* qzeros, const void* x, void* y,
                                const void* src_tok, const void* offsets,
                                int n_experts, int n, int k, int g, int bits,
                                int x_rows, bool gateup, hipStream_t stream) {
  if (k % 8 || g % 32 || g > k || (gateup && n % 2)) return hipErrorInvalidValue;
  const int per_wg = WAVES_PER_WG * (gateup ? 2 : 4);
  dim3 grid(ceil_div_u32(gateup ? n / 2 : n, per_wg), n_experts);
  dim3 block(WAVES_PER_WG * LFMQ_WAVE);
#define GCASE(B, GU)                                                        \
  hipLaunchKernelGGL((grouped_kernel<B, GU>), grid, block, 0, stream,       \
                     (const u32*)qweight, (const u32*)scales,               \
                     (const u32*)qzeros, (const u32*)x, (u32*)y,            \
                     (const i32*)src_tok, (const i32*)offsets, n, k, g,     \
                     x_rows)
  if (bits == 4 && gateup) GCASE(4, true);
  else if (bits == 4) GCASE(4, false);
  else if (bits == 8 && gateup) GCASE(8, true);
  else if (bits == 8) GCASE(8, false);
  else return hipErrorInvalidValue;
#undef GCASE
  return hipGetLastError();
}
