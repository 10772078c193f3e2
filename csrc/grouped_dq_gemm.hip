// K5 — grouped dequant-GEMV over E experts with ragged per-expert slot
// counts (Mixtral MoE).  One launch covers every
// expert: grid (n-tiles, E); token indirection via src_tok (null =
// identity, used for the w2 pass whose input is already slot-major).
// Offsets/ids live on device so the MoE decode path stays hipGraph-
// capturable.  Same packed layout + (128+c) dequant as dq_gemv.hip.
//
// Round-2 rework (the round-1 kernel was
// correctness-grade, 45.7+29.1 us/layer measured vs a ~28 us BW floor):
//   - hoisted scale/zero preload (lane g holds group g, fetched in-loop
//     by one cross-lane shuffle) — the round-1 dq_gemv lesson; the old
//     in-loop scattered 2-B scale loads serialized every k-iteration.
//   - KI-unrolled weight preload (all k-iterations' W loads issued up
//     front) for the decode shapes: gateup k=4096 (KI=2, COLS=4), w2
//     k=14336 (KI=7, COLS=2).
//   - optional fused scatter epilogue (dst_tok + slot_w + fp32 y with
//     atomicAdd): the w2 pass writes the weighted per-token combine
//     directly, replacing the aten index_add_ chain.
// GATEUP: W = [gate|up] fused rows, silu(gate)·up epilogue (out n/2).
#include "common.h"

namespace {

constexpr int WAVES_PER_WG = 4;
constexpr u32 ONE2_BF16 = 0x3F803F80u;
constexpr int MCHUNK = 2;

// scatter != null variant accumulates w_slot * v into y_f32[dst_tok[slot]]
template <int BITS, bool GATEUP, int COLS, int KI, bool SCATTER>
__global__ __launch_bounds__(WAVES_PER_WG * LFMQ_WAVE) void grouped_kernel(
    const u32* __restrict__ qweight, const u32* __restrict__ scales_u16,
    const u32* __restrict__ qzeros, const u32* __restrict__ x,
    u32* __restrict__ y, const i32* __restrict__ src_tok,
    const i32* __restrict__ offsets, const i32* __restrict__ dst_tok,
    const float* __restrict__ slot_w, const i32* __restrict__ active, int n,
    int k, int g, int x_rows) {
  static_assert(!GATEUP || COLS % 2 == 0, "gateup pairs");
  // active-expert indirection (moe_route): grid.y covers only the worst-
  // case ACTIVE count (min(E, slots)), not all E experts — the dead-
  // expert workgroup dispatch was measurable at decode (8x7B top-2: 6/8
  // of the grid exited early).
  const int e = active ? active[blockIdx.y] : (int)blockIdx.y;
  if (e < 0) return;
  const int s_begin = offsets[e];
  const int s_end = offsets[e + 1];
  if (s_begin >= s_end) return;

  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  const int cbase =
      (blockIdx.x * WAVES_PER_WG + wave) * (GATEUP ? COLS / 2 : COLS);
  int col[COLS];
#pragma unroll
  for (int c = 0; c < COLS; ++c)
    col[c] = GATEUP ? ((c & 1) + cbase + (c >> 1) * (n / 2)) : cbase + c;
  if (col[0] >= n) return;

  const int kw_row_bytes = k * BITS / 8;
  const int G = k / g;
  const int per_z = 32 / BITS;
  const int zw = (G + per_z - 1) / per_z;
  const u64 expert_w_words = (u64)n * k * BITS / 32;

  __amdgpu_buffer_rsrc_t w_rsrc[COLS];
#pragma unroll
  for (int c = 0; c < COLS; ++c) {
    const int cc = col[c] < n ? col[c] : n - 1;
    w_rsrc[c] = make_rsrc_uniform(
        (const char*)(qweight + (u64)e * expert_w_words) +
            (u64)cc * kw_row_bytes,
        kw_row_bytes);
  }
  auto x_rsrc = make_rsrc(x, (u32)((u64)x_rows * k * 2));
  auto s_rsrc = make_rsrc((const char*)scales_u16 + (u64)e * n * G * 2,
                          (u32)((u64)n * G * 2));
  auto z_rsrc = make_rsrc((const char*)qzeros + (u64)e * n * zw * 4,
                          (u32)((u64)n * zw * 4));

  // hoisted scale/zero preload: lane gg holds (s, s·zbias) of group gg
  // (and gg+64 in the hi pair when G > 64) — csrc/dq_gemv.hip pattern
  float s_lo[COLS], sz_lo[COLS], s_hi[COLS], sz_hi[COLS];
#pragma unroll
  for (int c = 0; c < COLS; ++c) {
    const int cc = col[c] < n ? col[c] : n - 1;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      if (half == 1 && G <= 64) {
        s_hi[c] = 0.f;
        sz_hi[c] = 0.f;
        continue;
      }
      const int gg = lane + 64 * half;
      const u64 sidx = (u64)cc * G + gg;
      const u32 sraw = __builtin_amdgcn_raw_buffer_load_b32(
          s_rsrc, (u32)(sidx * 2) & ~3u, 0, 0);
      const float s = bf16_bits_to_f32((sidx & 1) ? (sraw >> 16)
                                                  : (sraw & 0xFFFF));
      const u32 zword = __builtin_amdgcn_raw_buffer_load_b32(
          z_rsrc, (u32)(((u64)cc * zw + gg / per_z) * 4), 0, 0);
      const u32 z = BITS == 4 ? unpack4(zword, gg % 8) : unpack8(zword, gg % 4);
      const float zb = BITS == 4 ? (float)(z + 128u) : (float)z;
      if (half == 0) {
        s_lo[c] = gg < G ? s : 0.f;
        sz_lo[c] = gg < G ? s * zb : 0.f;
      } else {
        s_hi[c] = gg < G ? s : 0.f;
        sz_hi[c] = gg < G ? s * zb : 0.f;
      }
    }
  }

  const int kiters = (k + 2047) / 2048;
  for (int s0 = s_begin; s0 < s_end; s0 += MCHUNK) {
    const int ms = min(MCHUNK, s_end - s0);
    float acc[COLS][MCHUNK];
#pragma unroll
    for (int c = 0; c < COLS; ++c)
#pragma unroll
      for (int mm = 0; mm < MCHUNK; ++mm) acc[c][mm] = 0.f;

    if (BITS == 4 && KI > 0) {
      // all k-iterations' weight loads issued up front (per-iteration
      // loads make the compiler reuse the registers and serialize on
      // memory latency — dq_gemv KI pattern); scales applied in the
      // epilogue from the preloaded lane registers
      u32x4 w_all[KI > 0 ? KI : 1][COLS];
#pragma unroll
      for (int it = 0; it < KI; ++it) {
        const u32 wbyte = (u32)(it * 1024 + lane * 16);
#pragma unroll
        for (int c = 0; c < COLS; ++c)
          w_all[it][c] =
              __builtin_amdgcn_raw_buffer_load_b128(w_rsrc[c], wbyte, 0, 2);
      }
      float cxa[COLS][MCHUNK][KI > 0 ? KI : 1];
      float xsa[MCHUNK][KI > 0 ? KI : 1];
#pragma unroll
      for (int it = 0; it < KI; ++it) {
#pragma unroll
        for (int mm = 0; mm < MCHUNK; ++mm) {
          if (mm >= ms) break;
          const int slot = s0 + mm;
          const int row = src_tok ? src_tok[slot] : slot;
          const u32 xbyte = (u32)((u64)row * k * 2 + it * 4096 + lane * 64);
          u32x4 xq0 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte, 0, 0);
          u32x4 xq1 =
              __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 16, 0, 0);
          u32x4 xq2 =
              __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 32, 0, 0);
          u32x4 xq3 =
              __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 48, 0, 0);
          float xs = 0.f;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
            xs = dot2_bf16(ONE2_BF16, xq[0], xs);
            xs = dot2_bf16(ONE2_BF16, xq[1], xs);
            xs = dot2_bf16(ONE2_BF16, xq[2], xs);
            xs = dot2_bf16(ONE2_BF16, xq[3], xs);
          }
          xsa[mm][it] = xs;
#pragma unroll
          for (int c = 0; c < COLS; ++c) {
            float cx = 0.f;
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
              const DequantPairs4 dq = dequant4_pairs(w_all[it][c][j]);
              cx = dot2_bf16(dq.p01, xq[0], cx);
              cx = dot2_bf16(dq.p23, xq[1], cx);
              cx = dot2_bf16(dq.p45, xq[2], cx);
              cx = dot2_bf16(dq.p67, xq[3], cx);
            }
            cxa[c][mm][it] = cx;
          }
        }
      }
      // scale epilogue (preloaded s/sz fetched by one shuffle)
#pragma unroll
      for (int it = 0; it < KI; ++it) {
        const int k0 = it * 2048 + lane * 32;
        const int gi = k0 < k ? k0 / g : 0;
        const bool ghi = gi >= 64;
        const int gsel = ghi ? gi - 64 : gi;
        const float live = k0 < k ? 1.f : 0.f;
#pragma unroll
        for (int c = 0; c < COLS; ++c) {
          const float s_l = __shfl(s_lo[c], gsel, 64);
          const float s_h = G > 64 ? __shfl(s_hi[c], gsel, 64) : 0.f;
          const float sz_l = __shfl(sz_lo[c], gsel, 64);
          const float sz_h = G > 64 ? __shfl(sz_hi[c], gsel, 64) : 0.f;
          const float s = (ghi ? s_h : s_l) * live;
          const float sz = (ghi ? sz_h : sz_l) * live;
#pragma unroll
          for (int mm = 0; mm < MCHUNK; ++mm) {
            if (mm >= ms) break;
            acc[c][mm] = fmaf(s, cxa[c][mm][it], acc[c][mm]);
            acc[c][mm] = fmaf(-sz, xsa[mm][it], acc[c][mm]);
          }
        }
      }
    } else {
      // general path (ragged kiters / BITS == 8)
      for (int it = 0; it < kiters; ++it) {
        const int k0 = it * 2048 + lane * 32;
        const int gi = k0 < k ? k0 / g : 0;
        const bool ghi = gi >= 64;
        const int gsel = ghi ? gi - 64 : gi;
        const float live = k0 < k ? 1.f : 0.f;
        float s[COLS], sz[COLS];
#pragma unroll
        for (int c = 0; c < COLS; ++c) {
          const float s_l = __shfl(s_lo[c], gsel, 64);
          const float s_h = G > 64 ? __shfl(s_hi[c], gsel, 64) : 0.f;
          const float sz_l = __shfl(sz_lo[c], gsel, 64);
          const float sz_h = G > 64 ? __shfl(sz_hi[c], gsel, 64) : 0.f;
          s[c] = (ghi ? s_h : s_l) * live;
          sz[c] = (ghi ? sz_h : sz_l) * live;
        }
        u32x4 w4[COLS], w8b[COLS];
#pragma unroll
        for (int c = 0; c < COLS; ++c) {
          if (BITS == 4) {
            w4[c] = __builtin_amdgcn_raw_buffer_load_b128(
                w_rsrc[c], (u32)(it * 1024 + lane * 16), 0, 2);
          } else {
            w4[c] = __builtin_amdgcn_raw_buffer_load_b128(
                w_rsrc[c], (u32)(it * 2048 + lane * 32), 0, 2);
            w8b[c] = __builtin_amdgcn_raw_buffer_load_b128(
                w_rsrc[c], (u32)(it * 2048 + lane * 32 + 16), 0, 2);
          }
        }
        for (int mm = 0; mm < MCHUNK; ++mm) {
          if (mm >= ms) break;
          const int slot = s0 + mm;
          const int row = src_tok ? src_tok[slot] : slot;
          const u32 xbyte = (u32)((u64)row * k * 2 + it * 4096 + lane * 64);
          u32x4 xq[4];
#pragma unroll
          for (int h = 0; h < 4; ++h)
            xq[h] = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc,
                                                          xbyte + 16 * h, 0, 0);
          float xs = 0.f;
#pragma unroll
          for (int h = 0; h < 4; ++h) {
            xs = dot2_bf16(ONE2_BF16, xq[h][0], xs);
            xs = dot2_bf16(ONE2_BF16, xq[h][1], xs);
            xs = dot2_bf16(ONE2_BF16, xq[h][2], xs);
            xs = dot2_bf16(ONE2_BF16, xq[h][3], xs);
          }
#pragma unroll
          for (int c = 0; c < COLS; ++c) {
            float cx = 0.f;
            if (BITS == 4) {
#pragma unroll
              for (int j = 0; j < 4; ++j) {
                const DequantPairs4 dq = dequant4_pairs(w4[c][j]);
                cx = dot2_bf16(dq.p01, xq[j][0], cx);
                cx = dot2_bf16(dq.p23, xq[j][1], cx);
                cx = dot2_bf16(dq.p45, xq[j][2], cx);
                cx = dot2_bf16(dq.p67, xq[j][3], cx);
              }
            } else {
              float xv[32];
#pragma unroll
              for (int h = 0; h < 4; ++h)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                  bf16x2_to_f32(xq[h][j], xv[h * 8 + 2 * j],
                                xv[h * 8 + 2 * j + 1]);
#pragma unroll
              for (int half = 0; half < 2; ++half) {
                const u32x4 wv = half ? w8b[c] : w4[c];
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
            acc[c][mm] = fmaf(s[c], cx, acc[c][mm]);
            acc[c][mm] = fmaf(-sz[c], xs, acc[c][mm]);
          }
        }
      }
    }

    // epilogue per slot
    for (int mm = 0; mm < MCHUNK; ++mm) {
      if (mm >= ms) break;
      const int slot = s0 + mm;
      float v[COLS];
#pragma unroll
      for (int c = 0; c < COLS; ++c) v[c] = wave_reduce_sum(acc[c][mm]);
      if (lane == 0) {
        if (GATEUP) {
          unsigned short* yrow = (unsigned short*)y;
          const int inter = n / 2;
#pragma unroll
          for (int p = 0; p < COLS / 2; ++p) {
            const int oc = cbase + p;
            if (oc < inter) {
              const float gt = v[p];
              const float sil = gt / (1.f + __expf(-gt));
              yrow[(u64)slot * inter + oc] =
                  (unsigned short)f32_to_bf16_bits_rne(sil * v[p + COLS / 2]);
            }
          }
        } else if (SCATTER) {
          // fused weighted combine: y_f32[dst_tok[slot]] += w_slot * v
          const int dst = dst_tok[slot];
          const float wsl = slot_w[slot];
          float* yrow = (float*)y + (u64)dst * n;
#pragma unroll
          for (int c = 0; c < COLS; ++c)
            if (col[c] < n) atomicAdd(&yrow[col[c]], wsl * v[c]);
        } else {
          unsigned short* yrow = (unsigned short*)y;
#pragma unroll
          for (int c = 0; c < COLS; ++c)
            if (col[c] < n)
              yrow[(u64)slot * n + col[c]] =
                  (unsigned short)f32_to_bf16_bits_rne(v[c]);
        }
      }
    }
  }
}

template <int BITS, bool GATEUP, int COLS, bool SCATTER>
void launch_ki(dim3 grid, dim3 block, hipStream_t stream, const void* qweight,
               const void* scales, const void* qzeros, const void* x, void* y,
               const void* src_tok, const void* offsets, const void* dst_tok,
               const void* slot_w, const void* active, int n, int k, int g,
               int x_rows) {
  const int kiters = (k + 2047) / 2048;
#define GKI(KIV)                                                             \
  hipLaunchKernelGGL((grouped_kernel<BITS, GATEUP, COLS, KIV, SCATTER>),     \
                     grid, block, 0, stream, (const u32*)qweight,            \
                     (const u32*)scales, (const u32*)qzeros, (const u32*)x,  \
                     (u32*)y, (const i32*)src_tok, (const i32*)offsets,      \
                     (const i32*)dst_tok, (const float*)slot_w,              \
                     (const i32*)active, n, k, g, x_rows)
  if (BITS == 4 && kiters == 2) GKI(2);
  else if (BITS == 4 && kiters == 7 && COLS <= 2) GKI(7);
  else GKI(0);
#undef GKI
}

}  // namespace

hipError_t lfmq_grouped_dq_gemv(const void* qweight, const void* scales,
                                const void* qzeros, const void* x, void* y,
                                const void* src_tok, const void* offsets,
                                const void* active, int grid_e,
                                int n_experts, int n, int k, int g, int bits,
                                int x_rows, bool gateup, hipStream_t stream) {
  if (k % 8 || g % 32 || g > k || (gateup && n % 2)) return hipErrorInvalidValue;
  (void)n_experts;
  dim3 block(WAVES_PER_WG * LFMQ_WAVE);
  if (gateup) {
    const int per_wg = WAVES_PER_WG * 2;  // COLS=4 -> 2 output pairs/wave
    dim3 grid(ceil_div_u32(n / 2, per_wg), grid_e);
    if (bits == 4)
      launch_ki<4, true, 4, false>(grid, block, stream, qweight, scales,
                                   qzeros, x, y, src_tok, offsets, nullptr,
                                   nullptr, active, n, k, g, x_rows);
    else
      launch_ki<8, true, 4, false>(grid, block, stream, qweight, scales,
                                   qzeros, x, y, src_tok, offsets, nullptr,
                                   nullptr, active, n, k, g, x_rows);
  } else {
    const int per_wg = WAVES_PER_WG * 2;  // COLS=2 (deep-k shapes: KI=7)
    dim3 grid(ceil_div_u32(n, per_wg), grid_e);
    if (bits == 4)
      launch_ki<4, false, 2, false>(grid, block, stream, qweight, scales,
                                    qzeros, x, y, src_tok, offsets, nullptr,
                                    nullptr, active, n, k, g, x_rows);
    else
      launch_ki<8, false, 2, false>(grid, block, stream, qweight, scales,
                                    qzeros, x, y, src_tok, offsets, nullptr,
                                    nullptr, active, n, k, g, x_rows);
  }
  return hipGetLastError();
}

// w2 + fused weighted scatter: y_f32 [x_rows_out, n] zero-initialized by
// the caller; y[dst_tok[slot]] += slot_w[slot] * (x[slot] . W_e)
hipError_t lfmq_grouped_dq_gemv_scatter(
    const void* qweight, const void* scales, const void* qzeros,
    const void* x, void* y_f32, const void* offsets, const void* dst_tok,
    const void* slot_w, const void* active, int grid_e, int n, int k, int g,
    int bits, int x_rows, hipStream_t stream) {
  if (k % 8 || g % 32 || g > k) return hipErrorInvalidValue;
  dim3 block(WAVES_PER_WG * LFMQ_WAVE);
  const int per_wg = WAVES_PER_WG * 2;
  dim3 grid(ceil_div_u32(n, per_wg), grid_e);
  if (bits == 4)
    launch_ki<4, false, 2, true>(grid, block, stream, qweight, scales, qzeros,
                                 x, y_f32, nullptr, offsets, dst_tok, slot_w,
                                 active, n, k, g, x_rows);
  else
    launch_ki<8, false, 2, true>(grid, block, stream, qweight, scales, qzeros,
                                 x, y_f32, nullptr, offsets, dst_tok, slot_w,
                                 active, n, k, g, x_rows);
  return hipGetLastError();
}
