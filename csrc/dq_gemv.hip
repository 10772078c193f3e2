// K3a/K4a — W4A16 / W8A16 dequant-fused GEMV for decode (m <= 16).
//
// The decode hot kernel: y[m,n] = x[m,k] · W[n,k]^T,
// W int4/int8 group-quantized (layout: lfmq/ops/ref.py).  Memory-bound on
// the packed weight stream; measured design history on MI355X:
//   - per-wave buffer descriptors MUST be readfirstlane-uniform
//     (csrc/common.h make_rsrc_uniform) or hipcc waterfalls every load
//     (T20): 13 us/call -> 6.5.
//   - a raw-read probe of this exact pattern (csrc/probe.hip) sustains
//     6.3-6.7 TB/s, so the remaining gap was the per-iteration SCATTERED
//     2-byte scale + zero loads (8 TA-heavy lane-scattered loads per
//     k-iteration).  They are hoisted: each column's whole scale/zero
//     row is preloaded ONCE, coalesced, into per-lane registers (lane g
//     holds group g) and fetched in-loop by a single cross-lane shuffle.
//   - each wave owns COLS(=4) columns; lanes split k (lane l owns 32
//     consecutive k per 2048-wide k-iteration): every wave-instruction
//     reads 1 KiB of contiguous packed weight via hardware-bounds-
//     checked buffer loads (OOB -> 0: ragged k needs no branches),
//     non-temporal (each line is read once by one CU).
//   - int4 dequant off the critical path: v_perm assembles bf16 pairs
//     (128+c) straight from nibbles, dots via v_dot2c_f32_bf16, and the
//     (zero+128)·scale bias is folded per group:
//       s·Σ(c−z)x = s·(Σ(128+c)x − (128+z)·Σx),
//     Σx computed once per k-chunk and shared by all COLS columns.
//   - activations x are tiny and L1/L2-resident: read per wave
//     k-iteration from cache, never staged (GEMV row of the guide's
//     "glds vs register staging" table).
// GATEUP variant (K9 fused into the K3a epilogue): the wave's columns
// are {c, c+1} of gate and {inter+c, inter+c+1} of up of one fused
// gate_up projection; the epilogue writes silu(gate)·up directly.
// group_size must be a multiple of 32 (one lane-chunk never spans groups).
#include "common.h"

namespace {

constexpr int WAVES_PER_WG = 4;
constexpr u32 ONE2_BF16 = 0x3F803F80u;  // bf16x2 (1.0, 1.0)

template <int M, int BITS, int COLS, bool GATEUP, int KI, bool RMS,
          bool KSPLIT = false>
__global__ __launch_bounds__(WAVES_PER_WG * LFMQ_WAVE) void dq_gemv_kernel(
    const u32* __restrict__ qweight, const u32* __restrict__ scales_u16,
    const u32* __restrict__ qzeros, const u32* __restrict__ x,
    u32* __restrict__ y, const u32* __restrict__ bias,
    const u32* __restrict__ resid, int m, int n, int k, int g, float eps) {
  static_assert(!GATEUP || COLS == 4 || COLS == 2, "gateup needs COLS 2/4");
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  // KSPLIT: all 4 waves share the SAME columns and split the k-iteration
  // range (the decode GEMV is latency-bound on the per-wave serial
  // k-chain — SQ_WAIT_ANY 62% measured); partials combine through LDS.
  const int cbase = (KSPLIT ? (int)blockIdx.x
                            : ((int)blockIdx.x * WAVES_PER_WG + wave)) *
                    (GATEUP ? COLS / 2 : COLS);
  int col[COLS];
#pragma unroll
  for (int c = 0; c < COLS; ++c) {
    if (GATEUP && COLS == 2)
      col[c] = cbase + c * (n / 2);          // {gate c, up c}
    else if (GATEUP)
      col[c] = (c & 1) + cbase + (c >> 1) * (n / 2);  // {g c, g c+1, u c, u c+1}
    else
      col[c] = cbase + c;
  }
  if (col[0] >= n) return;

  const int kw_row_bytes = k * BITS / 8;
  const int G = k / g;
  const int per_z = 32 / BITS;
  const int zw = (G + per_z - 1) / per_z;

  __amdgpu_buffer_rsrc_t w_rsrc[COLS];
#pragma unroll
  for (int c = 0; c < COLS; ++c) {
    const int cc = col[c] < n ? col[c] : n - 1;
    w_rsrc[c] = make_rsrc_uniform((const char*)qweight + (u64)cc * kw_row_bytes,
                                  kw_row_bytes);
  }
  auto x_rsrc = make_rsrc(x, (u32)(m * k * 2));
  auto s_rsrc = make_rsrc(scales_u16, (u32)((u64)n * G * 2));
  auto z_rsrc = make_rsrc(qzeros, (u32)((u64)n * zw * 4));

  // ---- hoisted scale/zero preload: lane gg holds (s, s·zbias) of
  // group gg (and gg+64 in the hi pair when G > 64) --------------------
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
      const float s = bf16_bits_to_f32((sidx & 1) ? (sraw >> 16) : (sraw & 0xFFFF));
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

  float acc[COLS][M];
#pragma unroll
  for (int c = 0; c < COLS; ++c)
#pragma unroll
    for (int mm = 0; mm < M; ++mm) acc[c][mm] = 0.f;

  float ss_acc[M];  // Σx² per row (RMS mode)
#pragma unroll
  for (int mm = 0; mm < M; ++mm) ss_acc[mm] = 0.f;

  const int kiters = (k + 2047) / 2048;
  int it_begin = 0, it_end = kiters;
  if (KSPLIT) {
    const int per_wave = (kiters + WAVES_PER_WG - 1) / WAVES_PER_WG;
    it_begin = wave * per_wave;
    it_end = min(kiters, it_begin + per_wave);
  }

  if (BITS == 4 && KI > 0 && !KSPLIT) {
    // Fully-unrolled k-loop: per-iteration (cx, xs) partials accumulate in
    // registers; scales are applied ONLY in the epilogue, so the hoisted
    // scale/zero preload never blocks the weight stream (probe.hip mode 2
    // shows the full dequant+dot loop sustains >6 TB/s once unserialized).
    float cxa[COLS][M][KI > 0 ? KI : 1], xsa[M][KI > 0 ? KI : 1];
    // issue EVERY iteration's weight loads up front: with per-iteration
    // loads the compiler reuses the W registers and the loop serializes
    // on memory latency (SQ_WAIT_ANY 62% measured on the k=14336 shape)
    u32x4 w_all[KI > 0 ? KI : 1][COLS];
#pragma unroll
    for (int it = 0; it < KI; ++it) {
      const u32 wbyte = (u32)(it * 1024 + lane * 16);
#pragma unroll
      for (int c = 0; c < COLS; ++c)
        w_all[it][c] =
            __builtin_amdgcn_raw_buffer_load_b128(w_rsrc[c], wbyte, 0, 2);
    }
#pragma unroll
    for (int it = 0; it < KI; ++it) {
      u32x4 w[COLS];
#pragma unroll
      for (int c = 0; c < COLS; ++c) w[c] = w_all[it][c];
#pragma unroll
      for (int mm = 0; mm < M; ++mm) {
        if (mm >= m) break;
        const u32 xbyte = (u32)(mm * k * 2 + it * 4096 + lane * 64);
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
          if (RMS) {
            ss_acc[mm] = dot2_bf16(xq[0], xq[0], ss_acc[mm]);
            ss_acc[mm] = dot2_bf16(xq[1], xq[1], ss_acc[mm]);
            ss_acc[mm] = dot2_bf16(xq[2], xq[2], ss_acc[mm]);
            ss_acc[mm] = dot2_bf16(xq[3], xq[3], ss_acc[mm]);
          }
        }
        xsa[mm][it] = xs;
#pragma unroll
        for (int c = 0; c < COLS; ++c) {
          float cx = 0.f;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
            const DequantPairs4 dq = dequant4_pairs(w[c][j]);
            cx = dot2_bf16(dq.p01, xq[0], cx);
            cx = dot2_bf16(dq.p23, xq[1], cx);
            cx = dot2_bf16(dq.p45, xq[2], cx);
            cx = dot2_bf16(dq.p67, xq[3], cx);
          }
          cxa[c][mm][it] = cx;
        }
      }
    }
    // scale epilogue (the preloaded s/sz registers are ready by now)
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
        for (int mm = 0; mm < M; ++mm) {
          if (mm >= m) break;
          acc[c][mm] = fmaf(s, cxa[c][mm][it], acc[c][mm]);
          acc[c][mm] = fmaf(-sz, xsa[mm][it], acc[c][mm]);
        }
      }
    }
  } else
  for (int it = it_begin; it < it_end; ++it) {
    const int k0 = it * 2048 + lane * 32;
    const int gi = k0 < k ? k0 / g : 0;  // OOB lanes read s = 0 anyway
    const bool ghi = gi >= 64;
    const int gsel = ghi ? gi - 64 : gi;

    if (BITS == 4) {
      const u32 wbyte = (u32)(it * 1024 + lane * 16);
      u32x4 w[COLS];
#pragma unroll
      for (int c = 0; c < COLS; ++c)
        w[c] = __builtin_amdgcn_raw_buffer_load_b128(w_rsrc[c], wbyte, 0, 2);
#pragma unroll
      for (int mm = 0; mm < M; ++mm) {
        if (mm >= m) break;
        const u32 xbyte = (u32)(mm * k * 2 + it * 4096 + lane * 64);
        u32x4 xq0 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte, 0, 0);
        u32x4 xq1 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 16, 0, 0);
        u32x4 xq2 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 32, 0, 0);
        u32x4 xq3 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 48, 0, 0);
        float xs = 0.f;  // Σx over this lane chunk: shared by all columns
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
          xs = dot2_bf16(ONE2_BF16, xq[0], xs);
          xs = dot2_bf16(ONE2_BF16, xq[1], xs);
          xs = dot2_bf16(ONE2_BF16, xq[2], xs);
          xs = dot2_bf16(ONE2_BF16, xq[3], xs);
          if (RMS) {
            ss_acc[mm] = dot2_bf16(xq[0], xq[0], ss_acc[mm]);
            ss_acc[mm] = dot2_bf16(xq[1], xq[1], ss_acc[mm]);
            ss_acc[mm] = dot2_bf16(xq[2], xq[2], ss_acc[mm]);
            ss_acc[mm] = dot2_bf16(xq[3], xq[3], ss_acc[mm]);
          }
        }
#pragma unroll
        for (int c = 0; c < COLS; ++c) {
          // NB: __shfl evaluates its ARGUMENT in the source lane, so the
          // lo/hi array pick must use THIS lane's ghi after shuffling both.
          const float s_l = __shfl(s_lo[c], gsel, 64);
          const float s_h = G > 64 ? __shfl(s_hi[c], gsel, 64) : 0.f;
          const float sz_l = __shfl(sz_lo[c], gsel, 64);
          const float sz_h = G > 64 ? __shfl(sz_hi[c], gsel, 64) : 0.f;
          // inactive lanes (k0 >= k) must contribute 0: their x loads can
          // land in the NEXT row of x when m > 1 (whole-tensor bounds).
          const float live = k0 < k ? 1.f : 0.f;
          const float s = (ghi ? s_h : s_l) * live;
          const float sz = (ghi ? sz_h : sz_l) * live;
          float cx = 0.f;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
            const DequantPairs4 dq = dequant4_pairs(w[c][j]);
            cx = dot2_bf16(dq.p01, xq[0], cx);
            cx = dot2_bf16(dq.p23, xq[1], cx);
            cx = dot2_bf16(dq.p45, xq[2], cx);
            cx = dot2_bf16(dq.p67, xq[3], cx);
          }
          acc[c][mm] = fmaf(s, cx, acc[c][mm]);
          acc[c][mm] = fmaf(-sz, xs, acc[c][mm]);
        }
      }
    } else {  // BITS == 8
      const u32 wbyte = (u32)(it * 2048 + lane * 32);
      u32x4 w0[COLS], w1[COLS];
#pragma unroll
      for (int c = 0; c < COLS; ++c) {
        w0[c] = __builtin_amdgcn_raw_buffer_load_b128(w_rsrc[c], wbyte, 0, 2);
        w1[c] = __builtin_amdgcn_raw_buffer_load_b128(w_rsrc[c], wbyte + 16, 0, 2);
      }
#pragma unroll
      for (int mm = 0; mm < M; ++mm) {
        if (mm >= m) break;
        const u32 xbyte = (u32)(mm * k * 2 + it * 4096 + lane * 64);
        float xv[32];
#pragma unroll
        for (int h = 0; h < 4; ++h) {
          u32x4 xq = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 16 * h, 0, 0);
#pragma unroll
          for (int j = 0; j < 4; ++j)
            bf16x2_to_f32(xq[j], xv[h * 8 + 2 * j], xv[h * 8 + 2 * j + 1]);
        }
        float xs = 0.f;
#pragma unroll
        for (int i = 0; i < 32; ++i) xs += xv[i];
        if (RMS) {
#pragma unroll
          for (int i = 0; i < 32; ++i) ss_acc[mm] = fmaf(xv[i], xv[i], ss_acc[mm]);
        }
#pragma unroll
        for (int c = 0; c < COLS; ++c) {
          const float s_l = __shfl(s_lo[c], gsel, 64);
          const float s_h = G > 64 ? __shfl(s_hi[c], gsel, 64) : 0.f;
          const float sz_l = __shfl(sz_lo[c], gsel, 64);
          const float sz_h = G > 64 ? __shfl(sz_hi[c], gsel, 64) : 0.f;
          const float live = k0 < k ? 1.f : 0.f;
          const float s = (ghi ? s_h : s_l) * live;
          const float sz = (ghi ? sz_h : sz_l) * live;
          float cx = 0.f;
#pragma unroll
          for (int half = 0; half < 2; ++half) {
            const u32x4 wv = half ? w1[c] : w0[c];
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              const int e = half * 16 + j * 4;
              cx = fmaf((float)(wv[j] & 0xFFu), xv[e], cx);
              cx = fmaf((float)((wv[j] >> 8) & 0xFFu), xv[e + 1], cx);
              cx = fmaf((float)((wv[j] >> 16) & 0xFFu), xv[e + 2], cx);
              cx = fmaf((float)(wv[j] >> 24), xv[e + 3], cx);
            }
          }
          acc[c][mm] = fmaf(s, cx, acc[c][mm]);
          acc[c][mm] = fmaf(-sz, xs, acc[c][mm]);
        }
      }
    }
  }

  // ---- wave-reduce and store -----------------------------------------
  __shared__ float part_sh[WAVES_PER_WG][COLS][M];
  __shared__ float ss_sh[WAVES_PER_WG][M];
  if (KSPLIT) {
#pragma unroll
    for (int mm = 0; mm < M; ++mm) {
      if (mm >= m) break;
      if (RMS) {
        const float ssw = wave_reduce_sum(ss_acc[mm]);
        if (lane == 0) ss_sh[wave][mm] = ssw;
      }
#pragma unroll
      for (int c = 0; c < COLS; ++c) {
        const float v = wave_reduce_sum(acc[c][mm]);
        if (lane == 0) part_sh[wave][c][mm] = v;
      }
    }
    __syncthreads();
    if (wave != 0) return;
#pragma unroll
    for (int mm = 0; mm < M; ++mm) {
      if (mm >= m) break;
      float inv = 1.f;
      if (RMS) {
        float ss = 0.f;
#pragma unroll
        for (int w = 0; w < WAVES_PER_WG; ++w) ss += ss_sh[w][mm];
        inv = rsqrtf(ss / (float)k + eps);
      }
      if (lane == 0) {
        unsigned short* yrow = (unsigned short*)y;
#pragma unroll
        for (int c = 0; c < COLS; ++c) {
          if (col[c] >= n) continue;
          float v = 0.f;
#pragma unroll
          for (int w = 0; w < WAVES_PER_WG; ++w) v += part_sh[w][c][mm];
          v *= inv;
          if (bias != nullptr)
            v += bf16_bits_to_f32(((const unsigned short*)bias)[col[c]]);
          if (resid != nullptr)
            v += bf16_bits_to_f32(
                ((const unsigned short*)resid)[(u64)mm * n + col[c]]);
          yrow[(u64)mm * n + col[c]] = (unsigned short)f32_to_bf16_bits_rne(v);
        }
      }
    }
    return;
  }
#pragma unroll
  for (int mm = 0; mm < M; ++mm) {
    if (mm >= m) break;
    float inv = 1.f;
    if (RMS) {
      // fused RMSNorm: the wave's lanes saw the whole x row, so the
      // row's Σx² wave-reduces locally; the norm WEIGHT is pre-folded
      // into the quantized weights (lfmq.quant.fold_rms_norms), leaving
      // a scalar rsqrt on the accumulator: Σ wq·(x·inv) = inv·Σ wq·x.
      const float ss = wave_reduce_sum(ss_acc[mm]);
      inv = rsqrtf(ss / (float)k + eps);
    }
    float v[COLS];
#pragma unroll
    for (int c = 0; c < COLS; ++c) {
      v[c] = wave_reduce_sum(acc[c][mm]) * inv;
      if (bias != nullptr && col[c] < n)
        v[c] += bf16_bits_to_f32(((const unsigned short*)bias)[col[c]]);
      if (!GATEUP && resid != nullptr && col[c] < n)
        v[c] += bf16_bits_to_f32(
            ((const unsigned short*)resid)[(u64)mm * n + col[c]]);
    }
    if (lane == 0) {
      unsigned short* yrow = (unsigned short*)y;
      if (GATEUP) {
        const int inter = n / 2;
#pragma unroll
        for (int p = 0; p < COLS / 2; ++p) {  // pair p: gate v[p], up v[p + COLS/2]
          const int oc = cbase + p;
          if (oc < inter) {
            const float gt = v[p];
            const float sil = gt / (1.f + __expf(-gt));
            yrow[(u64)mm * inter + oc] =
                (unsigned short)f32_to_bf16_bits_rne(sil * v[p + COLS / 2]);
          }
        }
      } else {
#pragma unroll
        for (int c = 0; c < COLS; ++c)
          if (col[c] < n)
            yrow[(u64)mm * n + col[c]] =
                (unsigned short)f32_to_bf16_bits_rne(v[c]);
      }
    }
  }
}

template <int BITS, int COLS, bool GATEUP, int M, bool RMS>
void launch_one(dim3 grid, dim3 block, hipStream_t stream, const void* qweight,
                const void* scales, const void* qzeros, const void* x, void* y,
                const void* bias, const void* resid, int m, int n, int k,
                int g, float eps) {
  const int kiters = (k + 2047) / 2048;
#define LFMQ_KI(KIV)                                                          \
  hipLaunchKernelGGL((dq_gemv_kernel<M, BITS, COLS, GATEUP, KIV, RMS>), grid, \
                     block, 0, stream, (const u32*)qweight,                   \
                     (const u32*)scales, (const u32*)qzeros, (const u32*)x,   \
                     (u32*)y, (const u32*)bias, (const u32*)resid, m, n, k,   \
                     g, eps)
  if (BITS == 4 && (COLS == 4 || COLS == 2 || COLS == 1)) {
    if (kiters == 1) LFMQ_KI(1);
    else if (kiters == 2) LFMQ_KI(2);
    else if (kiters == 7) LFMQ_KI(7);
    else LFMQ_KI(0);
  } else {
    LFMQ_KI(0);
  }
#undef LFMQ_KI
}

template <int BITS, int COLS, bool GATEUP, int M, bool RMS>
void launch_ksplit(dim3 grid, dim3 block, hipStream_t stream,
                   const void* qweight, const void* scales, const void* qzeros,
                   const void* x, void* y, const void* bias, const void* resid,
                   int m, int n, int k, int g, float eps) {
  hipLaunchKernelGGL((dq_gemv_kernel<M, BITS, COLS, GATEUP, 0, RMS, true>),
                     grid, block, 0, stream, (const u32*)qweight,
                     (const u32*)scales, (const u32*)qzeros, (const u32*)x,
                     (u32*)y, (const u32*)bias, (const u32*)resid, m, n, k, g,
                     eps);
}

template <int BITS, bool GATEUP, bool RMS>
hipError_t launch_bits(const void* qweight, const void* scales,
                       const void* qzeros, const void* x, void* y,
                       const void* bias, const void* resid, int m, int n,
                       int k, int g, float eps, int cols_hint,
                       hipStream_t stream) {
  dim3 block(WAVES_PER_WG * LFMQ_WAVE);
  if (GATEUP || m <= 4) {
    // columns per wave: fewer columns -> more workgroups; small-n GEMVs
    // are occupancy/latency-bound (probe.hip: the inner loop itself
    // sustains >6 TB/s at large grids).  cols_hint 0 = auto.
    int cols = cols_hint;
    const int kiters = (k + 2047) / 2048;
    // measured: wg-internal split-k does NOT beat the multi-column wave
    // layout on the decode shapes (LDS combine + fewer weight streams
    // offset the shorter latency chain) — kept behind hint 5 for A/B.
    const bool want_ksplit = !GATEUP && BITS == 4 && kiters >= 2 &&
                             cols_hint == 5;
    if (want_ksplit) {
      // wg-internal split-k: the 4 waves share 4 columns
      dim3 gridk(ceil_div_u32(n, 4));
      if (m <= 1)
        launch_ksplit<BITS, 4, false, 1, RMS>(gridk, block, stream, qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, eps);
      else if (m <= 2)
        launch_ksplit<BITS, 4, false, 2, RMS>(gridk, block, stream, qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, eps);
      else
        launch_ksplit<BITS, 4, false, 4, RMS>(gridk, block, stream, qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, eps);
      return hipGetLastError();
    }
    if (cols == 0 || cols == 5) cols = (GATEUP || n >= 16384) ? 4 : 2;
    const int per_wg = WAVES_PER_WG * (GATEUP ? cols / 2 : cols);
    dim3 grid(ceil_div_u32(GATEUP ? n / 2 : n, per_wg));
#define LFMQ_M(CC)                                                            \
    do {                                                                      \
      if (m <= 1)                                                             \
        launch_one<BITS, CC, GATEUP, 1, RMS>(grid, block, stream, qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, eps); \
      else if (m <= 2)                                                        \
        launch_one<BITS, CC, GATEUP, 2, RMS>(grid, block, stream, qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, eps); \
      else                                                                    \
        launch_one<BITS, CC, GATEUP, 4, RMS>(grid, block, stream, qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, eps); \
    } while (0)
    if (cols == 4) LFMQ_M(4);
    else if (cols == 1) LFMQ_M(GATEUP ? 2 : 1);
    else LFMQ_M(2);
#undef LFMQ_M
  } else {
    // larger decode batches: 2 columns per wave (register budget)
    dim3 grid(ceil_div_u32(n, WAVES_PER_WG * 2));
    if (m <= 8)
      launch_one<BITS, 2, false, 8, false>(grid, block, stream, qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, eps);
    else
      launch_one<BITS, 2, false, 16, false>(grid, block, stream, qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, eps);
  }
  return hipGetLastError();
}

}  // namespace

// rms_eps >= 0 enables the fused-RMSNorm input path (norm weight must be
// pre-folded into the quantized weights); resid != null fuses a residual
// add into the epilogue (invalid under TP row-parallel world > 1 — the
// caller adds after the all-reduce instead).
hipError_t lfmq_dq_gemv_mfma(const void* qweight, const void* scales,
                             const void* qzeros, const void* x, void* y,
                             const void* bias, int m, int n, int k, int g,
                             hipStream_t stream);

hipError_t lfmq_dq_gemv(const void* qweight, const void* scales,
                        const void* qzeros, const void* x, void* y,
                        const void* bias, const void* resid, int m, int n,
                        int k, int g, int bits, float rms_eps, int cols_hint,
                        hipStream_t stream) {
  if (m > 16 || k % 8 != 0 || g % 32 != 0 || g > k) return hipErrorInvalidValue;
  // MFMA-tiled batched variant (csrc/dq_gemv_mfma.hip) with the 2-deep
  // wave-local pipeline: m=16 measured 18.6 us vs 47 on the lane-dot
  // path (m=8: 16.2 vs 18.3) — auto-routed for m > 4.
  if (bits == 4 && m > 4 && rms_eps < 0.f && resid == nullptr && g <= 128 &&
      k % g == 0 && k / g <= 128 && (cols_hint == 0 || cols_hint == 6))
    return lfmq_dq_gemv_mfma(qweight, scales, qzeros, x, y, bias, m, n, k, g,
                             stream);
  const bool rms = rms_eps >= 0.f;
  if (rms && m > 4) return hipErrorInvalidValue;
  if (bits == 4) {
    if (rms)
      return launch_bits<4, false, true>(qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, rms_eps, cols_hint, stream);
    return launch_bits<4, false, false>(qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, 0.f, cols_hint, stream);
  }
  if (bits == 8) {
    if (rms)
      return launch_bits<8, false, true>(qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, rms_eps, cols_hint, stream);
    return launch_bits<8, false, false>(qweight, scales, qzeros, x, y, bias, resid, m, n, k, g, 0.f, cols_hint, stream);
  }
  return hipErrorInvalidValue;
}

// fused gate_up GEMV + silu·mul epilogue: W is the fused gate_up
// projection [n = 2*inter, k]; output is [m, inter] = silu(gate)*up.
hipError_t lfmq_dq_gemv_gateup(const void* qweight, const void* scales,
                               const void* qzeros, const void* x, void* y,
                               int m, int n, int k, int g, int bits,
                               float rms_eps, int cols_hint,
                               hipStream_t stream) {
  if (m > 4 || k % 8 != 0 || g % 32 != 0 || g > k || n % 2)
    return hipErrorInvalidValue;
  const bool rms = rms_eps >= 0.f;
  if (bits == 4) {
    if (rms)
      return launch_bits<4, true, true>(qweight, scales, qzeros, x, y, nullptr, nullptr, m, n, k, g, rms_eps, cols_hint, stream);
    return launch_bits<4, true, false>(qweight, scales, qzeros, x, y, nullptr, nullptr, m, n, k, g, 0.f, cols_hint, stream);
  }
  if (bits == 8) {
    if (rms)
      return launch_bits<8, true, true>(qweight, scales, qzeros, x, y, nullptr, nullptr, m, n, k, g, rms_eps, cols_hint, stream);
    return launch_bits<8, true, false>(qweight, scales, qzeros, x, y, nullptr, nullptr, m, n, k, g, 0.f, cols_hint, stream);
  }
  return hipErrorInvalidValue;
}
