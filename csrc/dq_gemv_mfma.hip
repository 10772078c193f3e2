// K3a for batched decode (5 <= m <= 16): MFMA-tiled dequant GEMV.
//
// The lane-dot GEMV (dq_gemv.hip) pays per-row activation loads and
// per-row VALU — at m = 16 it measured 0.4 TB/s.  This variant maps the
// work onto mfma_f32_16x16x32_bf16: each WAVE owns one 16-col output
// tile with all 16 (padded) rows; the 4 waves of a workgroup split the
// k-range of the SAME tile and combine through LDS once at the end, so
// the k-loop needs NO workgroup barriers (all LDS traffic inside the
// loop is wave-local, ordered by the wave's own lgkmcnt).
//   - B fragment: ONE packed dword per lane per MFMA — dequant4_pairs
//     turns it into the 8-bf16 fragment (values 128+c) directly.
//   - per-group scale factoring keeps dequant off the critical path:
//     over one group  Σ s·(c−z)·x = s·(Σ(128+c)·x − (128+z)·Σx),
//     with Σ(128+c)·x accumulated by MFMA per group and Σx per
//     (row, group) precomputed cooperatively into LDS (xgs).
//   - A fragment: the wave stages its x k-tile [16][g] into its own LDS
//     quarter (rows >= m land OOB in the buffer and read 0), XOR-swizzled
//     for conflict-free ds_read_b128.
// Grid: ceil(n/16) workgroups — fills the chip for every decode shape.
// Requirements: bits == 4, g % 32 == 0, g <= 128 (LDS tile), k % g == 0.
#include "common.h"

namespace {

constexpr int TPB = 256;
constexpr int NW = 4;
typedef s16x8 frag_ab;

__device__ __forceinline__ u32 aswz(u32 byte_off, u32 row) {
  return byte_off ^ ((row & 7u) << 4);
}

__global__ __launch_bounds__(TPB, 1) void dq_gemv_mfma_kernel(
    const u32* __restrict__ qweight, const u32* __restrict__ scales_u16,
    const u32* __restrict__ qzeros, const u32* __restrict__ x,
    u32* __restrict__ y, const u32* __restrict__ bias, int m, int n, int k,
    int g) {
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  const int col0 = blockIdx.x * 16;
  if (col0 >= n) return;

  const int G = k / g;
  const int zw = (G + 7) / 8;
  const int gdw = g / 8;  // packed dwords per group per row

  // LDS: per-wave x tiles [16][g<=128] bf16 (4 KiB each) + per-wave
  // W tiles [16 cols][20-dword padded rows] + xgs [16][G<=128] fp32 +
  // cross-wave accumulator tiles.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // banks: [2][NW] x-tiles (4 KiB each), then [2][NW] padded W tiles,
  // then xgs [16][G], then the combine tiles
  char* xtile = smem + wave * 16 * 128 * 2;            // bank 0 slot
  u32* wtile = (u32*)(smem + 2 * 4 * 16 * 128 * 2) + wave * 16 * 20;
  float* xgs = (float*)(smem + 2 * 4 * 16 * 128 * 2 + 2 * 4 * 16 * 20 * 4);
  float* comb = xgs + 16 * 128;                        // [NW][16][16]

  auto x_rsrc = make_rsrc(x, (u32)(m * k * 2));

  // ---- cooperative xgs: Σx per (row, group) --------------------------
  for (int t = threadIdx.x; t < 16 * G; t += TPB) {
    const int row = t / G, gg = t % G;
    float s4 = 0.f;
    const u32 base = (u32)((row * k + gg * g) * 2);
    for (int j = 0; j < g / 8; ++j) {
      u32x4 v = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, base + 16 * j, 0, 0);
      s4 = dot2_bf16(0x3F803F80u, v[0], s4);
      s4 = dot2_bf16(0x3F803F80u, v[1], s4);
      s4 = dot2_bf16(0x3F803F80u, v[2], s4);
      s4 = dot2_bf16(0x3F803F80u, v[3], s4);
    }
    xgs[row * G + gg] = s4;
  }
  __syncthreads();

  // ---- per-lane column metadata --------------------------------------
  const int lc = lane & 15;                 // this lane's column-in-tile
  const int cc = min(col0 + lc, n - 1);
  const unsigned short* srow = (const unsigned short*)scales_u16 + (u64)cc * G;
  const u32* zrow = qzeros + (u64)cc * zw;
  // W staging source: lane reads a 16-B run of column (lane/4)'s packed
  // row — 64-B contiguous per 4 lanes (the per-lane-column read pattern
  // is 4-B scatter and measured slower than the old lane-dot path).
  const int wst_col = min(col0 + lane / 4, n - 1);
  const u32* wst_row = qweight + (u64)wst_col * (k / 8);

  // this wave's group range
  const int per_wave = (G + NW - 1) / NW;
  const int g_begin = wave * per_wave;
  const int g_end = min(G, g_begin + per_wave);

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};

  if (gdw == 16) {
    // g == 128 fast path: 2-deep wave-local pipeline — group gg+1's
    // global loads ISSUE before gg's MFMAs and their LDS writes land
    // after (T14 issue-early/write-late; no barriers: all LDS traffic
    // is wave-local, double-buffered in the wave's two tile slots).
    char* xt[2] = {xtile, xtile + NW * 16 * 128 * 2};   // bank 1 slot
    u32* wt[2] = {wtile, wtile + NW * 16 * 20};
    u32x4 xr[4], wr;
    auto issue = [&](int gg) {
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const int task = lane + 64 * t;
        const int row = task / 16, slot = task % 16;
        xr[t] = __builtin_amdgcn_raw_buffer_load_b128(
            x_rsrc, (u32)((row * k + gg * g) * 2 + slot * 16), 0, 0);
      }
      wr = *(const u32x4*)(wst_row + gg * 16 + (lane % 4) * 4);
    };
    auto write = [&](int buf) {
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const int task = lane + 64 * t;
        const int row = task / 16, slot = task % 16;
        *(u32x4*)(xt[buf] + aswz((u32)(row * 256 + slot * 16), row)) = xr[t];
      }
      *(u32x4*)(wt[buf] + (lane / 4) * 20 + (lane % 4) * 4) = wr;
    };
    issue(g_begin);
    write(0);
    for (int gg = g_begin; gg < g_end; ++gg) {
      const int buf = (gg - g_begin) & 1;
      if (gg + 1 < g_end) issue(gg + 1);   // loads in flight across MFMAs
      f32x4 accg = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        const u32 wd = wt[buf][lc * 20 + ks * 4 + (lane >> 4)];
        const DequantPairs4 dq = dequant4_pairs(wd);
        frag_ab b = __builtin_bit_cast(
            frag_ab, u32x4{dq.p01, dq.p23, dq.p45, dq.p67});
        const int krow = lane & 15;
        const int koff = ks * 32 + (lane >> 4) * 8;
        frag_ab a = *(frag_ab*)(xt[buf] + aswz((u32)(krow * 256 + koff * 2), krow));
        accg = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, accg, 0, 0, 0);
      }
      if (gg + 1 < g_end) write(buf ^ 1);  // write-late into the other slot
      const float s = bf16_bits_to_f32(srow[gg]);
      const float zb = s * (float)(unpack4(zrow[gg / 8], gg % 8) + 128u);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = (lane >> 4) * 4 + r;
        acc[r] = fmaf(s, accg[r], acc[r]);
        acc[r] = fmaf(-zb, xgs[row * G + gg], acc[r]);
      }
    }
  } else
  for (int gg = g_begin; gg < g_end; ++gg) {
    // stage this wave's x k-tile [16][g]: 16 rows x (g/8) 16-B slots
    {
      const int spr = g / 8;                  // 16-B slots per row
      for (int task = lane; task < 16 * spr; task += LFMQ_WAVE) {
        const int row = task / spr;
        const int slot = task % spr;
        const u32 gbyte = (u32)((row * k + gg * g) * 2 + slot * 16);
        u32x4 v = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, gbyte, 0, 0);
        *(u32x4*)(xtile + aswz((u32)(row * g * 2 + slot * 16), row)) = v;
      }
    }
    // stage the W tile for this group: 16 cols x gdw dwords; lane l
    // loads col (l/4), dwords (l%4)*4..+4 (dwordx4; only for gdw == 16,
    // i.e. g == 128 — smaller g stages with b32 loads)
    if (gdw == 16) {
      const u32x4 wv = *(const u32x4*)(wst_row + gg * 16 + (lane % 4) * 4);
      *(u32x4*)(wtile + (lane / 4) * 20 + (lane % 4) * 4) = wv;
    } else {
      for (int t = lane; t < 16 * gdw; t += LFMQ_WAVE)
        wtile[(t / gdw) * 20 + (t % gdw)] = (qweight +
            (u64)min(col0 + t / gdw, n - 1) * (k / 8))[gg * gdw + t % gdw];
    }
    // wave-local ordering only: the compiler waits the wave's own
    // vmcnt/lgkmcnt before the dependent ds_reads — no barrier.
    f32x4 accg = {0.f, 0.f, 0.f, 0.f};
#pragma unroll 4
    for (int ks = 0; ks < g / 32; ++ks) {
      // B: one dword per lane from the staged tile -> 8-bf16 fragment
      const u32 wd = wtile[lc * 20 + ks * 4 + (lane >> 4)];
      const DequantPairs4 dq = dequant4_pairs(wd);
      frag_ab b = __builtin_bit_cast(
          frag_ab, u32x4{dq.p01, dq.p23, dq.p45, dq.p67});
      // A: x rows, 16 B contiguous per lane
      const int krow = lane & 15;
      const int koff = ks * 32 + (lane >> 4) * 8;
      frag_ab a = *(frag_ab*)(xtile + aswz((u32)(krow * g * 2 + koff * 2), krow));
      accg = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, accg, 0, 0, 0);
    }
    // apply this group's scale and zero term
    const float s = bf16_bits_to_f32(srow[gg]);
    const float zb = s * (float)(unpack4(zrow[gg / 8], gg % 8) + 128u);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
      acc[r] = fmaf(s, accg[r], acc[r]);
      acc[r] = fmaf(-zb, xgs[row * G + gg], acc[r]);
    }
  }

  // ---- cross-wave combine (the only workgroup sync) -------------------
#pragma unroll
  for (int r = 0; r < 4; ++r)
    comb[(wave * 16 + (lane >> 4) * 4 + r) * 16 + lc] = acc[r];
  __syncthreads();
  if (wave == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
      if (row < m && col0 + lc < n) {
        float v = 0.f;
#pragma unroll
        for (int w = 0; w < NW; ++w) v += comb[(w * 16 + row) * 16 + lc];
        if (bias != nullptr)
          v += bf16_bits_to_f32(((const unsigned short*)bias)[col0 + lc]);
        ((unsigned short*)y)[(u64)row * n + col0 + lc] =
            (unsigned short)f32_to_bf16_bits_rne(v);
      }
    }
  }
}

}  // namespace

hipError_t lfmq_dq_gemv_mfma(const void* qweight, const void* scales,
                             const void* qzeros, const void* x, void* y,
                             const void* bias, int m, int n, int k, int g,
                             hipStream_t stream) {
  if (m > 16 || g % 32 != 0 || g > 128 || k % g != 0 || k / g > 128)
    return hipErrorInvalidValue;
  dim3 grid(ceil_div_u32(n, 16)), block(TPB);
  const size_t lds = 2 * (4 * 16 * 128 * 2 + 4 * 16 * 20 * 4) +
                     16 * 128 * 4 + NW * 16 * 16 * 4;
  hipLaunchKernelGGL(dq_gemv_mfma_kernel, grid, block, lds, stream,
                     (const u32*)qweight, (const u32*)scales,
                     (const u32*)qzeros, (const u32*)x, (u32*)y,
                     (const u32*)bias, m, n, k, g);
  return hipGetLastError();
}
