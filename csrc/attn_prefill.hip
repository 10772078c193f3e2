// Flash-style prefill attention on MFMA: o = softmax(q·Kᵀ·scale +
// causal)·V against the KV cache, s >= 1 queries at positions
// [pos0, pos0+s).  Replaces the rocBLAS matmul chain of
// lfmq/ops/functional.py:attn_prefill as the default GPU prefill path
// (SURVEY.md §2.3 K8 prefill note): measured 3.3-6.3x faster at
// s 256-2048, rel_err ~4e-3 vs the fp32 oracle
// (tools/prefill_flash_bench.py on 1xMI355X).
//
// Shape contract (the fused decode layouts, no transposes on the host):
//   q   [b, s, hq*128] bf16, already roped (qkv_rope_append output view)
//   k/v [b, hkv, tmax, 128] bf16 caches holding positions [0, pos0+s)
//   out [b, s, hq*128] bf16
//
// One workgroup = (64-query tile, head, batch); 4 waves × 16 query rows.
// Per 64-key tile: K staged to LDS (row-padded 16 B against bank
// conflicts), V staged TRANSPOSED (Vt[d][key], so the P·V B-fragment is a
// contiguous ds_read_b128), scores via mfma_f32_16x16x32_bf16 (A = q rows
// held in registers for the whole key loop), online softmax with the
// per-row running (m, l) entirely in registers — the C/D fragment rows
// (lane>>4)*4+r line up between the score and output accumulators, so the
// flash rescale is a per-lane multiply.  P transposes through a per-wave
// LDS slab into A-fragment layout (wave-local: LDS ops retire in order
// per wave; one lgkmcnt(0) wait, no barrier).  Causal + tail masking is
// done on the score fragment; q rows past s read zeros through the
// hardware-bounds buffer descriptor (T8) and are simply never written.
#include "common.h"

namespace {

constexpr int TPB = 256;         // 4 waves
constexpr int BM = 64;           // query rows per workgroup
constexpr int BN = 64;           // key tile
constexpr int HD = 128;          // head_dim (only supported size)
constexpr int KROWB = HD * 2 + 16;   // K LDS row bytes (16 B pad)
constexpr int VROWB = BN * 2 + 16;   // Vt LDS row bytes
constexpr int PROWB = BN * 2 + 16;   // P LDS row bytes

typedef s16x8 frag_ab;

__global__ __launch_bounds__(TPB) void attn_prefill_kernel(
    const u32* __restrict__ q, const u32* __restrict__ kc,
    const u32* __restrict__ vc, u32* __restrict__ o, int b, int s, int hq,
    int hkv, int tmax, int pos0, int T, float scale) {
  const int q0 = blockIdx.x * BM;
  const int h = blockIdx.y;
  const int bb = blockIdx.z;
  const int hk = h / (hq / hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  __shared__ char ldsK[BN * KROWB];            // K[key][d]
  __shared__ char ldsVt[HD * VROWB];           // V transposed: Vt[d][key]
  __shared__ char ldsP[4][16 * PROWB];         // per-wave P[q][key]

  // hardware-bounds descriptors: rows past s (q) / T (cache) read 0
  const __amdgpu_buffer_rsrc_t q_rsrc = make_rsrc_uniform(
      (const char*)q + (u64)bb * s * hq * (HD * 2), (u32)s * hq * (HD * 2));
  const __amdgpu_buffer_rsrc_t k_rsrc = make_rsrc_uniform(
      (const char*)kc + ((u64)bb * hkv + hk) * tmax * (HD * 2),
      (u32)T * (HD * 2));
  const __amdgpu_buffer_rsrc_t v_rsrc = make_rsrc_uniform(
      (const char*)vc + ((u64)bb * hkv + hk) * tmax * (HD * 2),
      (u32)T * (HD * 2));

  // ---- q A-fragments: 16 rows x 128 d, resident for the whole key loop
  frag_ab qf[HD / 32];
  {
    const u32 qrow = q0 + wave * 16 + l15;
    const u32 qbyte0 = qrow * (u32)(hq * HD * 2) + (u32)h * (HD * 2);
#pragma unroll
    for (int ks = 0; ks < HD / 32; ++ks)
      qf[ks] = __builtin_bit_cast(
          frag_ab, __builtin_amdgcn_raw_buffer_load_b128(
                       q_rsrc, qbyte0 + (u32)(ks * 32 + l4 * 8) * 2, 0, 0));
  }

  float mrun[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  float lrun[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4 oacc[HD / 16] = {};

  // last query of this tile sits at pos0+q0+BM-1 => keys < t_hi
  const int t_hi = min(T, pos0 + q0 + BM);
  for (int t0 = 0; t0 < t_hi; t0 += BN) {
    __syncthreads();  // previous tile's LDS readers are done
    // ---- stage K tile: 64 keys x 256 B; 16-B chunk per thread x 4
#pragma unroll
    for (int it = 0; it < (BN * HD * 2) / 16 / TPB; ++it) {
      const int c = it * TPB + threadIdx.x;       // 16-B chunk index
      const int key = c >> 4;                      // 16 chunks per row
      const int off = (c & 15) * 16;
      const u32x4 kd = __builtin_amdgcn_raw_buffer_load_b128(
          k_rsrc, (u32)(t0 + key) * (HD * 2) + off, 0, 0);
      *(u32x4*)(ldsK + key * KROWB + off) = kd;
    }
    // ---- stage V tile transposed: read [key][d0..d0+8), scatter to Vt
#pragma unroll
    for (int it = 0; it < (BN * HD * 2) / 16 / TPB; ++it) {
      const int c = it * TPB + threadIdx.x;
      const int key = c >> 4;
      const int d0 = (c & 15) * 8;
      const u32x4 vd = __builtin_amdgcn_raw_buffer_load_b128(
          v_rsrc, (u32)(t0 + key) * (HD * 2) + d0 * 2, 0, 0);
      const unsigned short* ve = (const unsigned short*)&vd;
#pragma unroll
      for (int i = 0; i < 8; ++i)
        *(unsigned short*)(ldsVt + (d0 + i) * VROWB + key * 2) = ve[i];
    }
    __syncthreads();

    // ---- scores: S[16 q][64 key] = q·Kᵀ, 4 n-tiles x 4 k-steps -------
    f32x4 sc[BN / 16] = {};
#pragma unroll
    for (int ks = 0; ks < HD / 32; ++ks) {
      const int koff = (ks * 32 + l4 * 8) * 2;
#pragma unroll
      for (int nt = 0; nt < BN / 16; ++nt) {
        const frag_ab kf =
            *(const frag_ab*)(ldsK + (nt * 16 + l15) * KROWB + koff);
        sc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[ks], kf, sc[nt],
                                                         0, 0, 0);
      }
    }

    // ---- scale + causal/tail mask on the C fragment -------------------
#pragma unroll
    for (int nt = 0; nt < BN / 16; ++nt) {
      const int key = t0 + nt * 16 + l15;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qpos = pos0 + q0 + wave * 16 + l4 * 4 + r;
        const bool ok = (key <= qpos) && (key < T);
        sc[nt][r] = ok ? sc[nt][r] * scale : -INFINITY;
      }
    }

    // ---- online softmax: per-row (m, l) across 4 n-tiles + 16 lanes ---
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = fmaxf(fmaxf(sc[0][r], sc[1][r]), fmaxf(sc[2][r], sc[3][r]));
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 64));   // within the 16-group
      const float mnew = fmaxf(mrun[r], mx);       // finite: key t0=0 valid
      const float alpha = __expf(mrun[r] - mnew);  // first tile: exp(-inf)=0
      float ps = 0.f;
#pragma unroll
      for (int nt = 0; nt < BN / 16; ++nt) {
        const float p = __expf(sc[nt][r] - mnew);  // masked -> 0
        sc[nt][r] = p;
        ps += p;
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) ps += __shfl_xor(ps, off, 64);
      lrun[r] = lrun[r] * alpha + ps;
      mrun[r] = mnew;
#pragma unroll
      for (int on = 0; on < HD / 16; ++on) oacc[on][r] *= alpha;
    }

    // ---- P: C layout -> A layout through the per-wave LDS slab --------
    char* pw = ldsP[wave];
#pragma unroll
    for (int nt = 0; nt < BN / 16; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        *(unsigned short*)(pw + (l4 * 4 + r) * PROWB + (nt * 16 + l15) * 2) =
            (unsigned short)f32_to_bf16_bits_rne(sc[nt][r]);
    // wave-local transpose: DS ops retire in order per wave; wait them out
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- O += P·V: 2 k-steps over keys x 8 n-tiles over d -------------
#pragma unroll
    for (int ks2 = 0; ks2 < BN / 32; ++ks2) {
      const int koff = (ks2 * 32 + l4 * 8) * 2;
      const frag_ab pf = *(const frag_ab*)(pw + l15 * PROWB + koff);
#pragma unroll
      for (int on = 0; on < HD / 16; ++on) {
        const frag_ab vf =
            *(const frag_ab*)(ldsVt + (on * 16 + l15) * VROWB + koff);
        oacc[on] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, oacc[on],
                                                           0, 0, 0);
      }
    }
  }

  // ---- epilogue: o[qr][h*128 + d] = oacc / l  (rows past s dropped) ----
  unsigned short* ob =
      (unsigned short*)o + (u64)bb * s * hq * HD;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qr = q0 + wave * 16 + l4 * 4 + r;
    if (qr >= s) continue;
    const float invl = 1.f / lrun[r];
    unsigned short* orow = ob + (u64)qr * hq * HD + h * HD;
#pragma unroll
    for (int on = 0; on < HD / 16; ++on)
      orow[on * 16 + l15] =
          (unsigned short)f32_to_bf16_bits_rne(oacc[on][r] * invl);
  }
}

}  // namespace

// q [b,s,hq*d] roped; caches [b,hkv,tmax,d] already hold rows [0, pos0+s).
// d == 128 only (callers fall back to the matmul-chain path otherwise).
hipError_t lfmq_attn_prefill(const void* q, const void* kc, const void* vc,
                             void* o, int b, int s, int hq, int hkv, int tmax,
                             int pos0, int d, float scale,
                             hipStream_t stream) {
  if (d != HD) return hipErrorInvalidValue;
  if (hq % hkv) return hipErrorInvalidValue;
  const int T = pos0 + s;
  if (T > tmax) return hipErrorInvalidValue;
  dim3 grid((s + BM - 1) / BM, hq, b), block(TPB);
  hipLaunchKernelGGL(attn_prefill_kernel, grid, block, 0, stream,
                     (const u32*)q, (const u32*)kc, (const u32*)vc, (u32*)o,
                     b, s, hq, hkv, tmax, pos0, T, scale);
  return hipGetLastError();
}
