// K8 — GQA decode attention: o = softmax(q·Kᵀ/√d)·V against the KV cache.
// Memory-bound cache scan: flash-decoding split over T.
//
// Grid (b·hq, ceil(T/256)); each 256-thread workgroup owns a 256-token
// chunk: one cache row per THREAD for q·K (scattered 16-B loads that
// still consume whole lines across the d-loop), then a per-wave PV pass
// (one V row per wave-instruction, coalesced 256 B, 8 rows in flight)
// with p broadcast from LDS.  nchunks == 1 (the common short-T decode
// case) writes the normalized output directly — no scratch round-trip,
// no combine launch; otherwise partials {o fp32[d], m, l} go to scratch
// and a combine kernel merges them with the standard flash rescale.
// Guards are branch-free: cache-row indices clamp to the allocation and
// p = 0 kills t >= T contributions.
#include "common.h"

namespace {

constexpr int TPB = 256;
constexpr int NWAVE = TPB / LFMQ_WAVE;

template <int D>
__global__ __launch_bounds__(TPB) void attn_decode_partial(
    const u32* __restrict__ q, const u32* __restrict__ kc,
    const u32* __restrict__ vc, float* __restrict__ scratch,
    u32* __restrict__ o_direct, int b, int hq, int hkv, int tmax, int T,
    const i32* __restrict__ T_dev, float scale, int nchunks) {
  if (T_dev != nullptr) T = *T_dev;  // hipGraph mode: T lives on device
  const int bh = blockIdx.x;
  const int chunk = blockIdx.y;
  if (chunk * TPB >= T) return;  // fixed max-chunk grid under graphs
  const int bb = bh / hq;
  const int h = bh % hq;
  const int hk = h / (hq / hkv);
  const int dw = D / 2;

  const u32* qrow = q + (u64)bh * dw;
  const u32* krow0 = kc + ((u64)bb * hkv + hk) * tmax * dw;
  const u32* vrow0 = vc + ((u64)bb * hkv + hk) * tmax * dw;

  __shared__ float p_sh[TPB];
  __shared__ float red[NWAVE];
  __shared__ float o_sh[NWAVE][D];
  __shared__ float m_sh, l_sh;

  // ---- q·K: one cache row per thread --------------------------------
  const int t = chunk * TPB + threadIdx.x;
  const int t_clamped = t < tmax ? t : tmax - 1;
  float s = -INFINITY;
  {
    const u32* krow = krow0 + (u64)t_clamped * dw;
    float acc = 0.f;
#pragma unroll
    for (int i = 0; i < dw / 4; ++i) {
      const u32x4 kv4 = *(const u32x4*)(krow + 4 * i);
      const u32x4 qv4 = *(const u32x4*)(qrow + 4 * i);
      acc = dot2_bf16(kv4[0], qv4[0], acc);
      acc = dot2_bf16(kv4[1], qv4[1], acc);
      acc = dot2_bf16(kv4[2], qv4[2], acc);
      acc = dot2_bf16(kv4[3], qv4[3], acc);
    }
    if (t < T) s = acc * scale;
  }

  // ---- chunk softmax --------------------------------------------------
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  float wm = wave_reduce_max(s);
  if (lane == 0) red[wave] = wm;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = red[0];
#pragma unroll
    for (int i = 1; i < NWAVE; ++i) m = fmaxf(m, red[i]);
    m_sh = m;
  }
  __syncthreads();
  const float m = m_sh;
  const float p = (t < T) ? __expf(s - m) : 0.f;
  p_sh[threadIdx.x] = p;
  float wl = wave_reduce_sum(p);
  if (lane == 0) red[wave] = wl;
  __syncthreads();
  if (threadIdx.x == 0) {
    float l = 0.f;
#pragma unroll
    for (int i = 0; i < NWAVE; ++i) l += red[i];
    l_sh = l;
  }

  // ---- PV: each wave sums its 64 tokens; 8 V rows in flight ----------
  float o0 = 0.f, o1 = 0.f;  // lane's dword (2 elems); lanes >= dw idle
  const int tbase = chunk * TPB + wave * LFMQ_WAVE;
  const int nt = min(LFMQ_WAVE, T - tbase);  // tokens this wave covers
  if (lane < dw) {
#pragma unroll 8
    for (int tt = 0; tt < LFMQ_WAVE; ++tt) {
      if (tt >= nt) break;
      const int tv = tbase + tt;
      const int tvc = tv < tmax ? tv : tmax - 1;
      float v0, v1;
      bf16x2_to_f32(vrow0[(u64)tvc * dw + lane], v0, v1);
      const float pv = p_sh[wave * LFMQ_WAVE + tt];
      o0 = fmaf(pv, v0, o0);
      o1 = fmaf(pv, v1, o1);
    }
    o_sh[wave][2 * lane] = o0;
    o_sh[wave][2 * lane + 1] = o1;
  }
  __syncthreads();

  if (nchunks == 1) {
    // direct write: o = Σ_w o_w / l
    const float invl = 1.f / l_sh;
    if (threadIdx.x < dw) {
      const int j = threadIdx.x;
      float a0 = 0.f, a1 = 0.f;
#pragma unroll
      for (int w = 0; w < NWAVE; ++w) {
        a0 += o_sh[w][2 * j];
        a1 += o_sh[w][2 * j + 1];
      }
      o_direct[(u64)bh * dw + j] = pack_bf16x2(a0 * invl, a1 * invl);
    }
    return;
  }

  // ---- write partial: scratch[bh][chunk] = {o[D], m, l} ---------------
  float* out = scratch + ((u64)bh * nchunks + chunk) * (D + 2);
  for (int j = threadIdx.x; j < D; j += TPB) {
    float v = 0.f;
#pragma unroll
    for (int w = 0; w < NWAVE; ++w) v += o_sh[w][j];
    out[j] = v;
  }
  if (threadIdx.x == 0) {
    out[D] = m;
    out[D + 1] = l_sh;
  }
}

// Fused decode attention (s == 1): consumes the RAW qkv projection
// buffer — ropes q and k in a cooperative prologue (pos = T-1), appends
// the new k/v row to the cache (the designated q-head-0 workgroup of
// each kv head), computes q·k_new by block reduction (so no chunk ever
// depends on another workgroup's cache write), then runs the same
// chunked flash-decoding as attn_decode_partial.  Replaces the separate
// qkv_rope_append launch on the decode path.
constexpr int FCH = 64;  // fused-path chunk: short PV chains, big grid

template <int D>
__global__ __launch_bounds__(TPB) void attn_decode_fused(
    const u32* __restrict__ qkv, u32* __restrict__ kc, u32* __restrict__ vc,
    float* __restrict__ scratch, u32* __restrict__ o_direct,
    const float* __restrict__ inv_freq, int b, int hq, int hkv, int tmax,
    int T, const i32* __restrict__ T_dev, float scale, int nchunks,
    u32* __restrict__ tickets) {
  if (T_dev != nullptr) T = *T_dev;
  const int bh = blockIdx.x;
  const int chunk = blockIdx.y;
  if (chunk * FCH >= T) return;
  const int bb = bh / hq;
  const int h = bh % hq;
  const int rep = hq / hkv;
  const int hk = h / rep;
  const int dw = D / 2;
  const int pos = T - 1;

  u32* krow_cache = kc + ((u64)bb * hkv + hk) * tmax * dw;
  u32* vrow_cache = vc + ((u64)bb * hkv + hk) * tmax * dw;
  const u32* qkv_row = qkv + (u64)bb * (hq + 2 * hkv) * dw;

  __shared__ float p_sh[TPB];
  __shared__ float red[NWAVE];
  __shared__ float o_sh[NWAVE][D];
  __shared__ float m_sh, l_sh, slast_sh;
  __shared__ unsigned short q_sh[D], k_sh[D];
  __shared__ u32 v_sh[D / 2];

  // ---- prologue: rope q (head h) and k (head hk), stage v ------------
  {
    const unsigned short* qk16 = (const unsigned short*)qkv_row;
    float prod = 0.f;
    if (threadIdx.x < D) {
      const int e = threadIdx.x;
      const int i = e & (D / 2 - 1);      // pair index
      const int hi = e >= D / 2;
      const float ang = (float)pos * inv_freq[i];
      float sv, cv;
      __sincosf(ang, &sv, &cv);
      const float q1 = bf16_bits_to_f32(qk16[h * D + i]);
      const float q2 = bf16_bits_to_f32(qk16[h * D + i + D / 2]);
      const float k1 = bf16_bits_to_f32(qk16[(hq + hk) * D + i]);
      const float k2 = bf16_bits_to_f32(qk16[(hq + hk) * D + i + D / 2]);
      const float qr = hi ? (q1 * sv + q2 * cv) : (q1 * cv - q2 * sv);
      const float kr = hi ? (k1 * sv + k2 * cv) : (k1 * cv - k2 * sv);
      q_sh[e] = (unsigned short)f32_to_bf16_bits_rne(qr);
      k_sh[e] = (unsigned short)f32_to_bf16_bits_rne(kr);
      prod = qr * kr;                      // q·k_new partial (fp32)
    }
    if (threadIdx.x < dw)                  // v passthrough (dwords)
      v_sh[threadIdx.x] = qkv_row[((u64)hq + hkv + hk) * dw + threadIdx.x];
    // block-reduce q·k_new
    float wsum = wave_reduce_sum(prod);
    const int wave0 = threadIdx.x / LFMQ_WAVE;
    if (threadIdx.x % LFMQ_WAVE == 0) red[wave0] = wsum;
    __syncthreads();
    if (threadIdx.x == 0) {
      float t = 0.f;
#pragma unroll
      for (int i = 0; i < NWAVE; ++i) t += red[i];
      slast_sh = t;
    }
    // designated writer appends the new k/v row to the cache
    if (h % rep == 0 && chunk == pos / FCH) {
      if (threadIdx.x < dw) {
        const unsigned short* ks = k_sh;
        krow_cache[(u64)pos * dw + threadIdx.x] =
            (u32)ks[2 * threadIdx.x] | ((u32)ks[2 * threadIdx.x + 1] << 16);
        vrow_cache[(u64)pos * dw + threadIdx.x] = v_sh[threadIdx.x];
      }
    }
    __syncthreads();
  }

  // ---- q·K over this 64-token chunk (wave 0 only; q from LDS) --------
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  const int t = chunk * FCH + lane;
  if (wave == 0) {
    const int t_clamped = t < tmax ? t : tmax - 1;
    float s = -INFINITY;
    const u32* krow = (const u32*)krow_cache + (u64)t_clamped * dw;
    const u32* q32 = (const u32*)q_sh;
    float acc = 0.f;
#pragma unroll
    for (int i = 0; i < dw / 4; ++i) {
      const u32x4 kv4 = *(const u32x4*)(krow + 4 * i);
      acc = dot2_bf16(kv4[0], q32[4 * i + 0], acc);
      acc = dot2_bf16(kv4[1], q32[4 * i + 1], acc);
      acc = dot2_bf16(kv4[2], q32[4 * i + 2], acc);
      acc = dot2_bf16(kv4[3], q32[4 * i + 3], acc);
    }
    if (t == pos) acc = slast_sh;          // never trust the stale cache row
    if (t < T) s = acc * scale;
    // single-wave softmax over the 64 scores
    const float m = wave_reduce_max(s);
    const float p = (t < T) ? __expf(s - m) : 0.f;
    p_sh[lane] = p;
    const float l = wave_reduce_sum(p);
    if (lane == 0) {
      m_sh = m;
      l_sh = l;
    }
  }
  __syncthreads();
  const float m = m_sh;

  // ---- PV: 4 waves x 16 tokens each (short dependent chains) ----------
  float o0 = 0.f, o1 = 0.f;
  const int tbase = chunk * FCH + wave * (FCH / NWAVE);
  const int nt = min(FCH / NWAVE, T - tbase);
  if (lane < dw) {
#pragma unroll
    for (int tt = 0; tt < FCH / NWAVE; ++tt) {
      if (tt >= nt) break;
      const int tv = tbase + tt;
      const int tvc = tv < tmax ? tv : tmax - 1;
      const u32 vd = (tv == pos) ? v_sh[lane]
                                 : vrow_cache[(u64)tvc * dw + lane];
      float v0, v1;
      bf16x2_to_f32(vd, v0, v1);
      const float pv = p_sh[wave * (FCH / NWAVE) + tt];
      o0 = fmaf(pv, v0, o0);
      o1 = fmaf(pv, v1, o1);
    }
    o_sh[wave][2 * lane] = o0;
    o_sh[wave][2 * lane + 1] = o1;
  }
  __syncthreads();

  if (nchunks == 1) {
    const float invl = 1.f / l_sh;
    if (threadIdx.x < dw) {
      const int j = threadIdx.x;
      float a0 = 0.f, a1 = 0.f;
#pragma unroll
      for (int w = 0; w < NWAVE; ++w) {
        a0 += o_sh[w][2 * j];
        a1 += o_sh[w][2 * j + 1];
      }
      o_direct[(u64)bh * dw + j] = pack_bf16x2(a0 * invl, a1 * invl);
    }
    return;
  }
  // In-launch combine by the LAST-ARRIVING chunk of this bh (tickets !=
  // null) instead of a separate combine kernel: ~4.8 us/launch x 32
  // layers/step of pure boundary cost measured in round 1 (VERDICT item
  // 7).  Guideline-16 discipline: partials stored write-through (agent
  // relaxed sc1), asm vmcnt(0) drain before the arrival ticket (the
  // compiler-hazard note in MI355X_MICROARCH.md), agent acquire by the
  // winner before its plain reads.  The winner RESETS the ticket so
  // hipGraph replays need no host-side re-zeroing.
  float* out = scratch + ((u64)bh * nchunks + chunk) * (D + 2);
  for (int j = threadIdx.x; j < D; j += TPB) {
    float v = 0.f;
#pragma unroll
    for (int w = 0; w < NWAVE; ++w) v += o_sh[w][j];
    if (tickets != nullptr)
      __hip_atomic_store((__attribute__((address_space(1))) float*)&out[j], v,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    else
      out[j] = v;
  }
  if (threadIdx.x == 0) {
    if (tickets != nullptr) {
      __hip_atomic_store((__attribute__((address_space(1))) float*)&out[D], m,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_store(
          (__attribute__((address_space(1))) float*)&out[D + 1], l_sh,
          __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    } else {
      out[D] = m;
      out[D + 1] = l_sh;
    }
  }
  if (tickets == nullptr) return;

  // arrival ticket.  vmcnt is a PER-WAVE counter and the partial stores
  // above were issued by every wave of the workgroup: each wave must
  // drain its own stores, and the barrier must order those drains before
  // lane 0 takes the ticket — a lane-0-only waitcnt would let another
  // CU's winner acquire while waves 1..3 still had partials in flight.
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  __shared__ int last_sh;
  const int used = (T + FCH - 1) / FCH;
  if (threadIdx.x == 0) {
    const unsigned old = __hip_atomic_fetch_add(
        (__attribute__((address_space(1))) unsigned*)&tickets[bh], 1u,
        __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    last_sh = (int)old == used - 1;
  }
  __syncthreads();
  if (!last_sh) return;
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");

  // combine (same math as attn_decode_combine), then reset the ticket
  const float* base = scratch + (u64)bh * nchunks * (D + 2);
  __shared__ float M_sh, invL_sh;
  if (threadIdx.x == 0) {
    float M = -INFINITY;
    for (int c = 0; c < used; ++c) M = fmaxf(M, base[c * (D + 2) + D]);
    float L = 0.f;
    for (int c = 0; c < used; ++c)
      L += __expf(base[c * (D + 2) + D] - M) * base[c * (D + 2) + D + 1];
    M_sh = M;
    invL_sh = 1.f / L;
    __hip_atomic_store((__attribute__((address_space(1))) unsigned*)&tickets[bh],
                       0u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  }
  __syncthreads();
  const int j = threadIdx.x;
  if (2 * j < D) {
    float acc0 = 0.f, acc1 = 0.f;
    for (int c = 0; c < used; ++c) {
      const float w = __expf(base[c * (D + 2) + D] - M_sh);
      acc0 = fmaf(w, base[c * (D + 2) + 2 * j], acc0);
      acc1 = fmaf(w, base[c * (D + 2) + 2 * j + 1], acc1);
    }
    o_direct[(u64)bh * dw + j] = pack_bf16x2(acc0 * invL_sh, acc1 * invL_sh);
  }
}

template <int D>
__global__ __launch_bounds__(128) void attn_decode_combine(
    const float* __restrict__ scratch, u32* __restrict__ o, int nchunks,
    int T, const i32* __restrict__ T_dev, int csz) {
  if (T_dev != nullptr) T = *T_dev;
  const int bh = blockIdx.x;
  const int used = (T + csz - 1) / csz;
  const float* base = scratch + (u64)bh * nchunks * (D + 2);

  float M = -INFINITY;
  for (int c = 0; c < used; ++c) M = fmaxf(M, base[c * (D + 2) + D]);
  float L = 0.f;
  for (int c = 0; c < used; ++c)
    L += __expf(base[c * (D + 2) + D] - M) * base[c * (D + 2) + D + 1];
  const float invL = 1.f / L;

  const int j = threadIdx.x;
  if (2 * j < D) {
    float acc0 = 0.f, acc1 = 0.f;
    for (int c = 0; c < used; ++c) {
      const float w = __expf(base[c * (D + 2) + D] - M);
      acc0 = fmaf(w, base[c * (D + 2) + 2 * j], acc0);
      acc1 = fmaf(w, base[c * (D + 2) + 2 * j + 1], acc1);
    }
    o[(u64)bh * (D / 2) + j] = pack_bf16x2(acc0 * invL, acc1 * invL);
  }
}

}  // namespace

// scratch must hold b*hq*nchunks*(d+2) floats (unused when nchunks == 1).
// T_dev != null => graph mode: T read on device, grid sized for max_T
// (pass max_T as T; nchunks = ceil(max_T/256)).
hipError_t lfmq_attn_decode(const void* q, const void* kc, const void* vc,
                            void* o, void* scratch, int b, int hq, int hkv,
                            int tmax, int T, const void* T_dev, int d,
                            float scale, hipStream_t stream) {
  if (d != 128 && d != 64) return hipErrorInvalidValue;
  if (hq % hkv) return hipErrorInvalidValue;
  const int nchunks = (T + TPB - 1) / TPB;
  dim3 gridA(b * hq, nchunks), blockA(TPB);
  dim3 gridB(b * hq), blockB(128);
#define LFMQ_ATTN(DD)                                                         \
  do {                                                                        \
    hipLaunchKernelGGL((attn_decode_partial<DD>), gridA, blockA, 0, stream,   \
                       (const u32*)q, (const u32*)kc, (const u32*)vc,         \
                       (float*)scratch, (u32*)o, b, hq, hkv, tmax, T,         \
                       (const i32*)T_dev, scale, nchunks);                    \
    if (nchunks > 1)                                                          \
      hipLaunchKernelGGL((attn_decode_combine<DD>), gridB, blockB, 0, stream, \
                         (const float*)scratch, (u32*)o, nchunks, T,          \
                         (const i32*)T_dev, TPB);                             \
  } while (0)
  if (d == 128)
    LFMQ_ATTN(128);
  else
    LFMQ_ATTN(64);
#undef LFMQ_ATTN
  return hipGetLastError();
}

// Fused rope+append+attention decode entry (s == 1).  qkv is the raw
// [b, (hq+2hkv)*d] projection output; writes the new k/v cache row
// itself.  scratch sized as lfmq_attn_decode.
// tickets != null: the last-arriving chunk combines in-launch (no combine
// kernel).  tickets = zero-initialized u32[b*hq]; the winner resets its
// entry, so graph replays stay correct with no host re-zeroing.
hipError_t lfmq_attn_decode_fused(const void* qkv, void* kc, void* vc,
                                  void* o, void* scratch,
                                  const void* inv_freq, int b, int hq,
                                  int hkv, int tmax, int T, const void* T_dev,
                                  int d, float scale, void* tickets,
                                  hipStream_t stream) {
  if (d != 128 && d != 64) return hipErrorInvalidValue;
  if (hq % hkv) return hipErrorInvalidValue;
  const int nchunks = (T + 63) / 64;  // FCH = 64
  dim3 gridA(b * hq, nchunks), blockA(TPB);
  dim3 gridB(b * hq), blockB(128);
#define LFMQ_ATTNF(DD)                                                        \
  do {                                                                        \
    hipLaunchKernelGGL((attn_decode_fused<DD>), gridA, blockA, 0, stream,     \
                       (const u32*)qkv, (u32*)kc, (u32*)vc, (float*)scratch,  \
                       (u32*)o, (const float*)inv_freq, b, hq, hkv, tmax, T,  \
                       (const i32*)T_dev, scale, nchunks, (u32*)tickets);     \
    if (nchunks > 1 && tickets == nullptr)                                    \
      hipLaunchKernelGGL((attn_decode_combine<DD>), gridB, blockB, 0, stream, \
                         (const float*)scratch, (u32*)o, nchunks, T,          \
                         (const i32*)T_dev, 64);                              \
  } while (0)
  if (d == 128)
    LFMQ_ATTNF(128);
  else
    LFMQ_ATTNF(64);
#undef LFMQ_ATTNF
  return hipGetLastError();
}
