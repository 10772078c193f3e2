// Persistent-engine primitives for the MI355X decode megakernel.
//
// These implement, verbatim where the guide gives code, the CDNA4
// inter-workgroup patterns of cdna_hip_programming.md §6 Guideline 16
// (agent-scope release/acquire, 8-byte {tag,data} granules) and the
// §5.6 weight-streaming engine structure (LDS-DMA loader wave + consumer
// waves over a 16 KiB-slot ring, nt weight stream).
//
// CRITICAL hardware facts this encodes (MI355X_MICROARCH.md):
//  - per-CU L1 is never refreshed by another CU's stores; per-XCD L2s
//    are not coherent: every cross-workgroup word moves ONLY via
//    agent-scope atomics (sc1) — R2 granules here.
//  - R2 form: the data IS the flag: one naturally-aligned 8-byte
//    {tag:32, value:32} per granule, one relaxed agent store, no fence;
//    consumer re-reads granules until every tag matches the epoch.
//  - epochs count within the call (phase index + 1, never 0); every
//    granule word is zeroed by a hipMemsetAsync before EVERY launch.
//  - every spin is bounded and surfaces a give-up code.
#pragma once
#include "common.h"

namespace lfmq_engine {

#define LFMQ_RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT
#define LFMQ_WG_SCOPE __HIP_MEMORY_SCOPE_WORKGROUP

typedef __attribute__((address_space(1))) unsigned int gu32;
typedef __attribute__((address_space(1))) unsigned long long gu64;

// bounded-spin budget: ~2^22 polls x s_sleep(2) ~= 0.25 s of wall per
// site — far beyond any healthy step; a trip means a lost producer and
// must surface FAST (cascading give-ups across layers must not look like
// a hung GPU to the harness).
constexpr unsigned SPIN_MAX = 1u << 22;

__device__ __forceinline__ bool spin_fail(unsigned& spins, gu32* tmo,
                                          unsigned code, int lane) {
  if (++spins < SPIN_MAX) {
    __builtin_amdgcn_s_sleep(2);
    return false;
  }
  // write-once: the FIRST give-up code is the diagnostic one — later
  // give-ups are cascade effects of the first missing producer.
  if (lane == 0 && tmo != nullptr) {
    unsigned expected = 0;
    __hip_atomic_compare_exchange_strong(tmo, &expected, code,
                                         __ATOMIC_RELAXED, __ATOMIC_RELAXED,
                                         __HIP_MEMORY_SCOPE_AGENT);
    // class bitmask of every give-up site (tmo[7]): bit = code high nibble
    __hip_atomic_fetch_or(tmo + 7, 1u << ((code >> 4) & 15), LFMQ_RLX_AGENT);
  }
  return true;
}

// ---- R2 granule publish: ONE aligned 8-byte relaxed agent store ----------
__device__ __forceinline__ void store_granule(gu64* g, unsigned epoch,
                                              unsigned value) {
  __hip_atomic_store(g, ((unsigned long long)epoch << 32) | value,
                     LFMQ_RLX_AGENT);
}

// ---- R2 granule sweep + LDS staging --------------------------------------
// ONE wave re-reads its lane-strided share of cnt granules every pass until
// every tag == epoch, ds-writing the payload dword into lds_dst[idx] as it
// goes (the final pass rewrites everything with matched values).  Returns
// false on spin give-up (outputs poisoned by the caller's memset protocol).
__device__ __forceinline__ bool sweep_stage(gu64* g, int cnt, unsigned epoch,
                                            u32* lds_dst, gu32* tmo,
                                            unsigned code, int lane,
                                            unsigned* passes = nullptr) {
  for (unsigned spins = 0;;) {
    bool ok = true;
    for (int idx = lane; idx < cnt; idx += LFMQ_WAVE) {
      const unsigned long long x = __hip_atomic_load(g + idx, LFMQ_RLX_AGENT);
      lds_dst[idx] = (unsigned)x;
      ok &= (unsigned)(x >> 32) == epoch;
    }
    if (passes) ++*passes;
    if (__all(ok)) return true;
    if (spin_fail(spins, tmo, code, lane)) return false;
    // back off: each pass re-reads cnt/64 sc1 granules per lane — real
    // fabric traffic from every CU; widen the interval once the edge is
    // clearly not ready yet (keeps gather latency, cuts poll bandwidth).
    if (spins > 2) __builtin_amdgcn_s_sleep(16);
  }
}

// variant without staging (caller consumes values in registers): sweeps its
// share into vals[] (each lane's granules at lane, lane+64, ...)
template <int MAXN>
__device__ __forceinline__ bool sweep_vals(gu64* g, int cnt, unsigned epoch,
                                           unsigned (&vals)[MAXN], gu32* tmo,
                                           unsigned code, int lane) {
  for (unsigned spins = 0;;) {
    bool ok = true;
    int j = 0;
    for (int idx = lane; idx < cnt; idx += LFMQ_WAVE, ++j) {
      const unsigned long long x = __hip_atomic_load(g + idx, LFMQ_RLX_AGENT);
      vals[j] = (unsigned)x;
      ok &= (unsigned)(x >> 32) == epoch;
    }
    if (__all(ok)) return true;
    if (spin_fail(spins, tmo, code, lane)) return false;
  }
}

// ---- intra-workgroup flags (LDS, workgroup scope) -------------------------
// gather/loader/consumer waves coordinate through LDS words: release store
// inserts the lgkmcnt drain ahead of the flag; acquire load orders the
// reader's later LDS reads after it.
__device__ __forceinline__ void lds_flag_store(unsigned* f, unsigned v) {
  __hip_atomic_store(f, v, __ATOMIC_RELEASE, LFMQ_WG_SCOPE);
}
// RELAXED variant for the loader's fill publication: a release store here
// would make the compiler drain vmcnt(0) (LDS-DMA completion is tracked by
// vmcnt), killing the fill pipeline.  The caller orders the store itself
// with an explicit lagged `s_waitcnt vmcnt(N)` asm (whose "memory" clobber
// keeps this store below it).
__device__ __forceinline__ void lds_flag_store_rlx(unsigned* f, unsigned v) {
  __hip_atomic_store(f, v, __ATOMIC_RELAXED, LFMQ_WG_SCOPE);
}
__device__ __forceinline__ unsigned lds_flag_load(const unsigned* f) {
  return __hip_atomic_load(f, __ATOMIC_ACQUIRE, LFMQ_WG_SCOPE);
}
// bounded poll until *f >= v (monotonic counters)
__device__ __forceinline__ bool lds_wait_ge(const unsigned* f, unsigned v,
                                            gu32* tmo, unsigned code,
                                            int lane,
                                            unsigned* acc = nullptr) {
  for (unsigned spins = 0;;) {
    if (lds_flag_load(f) >= v) {
      if (acc) *acc += spins;
      return true;
    }
    if (spin_fail(spins, tmo, code, lane)) return false;
  }
}

// ---- loader-side LDS access in asm ----------------------------------------
// hipcc tracks global_load_lds completion through vmcnt and conservatively
// drains vmcnt(0) before ANY visible ds_read/ds_write while fills are in
// flight — which would serialize the loader every time it polls consumer
// progress or publishes a slot.  These asm forms (guide §5.7 form (i):
// loads + s_waitcnt in ONE statement, early-clobber outputs) keep the fill
// pipeline open; ordering vs the DMA is the caller's explicit lagged
// s_waitcnt (in loader_publish) — never hipcc's.

// LDS byte offset of a __shared__ object, for asm ds operands
__device__ __forceinline__ unsigned lds_off(const void* p) {
  return (unsigned)(unsigned long long)(
      __attribute__((address_space(3))) const void*)p;
}

// read 3 LDS words (consumer progress counters) without a vmcnt drain
__device__ __forceinline__ void loader_read3(unsigned off0, unsigned off1,
                                             unsigned off2, unsigned& v0,
                                             unsigned& v1, unsigned& v2) {
  asm volatile(
      "ds_read_b32 %0, %3\n\t"
      "ds_read_b32 %1, %4\n\t"
      "ds_read_b32 %2, %5\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(v0), "=&v"(v1), "=&v"(v2)
      : "v"(off0), "v"(off1), "v"(off2)
      : "memory");
}

__device__ __forceinline__ unsigned loader_read1(unsigned off) {
  unsigned v;
  asm volatile("ds_read_b32 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(v) : "v"(off) : "memory");
  return v;
}

// publish a fill-progress value: wait until at most VMLAG fill instructions
// remain outstanding (i.e. everything older has LANDED in LDS), then
// ds_write the counter — all inside one asm so hipcc adds no drain.
template <int VMLAG>
__device__ __forceinline__ void loader_publish(unsigned flag_off,
                                               unsigned value) {
  asm volatile(
      "s_waitcnt vmcnt(%c2)\n\t"
      "ds_write_b32 %0, %1"
      :
      : "v"(flag_off), "v"(value), "i"(VMLAG)
      : "memory");
}

// ---- LDS-DMA fill: one 1-KiB chunk (64 lanes x 16 B), nt (aux = 2) --------
// gsrc: this lane's 16-B-aligned global source; lds_base: wave-uniform LDS
// byte destination (lane data lands at lds_base + lane*16).
typedef __attribute__((address_space(1))) const void* lfmq_gptr;
typedef __attribute__((address_space(3))) void* lfmq_lptr;

__device__ __forceinline__ void glds_fill_1k(const char* gsrc_lane,
                                             char* lds_base /*__shared__*/) {
  __builtin_amdgcn_global_load_lds((lfmq_gptr)gsrc_lane, (lfmq_lptr)lds_base,
                                   16, 0, /*aux nt*/ 2);
}

// ---- record dequant-dot, stream format v2 -------------------------------
// Record layout (builder: lfmq/models/engine.py):
//   [k/2 B nibbles, LANE-MAJOR dwords: dword for (iter i, lane l) at
//    (l*iters + i)*4 — one b128 per lane fetches a whole 4-iter chunk]
//   [G x 8 B fp32 (s, s*(z+128)) pairs — one broadcast b64 per iter
//    replaces the v1 u16+u8 scalar reads and their convert math]
//   [pad to 16 B]
// A chunk is 4 k-iterations (2048 values); 4 must divide k/512 (host
// asserts k % 2048 == 0) so chunk bodies are BRANCH-FREE (a runtime break
// inside #pragma unroll defeats the unroll).  Chunks are DOUBLE-BUFFERED:
// at 1 wave/SIMD nothing else hides the ~64-cycle LDS latency, so the
// wave's own ILP must (isolated microbench: 0.78 us/record chunk-serial,
// dominated by per-LDS-instruction cost — hence format v2's fewer, wider
// reads).
typedef float f32x2 __attribute__((ext_vector_type(2)));

struct RecChunk {
  u32x4 w;        // 4 iters of nibble dwords for this lane
  u32x4 xq[4];
  f32x2 ssz[4];   // (s, s*(z+128)) per iter
};

__device__ __forceinline__ void chunk_load(RecChunk& ch, int c0,
                                           const char* lds_rec, int iters,
                                           const char* lds_src, int log2g,
                                           int lane) {
  ch.w = *(const u32x4*)(lds_rec + (lane * iters + c0) * 4);
  const f32x2* sbase = (const f32x2*)(lds_rec + iters * 256);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int k0 = (c0 + j) * 512 + lane * 8;
    ch.xq[j] = *(const u32x4*)(lds_src + k0 * 2);
    ch.ssz[j] = sbase[k0 >> log2g];
  }
}

__device__ __forceinline__ void chunk_dot(const RecChunk& ch, float& acc) {
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const DequantPairs4 dq = dequant4_pairs(ch.w[j]);
    float cx = 0.f, xs = 0.f;
    cx = dot2_bf16(dq.p01, ch.xq[j][0], cx);
    cx = dot2_bf16(dq.p23, ch.xq[j][1], cx);
    cx = dot2_bf16(dq.p45, ch.xq[j][2], cx);
    cx = dot2_bf16(dq.p67, ch.xq[j][3], cx);
    xs = dot2_bf16(0x3F803F80u, ch.xq[j][0], xs);
    xs = dot2_bf16(0x3F803F80u, ch.xq[j][1], xs);
    xs = dot2_bf16(0x3F803F80u, ch.xq[j][2], xs);
    xs = dot2_bf16(0x3F803F80u, ch.xq[j][3], xs);
    acc = fmaf(ch.ssz[j][0], cx, acc);
    acc = fmaf(-ch.ssz[j][1], xs, acc);
  }
}

__device__ __forceinline__ float record_dot_nored(const char* lds_rec,
                                                  const char* lds_src, int k,
                                                  int G, int log2g, int lane) {
  (void)G;
  const int iters = k >> 9;
  float acc = 0.f;
  RecChunk A, B;
  chunk_load(A, 0, lds_rec, iters, lds_src, log2g, lane);
  int c0 = 0;
  for (;;) {  // ping-pong; iters may be an odd number of chunks (down: 7)
    if (c0 + 4 < iters)
      chunk_load(B, c0 + 4, lds_rec, iters, lds_src, log2g, lane);
    chunk_dot(A, acc);
    c0 += 4;
    if (c0 >= iters) break;
    if (c0 + 4 < iters)
      chunk_load(A, c0 + 4, lds_rec, iters, lds_src, log2g, lane);
    chunk_dot(B, acc);
    c0 += 4;
    if (c0 >= iters) break;
  }
  return acc;
}

// TWO records interleaved: a single record's chunk chain is latency-bound
// at 1 wave/SIMD (load -> dot -> load ...); a pair gives two independent
// chains and roughly doubles the load-level parallelism.  Records are
// published in pairs anyway (one granule per two outputs).
__device__ __forceinline__ void record_pair_dot(
    const char* lds_recA, const char* lds_recB, const char* lds_src, int k,
    int G, int log2g, int lane, float& vA, float& vB) {
  (void)G;
  const int iters = k >> 9;
  float accA = 0.f, accB = 0.f;
  RecChunk A0, A1, B0, B1;
  chunk_load(A0, 0, lds_recA, iters, lds_src, log2g, lane);
  chunk_load(B0, 0, lds_recB, iters, lds_src, log2g, lane);
  int c0 = 0;
  for (;;) {
    if (c0 + 4 < iters) {
      chunk_load(A1, c0 + 4, lds_recA, iters, lds_src, log2g, lane);
      chunk_load(B1, c0 + 4, lds_recB, iters, lds_src, log2g, lane);
    }
    chunk_dot(A0, accA);
    chunk_dot(B0, accB);
    c0 += 4;
    if (c0 >= iters) break;
    if (c0 + 4 < iters) {
      chunk_load(A0, c0 + 4, lds_recA, iters, lds_src, log2g, lane);
      chunk_load(B0, c0 + 4, lds_recB, iters, lds_src, log2g, lane);
    }
    chunk_dot(A1, accA);
    chunk_dot(B1, accB);
    c0 += 4;
    if (c0 >= iters) break;
  }
  vA = wave_reduce_sum(accA);
  vB = wave_reduce_sum(accB);
}

__device__ __forceinline__ float record_dot(const char* lds_rec,
                                            const char* lds_src, int k, int G,
                                            int log2g, int lane) {
  // shfl (ds_bpermute) reduce: the DPP ladder measured faster in isolation
  // but produced timing-dependent WRONG sums under full engine load
  // (race_hunt detector 0xC0: identical input, different dot) — a DPP
  // read-after-VALU hazard the compiler does not pad reliably here.
  return wave_reduce_sum(
      record_dot_nored(lds_rec, lds_src, k, G, log2g, lane));
}

}  // namespace lfmq_engine
