// Persistent weight-streaming decode engine for Llama-family W4A16 decode
// on MI355X (round-2 headline work; design: docs/persistent_decode_design.md,
// structure: cdna_hip_programming.md §5.6 + Guideline 16,
// prices: MI355X_MICROARCH.md persistent-kernel table).
//
// ONE kernel launch executes the WHOLE 32-layer decode step (b=1, s=1):
// embedding read, per layer [norm+qkv GEMV] -> [rope+append+attention] ->
// [o GEMV+resid] -> [norm+gate_up GEMV+silu] -> [down GEMV+resid], and the
// final RMSNorm.  Replaces ~194 launches/step whose ~3-4 us fixed cost was
// measured as ~0.7 ms of the 1.97 ms round-1 step (profiles/decode_profile_
// r01.md).  lm_head + sampling remain separate launches.
//
// Geometry: grid = 256 workgroups (one per CU, all co-resident — verified
// by csrc/engine_probe.hip census), 4 waves each:
//   wave 3           = loader: streams this CU's packed-weight byte stream
//                      through a 16 KiB-slot LDS ring via global_load_lds
//                      (nt, aux=2), run-ahead bounded by the ring, slot
//                      publication lagged by an explicit vmcnt (all loader
//                      LDS accesses are inline asm so hipcc inserts no
//                      vmcnt(0) drains — see engine_common.h).
//   waves 0-2        = consumers: dequant+dot records out of the ring
//                      (v_perm nibble dequant + v_dot2 bf16, the proven
//                      round-1 GEMV inner idioms), attention units, and
//                      3-way-split granule gathers; wave 0 leads (norm
//                      scalars, phase publication).
//
// Cross-CU dataflow: every op output is published as R2 granules (8-byte
// {epoch,2xbf16} relaxed agent stores — the data IS the flag) and gathered
// by all CUs; epochs are 6L+{1..6}, granule buffers are hipMemset to zero
// before EVERY launch (graph memset node).  Edge-buffer reuse period (one
// layer) exceeds the max CU run-ahead (one edge), so overwrite races are
// structurally impossible: a CU can publish edge E of layer L+1 only after
// gathering an edge of layer L that required every other CU to have passed
// its own E(L) publish.
//
// Weight stream layout (built by lfmq/models/engine.py, one contiguous
// region per CU, identical lengths): per layer, per op in [qkv, o,
// gate_up, down], per owned column record:
//   [k/2 B nibbles (row-major packed dwords — identical to the at-rest
//    qweight row)] [G x 2 B bf16 scales] [ceil(G/2) B packed int4 zeros]
//   [pad to 16 B]
// Records never straddle the ring wrap (ring_bytes boundary): the builder
// and the kernel both skip to the next ring boundary when one would.
// gate_up interleaves records [gate_j, up_j, ...] so one wave computes
// silu(g)*u locally.  Column ownership: CU c owns columns
// [c*n/256, (c+1)*n/256) of every op — all Llama n dims divide by 256.
#include <algorithm>

#include "common.h"
#include "engine_args.h"
#include "engine_common.h"

using namespace lfmq_engine;

namespace {

constexpr int SLOT = 16384;
constexpr int LAG = 4;     // publish slot s-LAG+1 after issuing slot s
constexpr int AT = 256;    // attention tokens per unit (one wave per unit)
constexpr int NCU = 256;
// 1 loader + NCONS consumer waves per CU.  NCONS=7 (8 waves, 2/SIMD) was
// tried for latency hiding but stalls in the gate_up/down ring window —
// parked pending a fill-flow redesign; NCONS=3 is the validated geometry.
constexpr int NCONS = 3;

struct Ctl {
  unsigned filled;       // loader: slots published
  unsigned consumed[NCONS];  // consumer waves: slots fully passed
  unsigned gath;         // gather arrivals (monotonic, +3 per event)
  unsigned phase;        // lead-published phase (monotonic)
  unsigned busy;         // gather window hint for loader thinning
  float inv_cur;         // norm scalar of the current phase
  float ss_part[NCONS];  // per-wave sum-of-squares partials
};

// record byte size v2 (padded to 16) — device twin of lfmq_engine_rec_bytes
__device__ inline int rec_bytes(int k, int G) {
  return (k / 2 + G * 8 + 15) & ~15;
}

// ---- record walking (must mirror lfmq/models/engine.py exactly) ----------
struct Cursor {
  u64 off;       // next free byte in the CU stream
  u32 ring_pos;  // off % ring_bytes
};
__device__ __forceinline__ u64 cursor_next(Cursor& c, int rec, u32 ring_bytes) {
  if (c.ring_pos + (u32)rec > ring_bytes) {  // would straddle the ring wrap
    c.off += ring_bytes - c.ring_pos;
    c.ring_pos = 0;
  }
  const u64 at = c.off;
  c.off += rec;
  c.ring_pos += rec;
  if (c.ring_pos == ring_bytes) c.ring_pos = 0;
  return at;
}

// ---- the dequant-dot of one record out of the ring ------------------------
// lane covers k indices {i*512 + lane*8 .. +8} for i in 0..k/512-1: the
// nibble dword at rec + (i*64+lane)*4 is exactly qweight dword (k0/8), x is
// one conflict-free ds_read_b128 (stride 16 B across lanes).  Math matches
// csrc/dq_gemv.hip: s*( sum (128+c)*x ) - s*(z+128)*sum x, fp32 accumulate.
// Chunked two-phase pipeline: issue a whole chunk's LDS loads up front
// (independent), then compute — the round-1 GEMV lesson (csrc/dq_gemv.hip
// KI-unroll): per-iteration loads make the compiler reuse registers and
// serialize the loop on LDS latency.  CHUNK=8 covers k=4096 ops in one
// chunk; k=14336 (down) runs 3.5 chunks.
// wait until the ring holds stream bytes [0, end_off): filled*SLOT >= end
__device__ __forceinline__ bool wait_fill(Ctl* ctl, u64 end_off, gu32* tmo,
                                          unsigned code, int lane,
                                          unsigned* acc = nullptr) {
  const unsigned need = (unsigned)((end_off + SLOT - 1) / SLOT);
  return lds_wait_ge(&ctl->filled, need, tmo, code, lane, acc);
}

// ---- 3-way gather helpers --------------------------------------------------
// stage this wave's third of cnt granules into lds_dst (u32 payload each),
// then bump the arrival counter.  Caller waits for 3*gseq arrivals.
__device__ __forceinline__ bool gather_share(gu64* g, int cnt, unsigned epoch,
                                             u32* lds_dst, int wave, Ctl* ctl,
                                             gu32* tmo, unsigned code,
                                             int lane,
                                             unsigned* passes = nullptr,
                                             unsigned dbg = 0) {
  const int b = cnt * wave / NCONS, e = cnt * (wave + 1) / NCONS;
  if (!sweep_stage(g + b, e - b, epoch, lds_dst + b, tmo, code, lane, passes))
    return false;
  if (dbg & 16) {  // re-read twice: active writer vs stale sweep read
    for (int idx = lane; idx < e - b; idx += LFMQ_WAVE) {
      const unsigned long long x1 = __hip_atomic_load(g + b + idx,
                                                      LFMQ_RLX_AGENT);
      const unsigned long long x2 = __hip_atomic_load(g + b + idx,
                                                      LFMQ_RLX_AGENT);
      if (x1 != x2)
        __hip_atomic_store(tmo + 6, 0xE9u, LFMQ_RLX_AGENT);  // 2nd writer!
      else if ((u32)x1 != lds_dst[b + idx])
        __hip_atomic_store(tmo + 6, 0xEAu, LFMQ_RLX_AGENT);  // stale sweep
    }
  }
  if (lane == 0)
      __hip_atomic_fetch_add(&ctl->gath, 1u, __ATOMIC_ACQ_REL, LFMQ_WG_SCOPE);
  return true;
}
__device__ __forceinline__ bool wait_gath(Ctl* ctl, unsigned gseq, gu32* tmo,
                                          unsigned code, int lane,
                                          unsigned* acc = nullptr) {
  return lds_wait_ge(&ctl->gath, NCONS * gseq, tmo, code, lane, acc);
}

// debug bit 8: read each record twice and byte-checksum it twice.
// checksum mismatch  -> ring bytes changed under us (overwrite race),
//   code 0xB0 | phase in tmo[5]
// checksums equal but dots differ -> compute nondeterminism (hazard in
//   the dot/reduce), code 0xC0 | phase in tmo[5]
__device__ __forceinline__ u32 rec_cksum(const char* lds_rec, int bytes,
                                         int lane) {
  const u32* p = (const u32*)lds_rec;
  u32 s = 0;
  for (int i = lane; i < bytes / 4; i += LFMQ_WAVE) s += p[i];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) s += __shfl_xor(s, off, 64);
  return s;
}
__device__ __forceinline__ float record_dot_checked(
    const char* lds_rec, const char* lds_src, int k, int G, int log2g,
    int lane, unsigned dbg, gu32* tmo, unsigned code) {
  if (!(dbg & 8)) return record_dot(lds_rec, lds_src, k, G, log2g, lane);
  // bracket BOTH dots with ring-record AND x-source checksums:
  //   0xB.. ring bytes changed, 0xA.. x bytes changed, 0xC.. checksums
  //   stable but dots differ (true compute nondeterminism)
  // 16-block partial checksums over x locate WHERE instability lands
  u32 xb1[16], xb2[16];
  const int xblk = (k * 2) / 16;
  const u32 r1 = rec_cksum(lds_rec, k / 2 + G * 8, lane);
#pragma unroll
  for (int b = 0; b < 16; ++b) xb1[b] = rec_cksum(lds_src + b * xblk, xblk, lane);
  const float v = record_dot(lds_rec, lds_src, k, G, log2g, lane);
  const u32 r2 = rec_cksum(lds_rec, k / 2 + G * 8, lane);
  const float v2 = record_dot(lds_rec, lds_src, k, G, log2g, lane);
  const u32 r3 = rec_cksum(lds_rec, k / 2 + G * 8, lane);
#pragma unroll
  for (int b = 0; b < 16; ++b) xb2[b] = rec_cksum(lds_src + b * xblk, xblk, lane);
  u32 bm = 0;
#pragma unroll
  for (int b = 0; b < 16; ++b) bm |= (xb1[b] != xb2[b]) ? (1u << b) : 0u;
  if (lane == 0) {
    if (r1 != r2 || r2 != r3)
      __hip_atomic_store(tmo + 5, 0xB00u | (code & 0xF), LFMQ_RLX_AGENT);
    else if (bm) {
      __hip_atomic_store(tmo + 5, 0xA00u | (code & 0xF), LFMQ_RLX_AGENT);
      __hip_atomic_fetch_or(tmo + 7, bm, LFMQ_RLX_AGENT);
    } else if (v != v2) {
      __hip_atomic_store(tmo + 5, 0xC00u | (code & 0xF), LFMQ_RLX_AGENT);
    }
  }
  return v;
}

// Pair form: the fast path runs the interleaved two-record dot (twice the
// independent load chains — the single-record chain is latency-bound at
// 1 wave/SIMD); the debug-8 detector path falls back to the bracketed
// single-record version so the corruption checks stay byte-exact.
__device__ __forceinline__ void record_pair_dot_checked(
    const char* recA, const char* recB, const char* lds_src, int k, int G,
    int log2g, int lane, unsigned dbg, gu32* tmo, unsigned code, float& vA,
    float& vB) {
  if (dbg & 8) {
    vA = record_dot_checked(recA, lds_src, k, G, log2g, lane, dbg, tmo, code);
    vB = record_dot_checked(recB, lds_src, k, G, log2g, lane, dbg, tmo, code);
    return;
  }
  record_pair_dot(recA, recB, lds_src, k, G, log2g, lane, vA, vB);
}

// single-granule bounded wait (combine path)
__device__ __forceinline__ u32 gran_wait(gu64* g, unsigned epoch, gu32* tmo,
                                         unsigned code, int lane, bool* okp) {
  for (unsigned spins = 0;;) {
    const unsigned long long x = __hip_atomic_load(g, LFMQ_RLX_AGENT);
    if ((unsigned)(x >> 32) == epoch) return (u32)x;
    if (spin_fail(spins, tmo, code, lane)) { *okp = false; return 0; }
  }
}

// ============================================================ main kernel
template <int D>
__global__ __launch_bounds__((NCONS + 1) * 64, 1) void lfmq_engine_kernel(EngineArgs a) {
  extern __shared__ char smem[];
  const int cu = blockIdx.x;
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  const u32 ring_bytes = (u32)a.nslot_ring * SLOT;

  char* ring = smem;
  unsigned short* xbuf = (unsigned short*)(smem + ring_bytes);
  char* unionb = (char*)(xbuf + a.hidden);
  const int union_bytes =
      max(a.inter * 2, max(a.nqkv * 2, a.hq * D * 2));
  char* ascr = unionb + union_bytes;   // attention scratch / resid staging
  unsigned short* q_scr = (unsigned short*)ascr;          // [NCONS][D]
  unsigned short* k_scr = q_scr + NCONS * D;              // [NCONS][D]
  float* p_scr = (float*)(k_scr + NCONS * D);             // [NCONS][AT]
  const int ascr_bytes =
      max(NCONS * D * 4 + NCONS * AT * 4, a.hidden * 2);
  Ctl* ctl = (Ctl*)(ascr + ((ascr_bytes + 63) & ~63));

  gu64* g_qkv = (gu64*)a.g_qkv;
  gu64* g_attn = (gu64*)a.g_attn;
  gu64* g_o = (gu64*)a.g_o;
  gu64* g_h = (gu64*)a.g_h;
  gu64* g_down = (gu64*)a.g_down;
  gu64* g_scr = (gu64*)a.g_scr;
  gu32* tmo = (gu32*)a.tmo;

  if (threadIdx.x == 0) {
    ctl->filled = 0;
    for (int w = 0; w < NCONS; ++w) ctl->consumed[w] = 0;
    ctl->gath = 0;
    ctl->phase = 0;
    ctl->busy = 0;
  }
  __syncthreads();  // last convergent barrier; waves diverge below

  // ======================================================== loader wave
  if (wave == NCONS) {
    const char* wg_src = a.stream + (u64)cu * a.stream_per_cu;
    const unsigned f_off = lds_off(&ctl->filled);
    unsigned c_off[NCONS];
    for (int w = 0; w < NCONS; ++w) c_off[w] = lds_off(&ctl->consumed[w]);
    const unsigned b_off = lds_off(&ctl->busy);
    const int total = a.n_slots_total;
    unsigned last_pub = 0;
    unsigned poll_spins_tot = 0;
#define ENG_PUB(VML, val)                                  \
  do {                                                     \
    const unsigned v_ = (val);                             \
    if (v_ > last_pub) {                                   \
      loader_publish<VML>(f_off, v_);                      \
      last_pub = v_;                                       \
    }                                                      \
  } while (0)
    for (int s = 0; s < total; ++s) {
      if (s >= a.nslot_ring) {
        // about to stall for ring space: ladder-publish the issued slots
        // as they land (otherwise up to LAG-1 landed slots stay invisible
        // and consumers lock-step with the loader across every stall);
        // progressive vmcnt steps hand the oldest slots over immediately.
        // Publishes are MONOTONIC (ENG_PUB): a later lagged in-loop
        // publish must never overwrite a ladder publish with a smaller
        // value — consumers poll `filled >= need` and a regressed value
        // re-blocks them on slots that landed long ago.
        if (s >= 3) {
          ENG_PUB(32, (unsigned)(s - 2));
          ENG_PUB(16, (unsigned)(s - 1));
        }
        ENG_PUB(0, (unsigned)s);
        unsigned spins = 0;
        for (;;) {
          unsigned m = 0xFFFFFFFFu;
          static_assert(NCONS % 3 == 0 || NCONS == 7, "poll reads in 3s");
          for (int w = 0; w + 3 <= NCONS; w += 3) {
            unsigned d0, d1, d2;
            loader_read3(c_off[w], c_off[w + 1], c_off[w + 2], d0, d1, d2);
            m = min(m, min(d0, min(d1, d2)));
          }
          if (NCONS % 3 == 1) m = min(m, loader_read1(c_off[NCONS - 1]));
          if ((int)m >= s - a.nslot_ring + 1) break;
          if (spin_fail(spins, tmo, 0x10, lane)) return;
        }
        poll_spins_tot += spins;
      }
      char* lbase = ring + (s % a.nslot_ring) * SLOT;
      const char* g = wg_src + (u64)s * SLOT + lane * 16;
#pragma unroll
      for (int i = 0; i < 16; ++i) glds_fill_1k(g + i * 1024, lbase + i * 1024);
      // loader timeline (debug): issue timestamp per slot, CU 0
      if (a.prof && cu == 0 && lane == 0)
        a.prof[(u64)a.n_layers * 8 + s] = __builtin_amdgcn_s_memtime();
      {  // gather-pass thinning: one outstanding fill while the CU sweeps
        unsigned busy;
        asm volatile("ds_read_b32 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                     : "=&v"(busy) : "v"(b_off) : "memory");
        if (busy) asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
      }
      if (a.debug & 1) {
        loader_publish<0>(f_off, (unsigned)(s + 1));
        last_pub = (unsigned)(s + 1);
      } else if (s >= LAG - 1) {
        ENG_PUB((LAG - 1) * 16, (unsigned)(s - LAG + 2));
      }
    }
    ENG_PUB(0, (unsigned)total);
#undef ENG_PUB
    if (a.prof && lane == 0)  // loader space-poll spins (all CUs summed)
      __hip_atomic_fetch_add((gu32*)a.tmo + 2, poll_spins_tot,
                             LFMQ_RLX_AGENT);
    return;
  }

  // ===================================================== consumer waves
  const int rep = a.hq / a.hkv;
  const int T = *a.T_dev;
  const int pos = T - 1;
  const int nch = (T + AT - 1) / AT;
  const int G1 = a.hidden >> a.log2g;       // qkv / gate_up k-groups
  const int Go = (a.hq * D) >> a.log2g;     // o k-groups
  const int G2 = a.inter >> a.log2g;        // down k-groups
  const int rb_qkv = rec_bytes(a.hidden, G1);
  const int rb_o = rec_bytes(a.hq * D, Go);
  const int rb_gu = rec_bytes(a.hidden, G1);
  const int rb_dn = rec_bytes(a.inter, G2);
  const int npc_qkv = a.nqkv / NCU;
  const int npc_o = a.hidden / NCU;
  const int npc_gu = 2 * a.inter / NCU;
  const int npc_dn = a.hidden / NCU;

  Cursor cur{0, 0};
  unsigned gseq = 0;
  unsigned phase_want = 0;
  bool ok = true;
  unsigned dbg_fill = 0, dbg_sweep = 0, dbg_sync = 0;
#define ENG_DBG_BARRIER()                                                  \
  do {                                                                     \
    if (a.debug & 2) {                                                     \
      if (lane == 0)                                                       \
        __hip_atomic_fetch_add(&ctl->gath, 1u, __ATOMIC_ACQ_REL,           \
                               LFMQ_WG_SCOPE);                             \
      ++gseq;                                                              \
      ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x7f, lane, &dbg_sync));         \
    }                                                                      \
  } while (0)
#define ENG_DBG_FENCE()                                                    \
  do {                                                                     \
    if (a.debug & 4)                                                       \
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");                   \
  } while (0)

#define ENG_CHECK(x) do { if (!(x)) return; } while (0)
#define ENG_PROF(slot_)                                                   \
  do {                                                                    \
    if (a.prof && cu == 0 && wave == 0 && lane == 0)                      \
      a.prof[(u64)L * 8 + (slot_)] = __builtin_amdgcn_s_memtime();        \
  } while (0)

  // ---- prologue: stage the embedding row into xbuf, 3-way -------------
  {
    const long long tk = *a.tok;
    const u32* erow = (const u32*)(a.embed + (u64)tk * a.hidden);
    u32* x32 = (u32*)xbuf;
    const int nd = a.hidden / 2;
    const int b = nd * wave / NCONS, e = nd * (wave + 1) / NCONS;
    float ss = 0.f;
    for (int j = b + lane; j < e; j += LFMQ_WAVE) {
      const u32 v = erow[j];
      x32[j] = v;
      ss = dot2_bf16(v, v, ss);
    }
    ss = wave_reduce_sum(ss);
    if (lane == 0) ctl->ss_part[wave] = ss;
    if (lane == 0)
      __hip_atomic_fetch_add(&ctl->gath, 1u, __ATOMIC_ACQ_REL, LFMQ_WG_SCOPE);
    ++gseq;
    ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x20, lane, &dbg_sync));
    if (wave == 0) {
      float sst = 0.f;
        for (int w = 0; w < NCONS; ++w) sst += ctl->ss_part[w];
      ctl->inv_cur = rsqrtf(sst / (float)a.hidden + a.rms_eps);
      lds_flag_store(&ctl->phase, ++phase_want);
    } else {
      ++phase_want;
      ENG_CHECK(lds_wait_ge(&ctl->phase, phase_want, tmo, 0x21, lane));
    }
  }

  for (int L = 0; L < a.n_layers; ++L) {
    const unsigned e_qkv = 6 * L + 1, e_scr = 6 * L + 2, e_attn = 6 * L + 3,
                   e_o = 6 * L + 4, e_h = 6 * L + 5, e_down = 6 * L + 6;
    ENG_PROF(0);

    // ---------------- P1: qkv records (norm-folded, scale inv_cur) -----
    {
      const float inv = ctl->inv_cur;
      for (int rp = 0; rp < npc_qkv / 2; ++rp) {
        const u64 atA = cursor_next(cur, rb_qkv, ring_bytes);
        const u64 atB = cursor_next(cur, rb_qkv, ring_bytes);
        if (rp % NCONS != wave) continue;
        // fills are stream-monotonic: waiting for B's end covers A
        ENG_CHECK(wait_fill(ctl, atB + rb_qkv, tmo, 0x30, lane, &dbg_fill));
        float vA, vB;
        record_pair_dot_checked(ring + (u32)(atA % ring_bytes),
                                ring + (u32)(atB % ring_bytes),
                                (const char*)xbuf, a.hidden, G1, a.log2g,
                                lane, a.debug, tmo, 0xE1, vA, vB);
        if (lane == 0)
          store_granule(g_qkv + cu * (npc_qkv / 2) + rp, e_qkv,
                        pack_bf16x2(inv * vA, inv * vB));
        lds_flag_store_rlx(&ctl->consumed[wave], (unsigned)(cur.off / SLOT));
      }
      lds_flag_store_rlx(&ctl->consumed[wave], (unsigned)(cur.off / SLOT));
    }
    ENG_DBG_BARRIER();
    ENG_PROF(1);

    // ---------------- P2: gather qkv into union ------------------------
    if (wave == 0) lds_flag_store_rlx(&ctl->busy, 1u);
    ENG_CHECK(gather_share(g_qkv, a.nqkv / 2, e_qkv, (u32*)unionb, wave, ctl,
                           tmo, 0x31, lane, &dbg_sweep, a.debug));
    ++gseq;
    ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x32, lane, &dbg_sync));
    if (wave == 0) lds_flag_store_rlx(&ctl->busy, 0u);
    ENG_DBG_FENCE();
    ENG_PROF(2);

    // ---------------- P3: attention units ------------------------------
    {
      const unsigned short* qkv16 = (const unsigned short*)unionb;
      const u32* qkv32 = (const u32*)unionb;
      unsigned short* qs = q_scr + wave * D;
      unsigned short* ks = k_scr + wave * D;
      float* ps = p_scr + wave * AT;
      const int total_units = a.hq * nch;
      for (int u = cu + NCU * wave; u < total_units; u += NCU * NCONS) {
        const int h = u / nch, c = u % nch;
        const int hk = h / rep;
        const unsigned short* kcache =
            a.kc + (u64)L * a.kv_layer_stride + (u64)hk * a.tmax * D;
        const unsigned short* vcache =
            a.vc + (u64)L * a.kv_layer_stride + (u64)hk * a.tmax * D;
        // rope q (head h) and k (head hk); lane i holds dims i and i+D/2
        const int i = lane;
        float sv, cv;
        __sincosf((float)pos * a.inv_freq[i], &sv, &cv);
        const float q1 = bf16_bits_to_f32(qkv16[h * D + i]);
        const float q2 = bf16_bits_to_f32(qkv16[h * D + i + D / 2]);
        const float k1 = bf16_bits_to_f32(qkv16[(a.hq + hk) * D + i]);
        const float k2 = bf16_bits_to_f32(qkv16[(a.hq + hk) * D + i + D / 2]);
        const float qlo = q1 * cv - q2 * sv, qhi = q1 * sv + q2 * cv;
        const float klo = k1 * cv - k2 * sv, khi = k1 * sv + k2 * cv;
        qs[i] = (unsigned short)f32_to_bf16_bits_rne(qlo);
        qs[i + D / 2] = (unsigned short)f32_to_bf16_bits_rne(qhi);
        ks[i] = (unsigned short)f32_to_bf16_bits_rne(klo);
        ks[i + D / 2] = (unsigned short)f32_to_bf16_bits_rne(khi);
        const float s_last = wave_reduce_sum(qlo * klo + qhi * khi);
        // cache append by the designated unit (plain stores: nothing reads
        // row `pos` from the cache until the next kernel launch)
        if (h % rep == 0 && c == pos / AT) {
          u32* kdst = (u32*)(kcache + (u64)pos * D);
          u32* vdst = (u32*)(vcache + (u64)pos * D);
          kdst[lane] = (u32)ks[2 * lane] | ((u32)ks[2 * lane + 1] << 16);
          vdst[lane] = qkv32[((u64)a.hq + a.hkv + hk) * (D / 2) + lane];
        }
        // scores: 4 tokens per lane (t = c*AT + j*64 + lane)
        const u32* q32 = (const u32*)qs;
        float sc[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int t = c * AT + j * LFMQ_WAVE + lane;
          float s = -INFINITY;
          if (t < T) {
            if (t == pos) {
              s = s_last * a.attn_scale;
            } else {
              const u32* krow = (const u32*)(kcache + (u64)t * D);
              float acc = 0.f;
#pragma unroll
              for (int q = 0; q < D / 8; ++q) {
                const u32x4 kv4 = *(const u32x4*)(krow + 4 * q);
                acc = dot2_bf16(kv4[0], q32[4 * q + 0], acc);
                acc = dot2_bf16(kv4[1], q32[4 * q + 1], acc);
                acc = dot2_bf16(kv4[2], q32[4 * q + 2], acc);
                acc = dot2_bf16(kv4[3], q32[4 * q + 3], acc);
              }
              s = acc * a.attn_scale;
            }
          }
          sc[j] = s;
        }
        float m = fmaxf(fmaxf(sc[0], sc[1]), fmaxf(sc[2], sc[3]));
        m = wave_reduce_max(m);
        float l = 0.f;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float p = sc[j] == -INFINITY ? 0.f : __expf(sc[j] - m);
          ps[j * LFMQ_WAVE + lane] = p;
          l += p;
        }
        l = wave_reduce_sum(l);
        // PV: lane = output dword (dims 2*lane, 2*lane+1).  4 independent
        // partial pairs + unroll keep ~8 V-row loads in flight (a single
        // accumulate chain serializes on global-load latency — the same
        // structure as csrc/attn_decode.hip's PV pass).
        // branch-free main blocks of 4 + a short tail (a break inside
        // #pragma unroll kills the unroll — same lesson as record_dot)
        float oa[4][2] = {{0.f, 0.f}, {0.f, 0.f}, {0.f, 0.f}, {0.f, 0.f}};
        const int nt = min(AT, T - c * AT);
        const int nt4 = nt & ~3;
        const u32 v_last = qkv32[((u64)a.hq + a.hkv + hk) * (D / 2) + lane];
#pragma unroll 2
        for (int tb = 0; tb < nt4; tb += 4) {
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            const int tt = tb + q;
            const int t = c * AT + tt;
            const u32 vd =
                (t == pos) ? v_last : ((const u32*)(vcache + (u64)t * D))[lane];
            float v0, v1;
            bf16x2_to_f32(vd, v0, v1);
            const float p = ps[tt];
            oa[q][0] = fmaf(p, v0, oa[q][0]);
            oa[q][1] = fmaf(p, v1, oa[q][1]);
          }
        }
        for (int tt = nt4; tt < nt; ++tt) {
          const int t = c * AT + tt;
          const u32 vd =
              (t == pos) ? v_last : ((const u32*)(vcache + (u64)t * D))[lane];
          float v0, v1;
          bf16x2_to_f32(vd, v0, v1);
          const float p = ps[tt];
          oa[tt & 3][0] = fmaf(p, v0, oa[tt & 3][0]);
          oa[tt & 3][1] = fmaf(p, v1, oa[tt & 3][1]);
        }
        const float o0 = (oa[0][0] + oa[1][0]) + (oa[2][0] + oa[3][0]);
        const float o1 = (oa[0][1] + oa[1][1]) + (oa[2][1] + oa[3][1]);
        if (nch == 1) {  // direct output, no combine
          const float invl = 1.f / l;
          store_granule(g_attn + h * (D / 2) + lane, e_attn,
                        pack_bf16x2(o0 * invl, o1 * invl));
        } else {  // publish fp32 partial {o[D], m, l} to scratch
          gu64* base = g_scr + ((u64)h * a.nch_max + c) * (D + 2);
          store_granule(base + 2 * lane, e_scr, __builtin_bit_cast(u32, o0));
          store_granule(base + 2 * lane + 1, e_scr,
                        __builtin_bit_cast(u32, o1));
          if (lane == 0) {
            store_granule(base + D, e_scr, __builtin_bit_cast(u32, m));
            store_granule(base + D + 1, e_scr, __builtin_bit_cast(u32, l));
          }
        }
      }
      // combines: this wave merges every head whose chunk-0 unit it ran
      if (nch > 1) {
        for (int u = cu + NCU * wave; u < total_units; u += NCU * NCONS) {
          if (u % nch != 0) continue;
          const int h = u / nch;
          gu64* hb = g_scr + (u64)h * a.nch_max * (D + 2);
          float mc = -INFINITY, lc = 0.f;
          if (lane < nch) {
            mc = __builtin_bit_cast(float, gran_wait(hb + lane * (D + 2) + D,
                                                     e_scr, tmo, 0x40, lane,
                                                     &ok));
            lc = __builtin_bit_cast(float, gran_wait(
                     hb + lane * (D + 2) + D + 1, e_scr, tmo, 0x41, lane,
                     &ok));
          }
          ENG_CHECK(ok);
          const float M = wave_reduce_max(mc);
          const float wgt = lane < nch ? __expf(mc - M) : 0.f;
          const float Lsum = wave_reduce_sum(wgt * lc);
          float a0 = 0.f, a1 = 0.f;
          for (int c = 0; c < nch; ++c) {
            const float wc = __shfl(wgt, c, 64);
            gu64* cb = hb + c * (D + 2);
            const float p0 = __builtin_bit_cast(
                float, gran_wait(cb + 2 * lane, e_scr, tmo, 0x42, lane, &ok));
            const float p1 = __builtin_bit_cast(
                float,
                gran_wait(cb + 2 * lane + 1, e_scr, tmo, 0x43, lane, &ok));
            a0 = fmaf(wc, p0, a0);
            a1 = fmaf(wc, p1, a1);
          }
          ENG_CHECK(ok);
          const float invL = 1.f / Lsum;
          store_granule(g_attn + h * (D / 2) + lane, e_attn,
                        pack_bf16x2(a0 * invL, a1 * invL));
        }
      }
    }
    ENG_DBG_BARRIER();
    ENG_PROF(3);

    // ---------------- P4: gather attention output ----------------------
    // intra-CU barrier FIRST: the gather stages into unionb, overwriting
    // qkvbuf, while this CU's other waves may still be inside attention
    // units READING qkvbuf (this race was caught as run-to-run engine
    // nondeterminism, max|run1-run2| up to 1e-2).
    if (lane == 0)
      __hip_atomic_fetch_add(&ctl->gath, 1u, __ATOMIC_ACQ_REL, LFMQ_WG_SCOPE);
    ++gseq;
    ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x4f, lane, &dbg_sync));
    if (wave == 0) lds_flag_store_rlx(&ctl->busy, 1u);
    ENG_CHECK(gather_share(g_attn, a.hq * D / 2, e_attn, (u32*)unionb, wave,
                           ctl, tmo, 0x50, lane, &dbg_sweep, a.debug));
    ++gseq;
    ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x51, lane, &dbg_sync));
    if (wave == 0) lds_flag_store_rlx(&ctl->busy, 0u);
    ENG_DBG_FENCE();
    ENG_PROF(4);

    // ---------------- P5: o records (src = attnbuf) --------------------
    {
      for (int rp = 0; rp < npc_o / 2; ++rp) {
        const u64 atA = cursor_next(cur, rb_o, ring_bytes);
        const u64 atB = cursor_next(cur, rb_o, ring_bytes);
        if (rp % NCONS != wave) continue;
        ENG_CHECK(wait_fill(ctl, atB + rb_o, tmo, 0x52, lane, &dbg_fill));
        float vA, vB;
        record_pair_dot_checked(ring + (u32)(atA % ring_bytes),
                                ring + (u32)(atB % ring_bytes), unionb,
                                a.hq * D, Go, a.log2g, lane, a.debug, tmo,
                                0xE2, vA, vB);
        if (lane == 0)
          store_granule(g_o + cu * (npc_o / 2) + rp, e_o,
                        pack_bf16x2(vA, vB));
      }
      lds_flag_store_rlx(&ctl->consumed[wave], (unsigned)(cur.off / SLOT));
    }

    // ---------------- P6: gather o, residual-add into xbuf, inv2 -------
    if (wave == 0) lds_flag_store_rlx(&ctl->busy, 1u);
    ENG_CHECK(gather_share(g_o, a.hidden / 2, e_o, (u32*)ascr, wave, ctl, tmo,
                           0x53, lane, &dbg_sweep, a.debug));
    ++gseq;
    ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x54, lane, &dbg_sync));
    ENG_DBG_FENCE();
    {
      u32* x32 = (u32*)xbuf;
      const u32* add32 = (const u32*)ascr;
      const int nd = a.hidden / 2;
      const int b = nd * wave / NCONS, e = nd * (wave + 1) / NCONS;
      float ss = 0.f;
      for (int j = b + lane; j < e; j += LFMQ_WAVE) {
        float x0, x1, a0, a1;
        bf16x2_to_f32(x32[j], x0, x1);
        bf16x2_to_f32(add32[j], a0, a1);
        x0 += a0;
        x1 += a1;
        const u32 nv = pack_bf16x2(x0, x1);
        x32[j] = nv;
        ss = dot2_bf16(nv, nv, ss);
      }
      ss = wave_reduce_sum(ss);
      if (lane == 0) ctl->ss_part[wave] = ss;
      if (lane == 0)
      __hip_atomic_fetch_add(&ctl->gath, 1u, __ATOMIC_ACQ_REL, LFMQ_WG_SCOPE);
      ++gseq;
      ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x55, lane, &dbg_sync));
      if (wave == 0) {
        lds_flag_store_rlx(&ctl->busy, 0u);
        float sst = 0.f;
        for (int w = 0; w < NCONS; ++w) sst += ctl->ss_part[w];
        ctl->inv_cur = rsqrtf(sst / (float)a.hidden + a.rms_eps);
        lds_flag_store(&ctl->phase, ++phase_want);
      } else {
        ++phase_want;
        ENG_CHECK(lds_wait_ge(&ctl->phase, phase_want, tmo, 0x56, lane));
      }
    }
    ENG_PROF(5);

    // ---------------- P7: gate_up records (scale inv2, silu pairs) -----
    {
      const float inv = ctl->inv_cur;
      for (int rq = 0; rq < npc_gu / 4; ++rq) {
        // quad = (g0, u0, g1, u1): two pair-dots, silu(g)*u per pair
        const u64 at0 = cursor_next(cur, rb_gu, ring_bytes);
        const u64 at1 = cursor_next(cur, rb_gu, ring_bytes);
        const u64 at2 = cursor_next(cur, rb_gu, ring_bytes);
        const u64 at3 = cursor_next(cur, rb_gu, ring_bytes);
        if (rq % NCONS != wave) continue;
        ENG_CHECK(wait_fill(ctl, at1 + rb_gu, tmo, 0x60, lane, &dbg_fill));
        float g0, u0;
        record_pair_dot_checked(ring + (u32)(at0 % ring_bytes),
                                ring + (u32)(at1 % ring_bytes),
                                (const char*)xbuf, a.hidden, G1, a.log2g,
                                lane, a.debug, tmo, 0xE3, g0, u0);
        ENG_CHECK(wait_fill(ctl, at3 + rb_gu, tmo, 0x60, lane, &dbg_fill));
        float g1, u1;
        record_pair_dot_checked(ring + (u32)(at2 % ring_bytes),
                                ring + (u32)(at3 % ring_bytes),
                                (const char*)xbuf, a.hidden, G1, a.log2g,
                                lane, a.debug, tmo, 0xE3, g1, u1);
        g0 *= inv; u0 *= inv; g1 *= inv; u1 *= inv;
        const float h0 = (g0 / (1.f + __expf(-g0))) * u0;
        const float h1 = (g1 / (1.f + __expf(-g1))) * u1;
        if (lane == 0)
          store_granule(g_h + cu * (npc_gu / 4) + rq, e_h,
                        pack_bf16x2(h0, h1));
        lds_flag_store_rlx(&ctl->consumed[wave], (unsigned)(cur.off / SLOT));
      }
      lds_flag_store_rlx(&ctl->consumed[wave], (unsigned)(cur.off / SLOT));
    }
    ENG_DBG_BARRIER();
    ENG_PROF(6);

    // ---------------- P8: gather h ------------------------------------
    if (wave == 0) lds_flag_store_rlx(&ctl->busy, 1u);
    ENG_CHECK(gather_share(g_h, a.inter / 2, e_h, (u32*)unionb, wave, ctl,
                           tmo, 0x61, lane, &dbg_sweep, a.debug));
    ++gseq;
    ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x62, lane, &dbg_sync));
    ENG_DBG_FENCE();
    if (wave == 0) lds_flag_store_rlx(&ctl->busy, 0u);

    // ---------------- P9: down records (src = hbuf) --------------------
    {
      for (int rp = 0; rp < npc_dn / 2; ++rp) {
        const u64 atA = cursor_next(cur, rb_dn, ring_bytes);
        const u64 atB = cursor_next(cur, rb_dn, ring_bytes);
        if (rp % NCONS != wave) continue;
        ENG_CHECK(wait_fill(ctl, atB + rb_dn, tmo, 0x63, lane, &dbg_fill));
        float vA, vB;
        record_pair_dot_checked(ring + (u32)(atA % ring_bytes),
                                ring + (u32)(atB % ring_bytes), unionb,
                                a.inter, G2, a.log2g, lane, a.debug, tmo,
                                0xE4, vA, vB);
        if (lane == 0)
          store_granule(g_down + cu * (npc_dn / 2) + rp, e_down,
                        pack_bf16x2(vA, vB));
        lds_flag_store_rlx(&ctl->consumed[wave], (unsigned)(cur.off / SLOT));
      }
      lds_flag_store_rlx(&ctl->consumed[wave], (unsigned)(cur.off / SLOT));
    }
    ENG_DBG_BARRIER();
    ENG_PROF(7);

    // ---------------- P10: gather down, resid-add, inv1 of next layer --
    if (wave == 0) lds_flag_store_rlx(&ctl->busy, 1u);
    ENG_CHECK(gather_share(g_down, a.hidden / 2, e_down, (u32*)ascr, wave,
                           ctl, tmo, 0x64, lane, &dbg_sweep, a.debug));
    ++gseq;
    ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x65, lane, &dbg_sync));
    ENG_DBG_FENCE();
    {
      u32* x32 = (u32*)xbuf;
      const u32* add32 = (const u32*)ascr;
      const int nd = a.hidden / 2;
      const int b = nd * wave / NCONS, e = nd * (wave + 1) / NCONS;
      float ss = 0.f;
      for (int j = b + lane; j < e; j += LFMQ_WAVE) {
        float x0, x1, a0, a1;
        bf16x2_to_f32(x32[j], x0, x1);
        bf16x2_to_f32(add32[j], a0, a1);
        x0 += a0;
        x1 += a1;
        const u32 nv = pack_bf16x2(x0, x1);
        x32[j] = nv;
        ss = dot2_bf16(nv, nv, ss);
      }
      ss = wave_reduce_sum(ss);
      if (lane == 0) ctl->ss_part[wave] = ss;
      if (lane == 0)
      __hip_atomic_fetch_add(&ctl->gath, 1u, __ATOMIC_ACQ_REL, LFMQ_WG_SCOPE);
      ++gseq;
      ENG_CHECK(wait_gath(ctl, gseq, tmo, 0x66, lane, &dbg_sync));
      if (wave == 0) {
        lds_flag_store_rlx(&ctl->busy, 0u);
        float sst = 0.f;
        for (int w = 0; w < NCONS; ++w) sst += ctl->ss_part[w];
        ctl->inv_cur = rsqrtf(sst / (float)a.hidden + a.rms_eps);
        lds_flag_store(&ctl->phase, ++phase_want);
      } else {
        ++phase_want;
        ENG_CHECK(lds_wait_ge(&ctl->phase, phase_want, tmo, 0x67, lane));
      }
    }
  }

  if (a.prof && lane == 0) {
    __hip_atomic_fetch_add((gu32*)a.tmo + 1, dbg_fill, LFMQ_RLX_AGENT);
    __hip_atomic_fetch_add((gu32*)a.tmo + 3, dbg_sweep, LFMQ_RLX_AGENT);
    __hip_atomic_fetch_add((gu32*)a.tmo + 4, dbg_sync, LFMQ_RLX_AGENT);
  }

  // ---- epilogue: CU 0 wave 0 writes the final-norm output -------------
  if (cu == 0 && wave == 0) {
    const float inv = ctl->inv_cur;
    const u32* x32 = (const u32*)xbuf;
    const u32* w32 = (const u32*)a.final_w;
    u32* out32 = (u32*)a.xout;
    for (int j = lane; j < a.hidden / 2; j += LFMQ_WAVE) {
      float x0, x1, w0, w1;
      bf16x2_to_f32(x32[j], x0, x1);
      bf16x2_to_f32(w32[j], w0, w1);
      out32[j] = pack_bf16x2(x0 * inv * w0, x1 * inv * w1);
    }
  }
#undef ENG_CHECK
#undef ENG_PROF
#undef ENG_DBG_BARRIER
#undef ENG_DBG_FENCE
}

}  // namespace

// host-side launcher; returns required dynamic-LDS bytes via *lds_bytes
// when kernel launch is skipped (query mode: stream == nullptr).
extern "C" hipError_t lfmq_engine_step(const EngineArgs* args, int d,
                                       hipStream_t stream) {
  if (d != 128) return hipErrorInvalidValue;
  const u32 ring_bytes = (u32)args->nslot_ring * SLOT;
  const int union_bytes =
      std::max(args->inter * 2, std::max(args->nqkv * 2, args->hq * d * 2));
  const int ascr_bytes =
      std::max(NCONS * d * 4 + NCONS * AT * 4, args->hidden * 2);
  const size_t lds = ring_bytes + args->hidden * 2 + union_bytes +
                     ((ascr_bytes + 63) & ~63) + sizeof(Ctl) + 64;
  static size_t lds_set = 0;
  if (lds > lds_set) {
    hipError_t e = hipFuncSetAttribute(
        (const void*)lfmq_engine_kernel<128>,
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    if (e != hipSuccess) return e;
    lds_set = lds;
  }
  hipLaunchKernelGGL(lfmq_engine_kernel<128>, dim3(NCU),
                     dim3((NCONS + 1) * 64), lds, stream, *args);
  return hipGetLastError();
}
