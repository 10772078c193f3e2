// Standalone validation of the persistent-decode-engine primitives on a
// real MI355X (NOT part of the torch extension; no model code):
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 csrc/engine_probe.hip \
//         -o gpurun_out/engine_probe && ./gpurun_out/engine_probe
//
// Tests, in dependency order (each later test assumes the earlier holds):
//  1. census: all 256 one-per-CU workgroups are co-resident (a persistent
//     engine deadlocks otherwise — MI355X_MICROARCH.md residency rules).
//  2. ring: loader-wave LDS-DMA weight streaming (7 x 16 KiB ring,
//     global_load_lds nt, vmcnt-lagged slot publication) + 3 consumer
//     waves checksumming every byte through LDS; flow control via
//     workgroup-scope LDS atomics.  Verifies exact u64 sums per WG and
//     reports the achieved chip fill bandwidth.
//  3. allgather: R2 granule publish + sweep (Guideline 16) across all
//     256 CUs, ping-pong buffers over many epochs (the engine's edge
//     reuse discipline), exact sum verification per CU per epoch.
#include <hip/hip_runtime.h>
#include <stdio.h>
#include <stdlib.h>
#include <vector>

#include "common.h"
#include "engine_common.h"

using namespace lfmq_engine;

constexpr int NWG = 256;
constexpr int TPB = 256;
constexpr int SLOT = 16384;
constexpr int NSLOT = 7;
constexpr int LAG = 4;  // publish slot s-LAG+1 after issuing slot s

__device__ __forceinline__ unsigned long long wave_reduce_sum_u64(
    unsigned long long v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// ---------------------------------------------------------------- census
__global__ __launch_bounds__(TPB) void census_kernel(gu32* cnt, u32* out) {
  if (threadIdx.x == 0) __hip_atomic_fetch_add(cnt, 1u, LFMQ_RLX_AGENT);
  unsigned spins = 0;
  bool ok = true;
  while (__hip_atomic_load(cnt, LFMQ_RLX_AGENT) < NWG) {
    if (spin_fail(spins, nullptr, 0, 0)) { ok = false; break; }
  }
  if (threadIdx.x == 0) out[blockIdx.x] = ok ? 1u : 0u;
}

// ------------------------------------------------------------------ ring
// LDS: ring + control words.  Loader = wave 3, consumers = waves 0-2.
__global__ __launch_bounds__(TPB) void ring_kernel(
    const char* __restrict__ src, unsigned long long* __restrict__ out,
    gu32* tmo, int slots_per_wg) {
  __shared__ char ring[NSLOT * SLOT];
  __shared__ unsigned filled;      // published slot count (monotonic)
  __shared__ unsigned consumed[3];  // per consumer wave: slots consumed
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  if (threadIdx.x == 0) {
    filled = 0;
    consumed[0] = consumed[1] = consumed[2] = 0;
  }
  __syncthreads();  // safe: all waves still converged here

  const char* wg_src = src + (u64)blockIdx.x * slots_per_wg * SLOT;

  if (wave == 3) {  // ---- loader ----
    const unsigned f_off = lds_off(&filled);
    const unsigned c_off0 = lds_off(&consumed[0]);
    const unsigned c_off1 = lds_off(&consumed[1]);
    const unsigned c_off2 = lds_off(&consumed[2]);
    for (int s = 0; s < slots_per_wg; ++s) {
      if (s >= NSLOT) {
        // ring space: slot s overwrites ring slot s%NSLOT, last used by
        // stream slot s-NSLOT: every consumer must be past it.  The poll
        // is asm (loader_read3) so hipcc adds no vmcnt drain while fills
        // are in flight.
        unsigned spins = 0;
        for (;;) {
          unsigned d0, d1, d2;
          loader_read3(c_off0, c_off1, c_off2, d0, d1, d2);
          if ((int)min(d0, min(d1, d2)) >= s - NSLOT + 1) break;
          if (spin_fail(spins, tmo, 10, lane)) return;
        }
      }
      char* lbase = ring + (s % NSLOT) * SLOT;
      const char* g = wg_src + (u64)s * SLOT + lane * 16;
#pragma unroll
      for (int i = 0; i < 16; ++i) glds_fill_1k(g + i * 1024, lbase + i * 1024);
      if (s >= LAG - 1)  // publish slot s-LAG+1: <= 48 outstanding = landed
        loader_publish<(LAG - 1) * 16>(f_off, (unsigned)(s - LAG + 2));
    }
    loader_publish<0>(f_off, (unsigned)slots_per_wg);
    return;
  }

  // ---- consumers: each wave sums a lane-strided third of every slot ----
  unsigned long long acc = 0;
  for (int s = 0; s < slots_per_wg; ++s) {
    if (!lds_wait_ge(&filled, (unsigned)(s + 1), tmo, 20, lane)) return;
    const u32* slot32 = (const u32*)(ring + (s % NSLOT) * SLOT);
    for (int i = wave * LFMQ_WAVE + lane; i < SLOT / 4; i += 3 * LFMQ_WAVE)
      acc += slot32[i];
    lds_flag_store(&consumed[wave], (unsigned)(s + 1));
  }
  acc = wave_reduce_sum_u64(acc);
  if (lane == 0) out[blockIdx.x * 3 + wave] = acc;
}

// -------------------------------------------------------------- allgather
constexpr int PGRAN = 8;  // granules published per WG per epoch
__global__ __launch_bounds__(TPB) void allgather_kernel(
    gu64* gran0, gu64* gran1, unsigned long long* __restrict__ out, gu32* tmo,
    int epochs) {
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  if (wave != 0) return;  // one active wave per CU mirrors the gather wave
  __shared__ u32 stage[NWG * PGRAN];
  for (int e = 1; e <= epochs; ++e) {
    gu64* g = (e & 1) ? gran0 : gran1;  // ping-pong: reuse period 2 epochs
    if (lane < PGRAN)
      store_granule(g + blockIdx.x * PGRAN + lane, (unsigned)e,
                    blockIdx.x * PGRAN + lane + (unsigned)e * 1000003u);
    if (!sweep_stage(g, NWG * PGRAN, (unsigned)e, stage, tmo, 100u + e, lane))
      return;
    unsigned long long s = 0;
    for (int i = lane; i < NWG * PGRAN; i += LFMQ_WAVE) s += stage[i];
    s = wave_reduce_sum_u64(s);
    if (lane == 0) out[(u64)(e - 1) * NWG + blockIdx.x] = s;
  }
}

// ------------------------------------------------------------------ host
#define CHK(x)                                                    \
  do {                                                            \
    hipError_t e = (x);                                           \
    if (e != hipSuccess) {                                        \
      fprintf(stderr, "HIP error %s at %s:%d\n",                  \
              hipGetErrorString(e), __FILE__, __LINE__);          \
      exit(2);                                                    \
    }                                                             \
  } while (0)

int main() {
  int fails = 0;

  // ---- census ----
  {
    gu32* cnt;
    u32* out;
    CHK(hipMalloc((void**)&cnt, 4));
    CHK(hipMalloc((void**)&out, NWG * 4));
    CHK(hipMemset((void*)cnt, 0, 4));
    hipLaunchKernelGGL(census_kernel, dim3(NWG), dim3(TPB), 0, 0, cnt, out);
    CHK(hipDeviceSynchronize());
    std::vector<u32> h(NWG);
    CHK(hipMemcpy(h.data(), out, NWG * 4, hipMemcpyDeviceToHost));
    int ok = 0;
    for (auto v : h) ok += v == 1;
    printf("census: %d/%d workgroups co-resident\n", ok, NWG);
    if (ok != NWG) { fails++; printf("CENSUS FAIL — aborting\n"); return 1; }
    hipFree((void*)cnt); hipFree(out);
  }

  // ---- ring ----
  {
    const int slots_per_wg = 512;  // 8 MiB per WG -> 2 GiB total
    const u64 total = (u64)NWG * slots_per_wg * SLOT;
    char* src;
    unsigned long long* out;
    gu32* tmo;
    CHK(hipMalloc(&src, total));
    CHK(hipMalloc((void**)&out, NWG * 3 * 8));
    CHK(hipMalloc((void**)&tmo, 4));
    CHK(hipMemset((void*)tmo, 0, 4));
    // deterministic pattern, cheap to check per WG on host
    {
      std::vector<u32> host(total / 4);
      for (u64 i = 0; i < total / 4; ++i) host[i] = (u32)(i * 2654435761u);
      CHK(hipMemcpy(src, host.data(), total, hipMemcpyHostToDevice));
    }
    // warm + timed
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    for (int rep = 0; rep < 2; ++rep) {
      CHK(hipEventRecord(t0));
      hipLaunchKernelGGL(ring_kernel, dim3(NWG), dim3(TPB), 0, 0, src, out,
                         tmo, slots_per_wg);
      CHK(hipEventRecord(t1));
      CHK(hipDeviceSynchronize());
      float ms;
      CHK(hipEventElapsedTime(&ms, t0, t1));
      u32 tmo_h;
      CHK(hipMemcpy(&tmo_h, (void*)tmo, 4, hipMemcpyDeviceToHost));
      printf("ring rep %d: %.3f ms -> %.2f GB/s  (timeout code %u)\n", rep,
             ms, total / (ms * 1e6), tmo_h);
      if (tmo_h) fails++;
    }
    std::vector<unsigned long long> h(NWG * 3);
    CHK(hipMemcpy(h.data(), out, NWG * 3 * 8, hipMemcpyDeviceToHost));
    // host verify: sum of u32s per WG split by consumer-wave pattern
    {
      std::vector<u32> host(total / 4);
      for (u64 i = 0; i < total / 4; ++i) host[i] = (u32)(i * 2654435761u);
      int bad = 0;
      for (int wg = 0; wg < NWG; ++wg) {
        unsigned long long want[3] = {0, 0, 0};
        const u64 base = (u64)wg * slots_per_wg * (SLOT / 4);
        for (int s = 0; s < slots_per_wg; ++s)
          for (int i = 0; i < SLOT / 4; ++i)
            want[(i / 64) % 3] += host[base + (u64)s * (SLOT / 4) + i];
        for (int w = 0; w < 3; ++w)
          if (h[wg * 3 + w] != want[w]) bad++;
      }
      printf("ring checksum: %s (%d bad)\n", bad ? "FAIL" : "ok", bad);
      if (bad) fails++;
    }
    hipFree(src); hipFree(out); hipFree((void*)tmo);
  }

  // ---- allgather ----
  {
    const int epochs = 200;
    gu64 *g0, *g1;
    unsigned long long* out;
    gu32* tmo;
    CHK(hipMalloc((void**)&g0, NWG * PGRAN * 8));
    CHK(hipMalloc((void**)&g1, NWG * PGRAN * 8));
    CHK(hipMalloc((void**)&out, (u64)epochs * NWG * 8));
    CHK(hipMalloc((void**)&tmo, 4));
    CHK(hipMemset((void*)g0, 0, NWG * PGRAN * 8));
    CHK(hipMemset((void*)g1, 0, NWG * PGRAN * 8));
    CHK(hipMemset((void*)tmo, 0, 4));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    CHK(hipEventRecord(t0));
    hipLaunchKernelGGL(allgather_kernel, dim3(NWG), dim3(TPB), 0, 0, g0, g1,
                       out, tmo, epochs);
    CHK(hipEventRecord(t1));
    CHK(hipDeviceSynchronize());
    float ms;
    CHK(hipEventElapsedTime(&ms, t0, t1));
    u32 tmo_h;
    CHK(hipMemcpy(&tmo_h, (void*)tmo, 4, hipMemcpyDeviceToHost));
    std::vector<unsigned long long> h((u64)epochs * NWG);
    CHK(hipMemcpy(h.data(), out, (u64)epochs * NWG * 8, hipMemcpyDeviceToHost));
    int bad = 0;
    const int N = NWG * PGRAN;
    for (int e = 1; e <= epochs; ++e) {
      unsigned long long want = 0;
      for (int i = 0; i < N; ++i) want += (u32)(i + (unsigned)e * 1000003u);
      for (int wg = 0; wg < NWG; ++wg)
        if (h[(u64)(e - 1) * NWG + wg] != want) bad++;
    }
    printf("allgather: %d epochs, %.2f us/epoch, %s (%d bad, timeout %u)\n",
           epochs, ms * 1000.f / epochs, bad ? "FAIL" : "ok", bad, tmo_h);
    if (bad || tmo_h) fails++;
    hipFree((void*)g0); hipFree((void*)g1); hipFree(out); hipFree((void*)tmo);
  }

  printf(fails ? "ENGINE_PROBE FAIL (%d)\n" : "ENGINE_PROBE PASS\n", fails);
  return fails ? 1 : 0;
}
