// Bandwidth probes for tuning (not part of the inference path).
// read_probe: streams a buffer exactly like the GEMV weight path —
// each wave owns STREAMS row-segments read 16 B/lane per iteration —
// to isolate what bounds the decode GEMV (nt vs cached, stream count).
#include "common.h"

namespace {

// MODE 0: bare reads; 1: + x loads (shared 8KB buffer, cached);
// 2: + the GEMV's actual dequant+dot compute per stream.
template <int NT, int STREAMS, int MODE>
__global__ __launch_bounds__(256) void read_probe_kernel(
    const u32* __restrict__ src, float* __restrict__ sink, u64 row_bytes,
    int rows_per_wave, int iters) {
  const int wave = threadIdx.x / LFMQ_WAVE;
  const int lane = threadIdx.x % LFMQ_WAVE;
  const u64 gwave = (u64)blockIdx.x * 4 + wave;
  __amdgpu_buffer_rsrc_t rs[STREAMS];
#pragma unroll
  for (int c = 0; c < STREAMS; ++c)
    rs[c] = make_rsrc_uniform(
        (const char*)src + (gwave * STREAMS + c) * row_bytes, (u32)row_bytes);
  auto x_rsrc = make_rsrc(src, 8192);  // L1-resident fake activations
  float acc = 0.f;
  for (int it = 0; it < iters; ++it) {
    const u32 byte = (u32)(it * 1024 + lane * 16);
    u32x4 v[STREAMS];
#pragma unroll
    for (int c = 0; c < STREAMS; ++c)
      v[c] = __builtin_amdgcn_raw_buffer_load_b128(rs[c], byte, 0, NT);
    if (MODE == 0) {
#pragma unroll
      for (int c = 0; c < STREAMS; ++c)
        acc += (float)(v[c][0] ^ v[c][1] ^ v[c][2] ^ v[c][3]);
    } else {
      const u32 xbyte = (u32)((lane * 64) & 8191);
      u32x4 xq0 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte, 0, 0);
      u32x4 xq1 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 16, 0, 0);
      u32x4 xq2 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 32, 0, 0);
      u32x4 xq3 = __builtin_amdgcn_raw_buffer_load_b128(x_rsrc, xbyte + 48, 0, 0);
      if (MODE == 1) {
        acc += (float)(xq0[0] ^ xq1[1] ^ xq2[2] ^ xq3[3]);
#pragma unroll
        for (int c = 0; c < STREAMS; ++c)
          acc += (float)(v[c][0] ^ v[c][1] ^ v[c][2] ^ v[c][3]);
      } else {
        float xs = 0.f;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
          xs = dot2_bf16(0x3F803F80u, xq[0], xs);
          xs = dot2_bf16(0x3F803F80u, xq[1], xs);
          xs = dot2_bf16(0x3F803F80u, xq[2], xs);
          xs = dot2_bf16(0x3F803F80u, xq[3], xs);
        }
#pragma unroll
        for (int c = 0; c < STREAMS; ++c) {
          float cx = 0.f;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const u32x4 xq = j == 0 ? xq0 : j == 1 ? xq1 : j == 2 ? xq2 : xq3;
            const DequantPairs4 dq = dequant4_pairs(v[c][j]);
            cx = dot2_bf16(dq.p01, xq[0], cx);
            cx = dot2_bf16(dq.p23, xq[1], cx);
            cx = dot2_bf16(dq.p45, xq[2], cx);
            cx = dot2_bf16(dq.p67, xq[3], cx);
          }
          acc = fmaf(1.1f, cx, acc);
          acc = fmaf(-0.9f, xs, acc);
        }
      }
    }
  }
  if (acc == 1234.5f) sink[0] = acc;  // never true; keeps loads live
}

}  // namespace

// streams a tensor: grid*4 waves * STREAMS rows of row_bytes each
hipError_t lfmq_read_probe(const void* src, void* sink, long row_bytes,
                           long n_rows, int nt, int streams, int mode,
                           hipStream_t stream) {
  const int iters = (int)(row_bytes / 1024);
  const long waves = n_rows / streams;
  dim3 grid((u32)(waves / 4)), block(256);
#define CASE(NT, ST, MO)                                                     \
  hipLaunchKernelGGL((read_probe_kernel<NT, ST, MO>), grid, block, 0,        \
                     stream, (const u32*)src, (float*)sink, (u64)row_bytes,  \
                     streams, iters)
  if (streams != 4) {
    if (nt && streams == 8) CASE(2, 8, 0);
    else if (!nt && streams == 8) CASE(0, 8, 0);
    else if (nt) CASE(2, 1, 0);
    else CASE(0, 1, 0);
  } else if (mode == 0) {
    if (nt) CASE(2, 4, 0); else CASE(0, 4, 0);
  } else if (mode == 1) {
    if (nt) CASE(2, 4, 1); else CASE(0, 4, 1);
  } else {
    if (nt) CASE(2, 4, 2); else CASE(0, 4, 2);
  }
#undef CASE
  return hipGetLastError();
}
