// Bandwidth probes for tuning (not part of the inference path).
// read_probe: streams a buffer exactly like the GEMV weight path —
// each wave owns STREAMS row-segments read 16 B/lane per iteration —
// to isolate what bounds the decode GEMV (nt vs cached, stream count).
#include "common.h"

namespace {

template <int NT, int STREAMS>
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
  float acc = 0.f;
  for (int it = 0; it < iters; ++it) {
    const u32 byte = (u32)(it * 1024 + lane * 16);
    u32x4 v[STREAMS];
#pragma unroll
    for (int c = 0; c < STREAMS; ++c)
      v[c] = __builtin_amdgcn_raw_buffer_load_b128(rs[c], byte, 0, NT);
#pragma unroll
    for (int c = 0; c < STREAMS; ++c)
      acc += (float)(v[c][0] ^ v[c][1] ^ v[c][2] ^ v[c][3]);
  }
  if (acc == 1234.5f) sink[0] = acc;  // never true; keeps loads live
}

}  // namespace

// streams a tensor: grid*4 waves * STREAMS rows of row_bytes each
hipError_t lfmq_read_probe(const void* src, void* sink, long row_bytes,
                           long n_rows, int nt, int streams,
                           hipStream_t stream) {
  const int iters = (int)(row_bytes / 1024);
  const long waves = n_rows / streams;
  dim3 grid((u32)(waves / 4)), block(256);
#define CASE(NT, ST)                                                        \
  hipLaunchKernelGGL((read_probe_kernel<NT, ST>), grid, block, 0, stream,   \
                     (const u32*)src, (float*)sink, (u64)row_bytes,         \
                     streams, iters)
  if (nt && streams == 4) CASE(2, 4);
  else if (!nt && streams == 4) CASE(0, 4);
  else if (nt && streams == 8) CASE(2, 8);
  else if (!nt && streams == 8) CASE(0, 8);
  else if (nt) CASE(2, 1);
  else CASE(0, 1);
#undef CASE
  return hipGetLastError();
}
