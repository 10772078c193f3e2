// Greedy sampling: row-wise argmax over the logits [b, vocab] bf16.
// torch's last-dim argmax on a [1, 128k] row measured 125 µs (one-row
// reduce has no parallelism there); this two-phase version splits the
// row over 64 workgroups, then one tiny combine. Ties break to the
// lowest index (torch semantics).
#include "common.h"

namespace {

constexpr int TPB = 256;
constexpr int NSPLIT = 64;

__global__ __launch_bounds__(TPB) void argmax_partial(
    const u32* __restrict__ logits, float* __restrict__ pmax,
    i32* __restrict__ pidx, int vocab) {
  const int row = blockIdx.x;
  const int part = blockIdx.y;
  const int vw = (vocab + 1) / 2;  // dwords per row
  const int per = (vw + NSPLIT - 1) / NSPLIT;
  const int begin = part * per;
  const int end = min(begin + per, vw);
  const u32* src = logits + (u64)row * vw;

  float best = -INFINITY;
  int bidx = 0;
  for (int i = begin + threadIdx.x; i < end; i += TPB) {
    float lo, hi;
    bf16x2_to_f32(src[i], lo, hi);
    const int e0 = 2 * i, e1 = 2 * i + 1;
    if (lo > best || (lo == best && e0 < bidx)) { best = lo; bidx = e0; }
    if (e1 < vocab && (hi > best || (hi == best && e1 < bidx))) { best = hi; bidx = e1; }
  }
  // wave then block reduce (max, tie -> lower index)
  __shared__ float smax[TPB / LFMQ_WAVE];
  __shared__ int sidx[TPB / LFMQ_WAVE];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ob = __shfl_xor(best, off, 64);
    const int oi = __shfl_xor(bidx, off, 64);
    if (ob > best || (ob == best && oi < bidx)) { best = ob; bidx = oi; }
  }
  const int wave = threadIdx.x / LFMQ_WAVE;
  if (threadIdx.x % LFMQ_WAVE == 0) { smax[wave] = best; sidx[wave] = bidx; }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int w = 1; w < TPB / LFMQ_WAVE; ++w)
      if (smax[w] > best || (smax[w] == best && sidx[w] < bidx)) {
        best = smax[w]; bidx = sidx[w];
      }
    pmax[(u64)row * NSPLIT + part] = best;
    pidx[(u64)row * NSPLIT + part] = bidx;
  }
}

__global__ __launch_bounds__(64) void argmax_combine(
    const float* __restrict__ pmax, const i32* __restrict__ pidx,
    long* __restrict__ out) {
  const int row = blockIdx.x;
  float best = pmax[(u64)row * NSPLIT + threadIdx.x];
  int bidx = pidx[(u64)row * NSPLIT + threadIdx.x];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ob = __shfl_xor(best, off, 64);
    const int oi = __shfl_xor(bidx, off, 64);
    if (ob > best || (ob == best && oi < bidx)) { best = ob; bidx = oi; }
  }
  if (threadIdx.x == 0) out[row] = (long)bidx;
}

}  // namespace

// scratch: b*64 floats + b*64 ints
hipError_t lfmq_argmax_rows(const void* logits, void* out, void* scratch_f,
                            void* scratch_i, int b, int vocab,
                            hipStream_t stream) {
  if (vocab % 2) return hipErrorInvalidValue;
  hipLaunchKernelGGL(argmax_partial, dim3(b, NSPLIT), dim3(TPB), 0, stream,
                     (const u32*)logits, (float*)scratch_f, (i32*)scratch_i,
                     vocab);
  hipLaunchKernelGGL(argmax_combine, dim3(b), dim3(64), 0, stream,
                     (const float*)scratch_f, (const i32*)scratch_i,
                     (long*)out);
  return hipGetLastError();
}

namespace {

// hipGraph decode-step bookkeeping: record the sampled token into the
// output ring, feed it back as the next step's input, and advance the
// device-side decode state {pos0, T_after, step} — all inside the graph,
// so replay needs no host work at all.
__global__ __launch_bounds__(64) void advance_record_kernel(
    i32* __restrict__ state, long* __restrict__ tok_buf,
    const long* __restrict__ sampled, long* __restrict__ ring, int b,
    int ring_len) {
  if (threadIdx.x == 0) {
    const int step = state[2];
    for (int bb = 0; bb < b; ++bb) {
      // record the token this step CONSUMED (t_r), then feed the newly
      // sampled t_{r+1} back as the next replay's input
      if (step < ring_len) ring[(u64)bb * ring_len + step] = tok_buf[bb];
      tok_buf[bb] = sampled[bb];
    }
    state[0] += 1;
    state[1] += 1;
    state[2] += 1;
  }
}

}  // namespace

hipError_t lfmq_advance_record(void* state, void* tok_buf, const void* sampled,
                               void* ring, int b, int ring_len,
                               hipStream_t stream) {
  hipLaunchKernelGGL(advance_record_kernel, dim3(1), dim3(64), 0, stream,
                     (i32*)state, (long*)tok_buf, (const long*)sampled,
                     (long*)ring, b, ring_len);
  return hipGetLastError();
}
