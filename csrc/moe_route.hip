// MoE decode routing in ONE kernel (Mixtral top-k): replaces the aten
// chain topk/softmax/argsort/bincount/cumsum/gathers (~12 launches per
// MoE layer, ~3-5 us each — measured dominant in the 172 tok/s round-1
// Mixtral decode) with a single tiny launch.
//
// Input: router logits [t, E] bf16 (the dense gate GEMV output).
// Outputs, slot-major sorted by expert (the grouped kernel's layout):
//   src_tok  i32[t*k]  token index of each slot
//   slot_w   f32[t*k]  softmax weight of the slot (over the token's top-k)
//   offsets  i32[E+1]  per-expert slot ranges
// One workgroup; lane = token (t <= 64, decode shapes); within-expert
// slot order is by atomic cursor (order irrelevant: the combine is a
// weighted scatter-add).
#include "common.h"

namespace {

constexpr int MAXE = 64;
constexpr int MAXK = 4;

// active[j] = j-th expert with a nonempty slot range (dense, grid.y
// indexes it so dead experts never dispatch workgroups); active[j] = -1
// beyond n_active.
__global__ __launch_bounds__(64) void moe_route_kernel(
    const unsigned short* __restrict__ logits, i32* __restrict__ src_tok,
    float* __restrict__ slot_w, i32* __restrict__ offsets,
    i32* __restrict__ active, int t, int E, int topk) {
  __shared__ int cnt[MAXE];
  __shared__ int offs[MAXE + 1];
  const int lane = threadIdx.x;
  if (lane < E) cnt[lane] = 0;
  __syncthreads();

  float best[MAXK];
  int bid[MAXK];
  const bool live = lane < t;
  if (live) {
    // serial top-k over E logits (E <= 64, k <= 4)
#pragma unroll
    for (int j = 0; j < MAXK; ++j) {
      best[j] = -INFINITY;
      bid[j] = 0;
    }
    for (int e = 0; e < E; ++e) {
      float v = bf16_bits_to_f32(logits[lane * E + e]);
#pragma unroll
      for (int j = 0; j < MAXK; ++j) {
        if (j >= topk) break;
        if (v > best[j]) {
          float tv = best[j];
          int ti = bid[j];
          best[j] = v;
          bid[j] = e;
          v = tv;
          // shift the displaced entry down
          for (int l = j + 1; l < topk; ++l) {
            const float t2 = best[l];
            const int i2 = bid[l];
            best[l] = v;
            bid[l] = ti;
            v = t2;
            ti = i2;
          }
          break;
        }
      }
    }
    for (int j = 0; j < topk; ++j)
      atomicAdd(&cnt[bid[j]], 1);
  }
  __syncthreads();
  if (lane == 0) {
    int acc = 0;
    int na = 0;
    for (int e = 0; e <= E; ++e) {
      offs[e] = acc;
      if (e < E) {
        offsets[e] = acc;
        if (cnt[e] > 0) active[na++] = e;
        acc += cnt[e];
        cnt[e] = 0;  // becomes the write cursor
      } else {
        offsets[E] = acc;
      }
    }
    for (int j = na; j < E; ++j) active[j] = -1;
  }
  __syncthreads();
  if (live) {
    // softmax over the token's top-k
    float mx = best[0];
    float den = 0.f;
    for (int j = 0; j < topk; ++j) den += __expf(best[j] - mx);
    for (int j = 0; j < topk; ++j) {
      const int slot = offs[bid[j]] + atomicAdd(&cnt[bid[j]], 1);
      src_tok[slot] = lane;
      slot_w[slot] = __expf(best[j] - mx) / den;
    }
  }
}

}  // namespace

hipError_t lfmq_moe_route(const void* logits, void* src_tok, void* slot_w,
                          void* offsets, void* active, int t, int E, int topk,
                          hipStream_t stream) {
  if (t > 64 || E > MAXE || topk > MAXK) return hipErrorInvalidValue;
  hipLaunchKernelGGL(moe_route_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)logits, (i32*)src_tok,
                     (float*)slot_w, (i32*)offsets, (i32*)active, t, E, topk);
  return hipGetLastError();
}
