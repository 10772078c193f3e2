// K1 — GPTQ Hessian accumulate: H += 2·XᵀX (X bf16 [t, k], H fp32 [k, k]).
//
// MFMA rank-k update: both operands of the (I,J)
// output tile are column-slices of the SAME X, staged into LDS
// TRANSPOSED ([col][t] rows, +16-ushort pad for conflict-free
// ds_read_b128 fragment reads) so A- and B-fragments are contiguous.
// Full matrix is computed (no triangle+mirror pass: the mirror would
// re-stream the 822 MB fp32 H of a 14336-wide layer for no win —
// MFMA flops are free at calibration scale).  One workgroup per 128x128
// tile; read-modify-write accumulate (one WG owns each tile).
#include "common.h"

namespace {

constexpr int BC = 128;   // columns per tile side
constexpr int BT = 64;    // token rows per staging step
constexpr int TPB = 512;
constexpr int LDT = BT + 16;  // padded [col][t] row length (ushorts)
typedef s16x8 frag_ab;

__global__ __launch_bounds__(TPB, 1) void hessian_kernel(
    const u32* __restrict__ x, float* __restrict__ h, int t, int k) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned short* ldsI = (unsigned short*)smem;             // [BC][LDT]
  unsigned short* ldsJ = ldsI + BC * LDT;

  const u32 nbx = (k + BC - 1) / BC;
  const int cI = (blockIdx.x / nbx) * BC;
  const int cJ = (blockIdx.x % nbx) * BC;

  const int tid = threadIdx.x;
  const int lane = tid % LFMQ_WAVE;
  const int wave = tid / LFMQ_WAVE;
  const int wm = wave / 4, wn = wave % 4;

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const unsigned short* xs = (const unsigned short*)x;
  const int tsteps = (t + BT - 1) / BT;
  for (int ts = 0; ts < tsteps; ++ts) {
    // stage transposed: thread -> (t_row, 8-col slot); 64 x 16 slots,
    // each thread handles 2 slots of 8 ushort scattered writes.
    for (int sl = tid; sl < BT * 16; sl += TPB) {
      const int tr = sl % BT;          // lanes 0..31 span t -> ~2-way banks
      const int c8 = (sl / BT) * 8;
      const int tg = ts * BT + tr;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        unsigned short* lds = half ? ldsJ : ldsI;
        const int cbase = half ? cJ : cI;
        u32x4 v = {0, 0, 0, 0};
        if (tg < t && cbase + c8 < k) {
          const unsigned short* src = xs + (u64)tg * k + cbase + c8;
          v = *(const u32x4*)src;  // 8 cols (16B); k%8==0 required
        }
        const unsigned short* vs = (const unsigned short*)&v;
#pragma unroll
        for (int i = 0; i < 8; ++i) lds[(c8 + i) * LDT + tr] = vs[i];
      }
    }
    __syncthreads();

    // MFMA over this 64-token slab: A rows = cI cols, B cols = cJ cols
#pragma unroll
    for (int ks = 0; ks < BT / 32; ++ks) {
      const int t8 = ks * 32 + (lane >> 4) * 8;
      frag_ab a[4], b[2];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wm * 64 + i * 16 + (lane & 15);
        a[i] = *(frag_ab*)(ldsI + row * LDT + t8);
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int col = wn * 32 + j * 16 + (lane & 15);
        b[j] = *(frag_ab*)(ldsJ + col * LDT + t8);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: H[cI+row][cJ+col] += 2*acc
  const int ccol = cJ + wn * 32 + (lane & 15);
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int col = ccol + j * 16;
      if (col >= k) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = cI + wm * 64 + i * 16 + (lane >> 4) * 4 + r;
        if (row < k) {
          float* p = h + (u64)row * k + col;
          *p = fmaf(2.f, acc[i][j][r], *p);
        }
      }
    }
}

}  // namespace

hipError_t lfmq_hessian_accum(const void* x, void* h, int t, int k,
                              hipStream_t stream) {
  if (k % 8) return hipErrorInvalidValue;
  const u32 nb = ceil_div_u32(k, BC);
  dim3 grid(nb * nb), block(TPB);
  const size_t lds = 2 * BC * (BT + 16) * sizeof(unsigned short);
  hipLaunchKernelGGL(hessian_kernel, grid, block, lds, stream, (const u32*)x,
                     (float*)h, t, k);
  return hipGetLastError();
}
