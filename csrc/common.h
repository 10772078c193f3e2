// Shared helpers for the lfmq gfx950 (CDNA4) HIP kernels.
// Written for MI355X only: wave64, MFMA, 160 KiB LDS, 8 XCDs.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define LFMQ_WAVE 64

typedef uint32_t u32;
typedef uint64_t u64;
typedef int32_t i32;
typedef __hip_bfloat16 bf16_t;

typedef __bf16 bf16x2_native __attribute__((ext_vector_type(2)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short s16x4 __attribute__((ext_vector_type(4)));
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef u32 u32x4 __attribute__((ext_vector_type(4)));

// f32 <-> bf16 bit helpers (bf16 = high 16 bits of f32; RNE convert on pack)
__device__ __forceinline__ float bf16_bits_to_f32(u32 hi16) {
  union { u32 u; float f; } v; v.u = hi16 << 16; return v.f;
}
__device__ __forceinline__ u32 f32_to_bf16_bits_rne(float x) {
  union { u32 u; float f; } v; v.f = x;
  u32 r = v.u + 0x7FFF + ((v.u >> 16) & 1);
  return r >> 16;
}
// pack two f32 into a bf16x2 dword (RNE)
__device__ __forceinline__ u32 pack_bf16x2(float lo, float hi) {
  return f32_to_bf16_bits_rne(lo) | (f32_to_bf16_bits_rne(hi) << 16);
}

// dword of 2 bf16 -> two f32
__device__ __forceinline__ void bf16x2_to_f32(u32 d, float& lo, float& hi) {
  lo = bf16_bits_to_f32(d & 0xFFFFu);
  hi = bf16_bits_to_f32(d >> 16);
}

// fdot2 on a packed bf16x2 pair: acc += a.lo*b.lo + a.hi*b.hi
__device__ __forceinline__ float dot2_bf16(u32 a, u32 b, float acc) {
  return __builtin_amdgcn_fdot2_f32_bf16(__builtin_bit_cast(bf16x2_native, a),
                                         __builtin_bit_cast(bf16x2_native, b),
                                         acc, false);
}

// ---- int4 -> bf16x2 dequant (the GEMV hot path) -------------------------
// A packed qweight dword w holds nibbles c0..c7 (little-endian; code for
// k = 8*word + i at bits 4i..4i+3).  Layout doc: lfmq/ops/ref.py.
// even/odd split puts c0,c2,c4,c6 / c1,c3,c5,c7 into bytes 0..3; v_perm
// (sel 0x0C = constant 0x00, verified by constant-fold on this toolchain)
// assembles bf16 pairs biased by 128: bf16(0x4300 | c) == 128 + c exactly.
// The caller removes the bias via the group zero term: (c - z)·s·x
//  == s·(Σ(128+c)x − (128+z)·Σx).
struct DequantPairs4 { u32 p01, p23, p45, p67; };  // bf16x2 pairs (128+c)
__device__ __forceinline__ DequantPairs4 dequant4_pairs(u32 w) {
  u32 even = w & 0x0F0F0F0Fu;
  u32 odd = (w >> 4) & 0x0F0F0F0Fu;
  DequantPairs4 r;
  r.p01 = __builtin_amdgcn_perm(odd, even, 0x0C040C00u) | 0x43004300u;
  r.p23 = __builtin_amdgcn_perm(odd, even, 0x0C050C01u) | 0x43004300u;
  r.p45 = __builtin_amdgcn_perm(odd, even, 0x0C060C02u) | 0x43004300u;
  r.p67 = __builtin_amdgcn_perm(odd, even, 0x0C070C03u) | 0x43004300u;
  return r;
}

// extract int4 zero code i from a packed zeros word
__device__ __forceinline__ u32 unpack4(u32 word, u32 idx) {
  return (word >> (idx * 4)) & 0xFu;
}
__device__ __forceinline__ u32 unpack8(u32 word, u32 idx) {
  return (word >> (idx * 8)) & 0xFFu;
}

// ---- wave reduction ------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// DPP-based wave sum: ~6 VALU ops instead of 6 LDS-unit bpermute round
// trips (~20 vs ~500 cycles — decisive at 1 wave/SIMD where nothing hides
// the bpermute latency).  rocPRIM's classic CDNA ladder: row_shr 1/2/4/8
// (bound_ctrl: shifted-in lanes read 0), then row_bcast:15 into rows 1|3,
// row_bcast:31 into rows 2|3; lane 63 holds the total, broadcast via
// readlane.  The update_dpp builtin is compiler-modeled (hazards padded).
__device__ __forceinline__ float wave_reduce_sum_dpp(float v) {
#define LFMQ_DPP_ADD(ctrl, rmask)                                         \
  v += __builtin_bit_cast(float,                                          \
      __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v), (ctrl),  \
                                  (rmask), 0xf, true))
  LFMQ_DPP_ADD(0x111, 0xf);  // row_shr:1
  LFMQ_DPP_ADD(0x112, 0xf);  // row_shr:2
  LFMQ_DPP_ADD(0x114, 0xf);  // row_shr:4
  LFMQ_DPP_ADD(0x118, 0xf);  // row_shr:8
  LFMQ_DPP_ADD(0x142, 0xa);  // row_bcast:15 -> rows 1,3
  LFMQ_DPP_ADD(0x143, 0xc);  // row_bcast:31 -> rows 2,3
#undef LFMQ_DPP_ADD
  return __builtin_bit_cast(
      float, __builtin_amdgcn_readlane(__builtin_bit_cast(int, v), 63));
}
__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// ---- buffer resource (T8): hardware-bounds-checked loads (OOB -> 0) ------
__device__ __forceinline__ __amdgpu_buffer_rsrc_t make_rsrc(const void* p, u32 bytes) {
  return __builtin_amdgcn_make_buffer_rsrc((void*)p, /*stride*/ 0, bytes,
                                           /*swizzle fmt*/ 0x00020000u);
}

// Wave-uniform variant (guide T20): a descriptor built from a value the
// compiler cannot PROVE uniform (e.g. a per-wave row pointer) gets every
// buffer op wrapped in a ~10-instruction readfirstlane/saveexec waterfall
// loop that serializes consecutive loads — measured 13 µs/call on the
// decode GEMV.  readfirstlane the inputs once; descriptor becomes SGPRs.
__device__ __forceinline__ __amdgpu_buffer_rsrc_t make_rsrc_uniform(
    const void* p, u32 bytes) {
  const u64 a = (u64)p;
  const u32 lo = __builtin_amdgcn_readfirstlane((u32)a);
  const u32 hi = __builtin_amdgcn_readfirstlane((u32)(a >> 32));
  const u32 nb = __builtin_amdgcn_readfirstlane(bytes);
  return __builtin_amdgcn_make_buffer_rsrc((void*)(((u64)hi << 32) | lo), 0,
                                           nb, 0x00020000u);
}

// XCD-aware bijective block-id swizzle (guide §5: nwg need not divide 8)
__device__ __forceinline__ u32 xcd_swizzle(u32 bid, u32 nwg) {
  const u32 nx = 8;
  u32 q = nwg / nx, r = nwg % nx;
  u32 xcd = bid % nx, i = bid / nx;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
}

#define LFMQ_CHECK_HIP(expr)                                        \
  do {                                                              \
    hipError_t _e = (expr);                                         \
    if (_e != hipSuccess) return _e;                                \
  } while (0)

static inline u32 ceil_div_u32(u32 a, u32 b) { return (a + b - 1) / b; }
