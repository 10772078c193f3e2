// lfmq_hip — torch-extension bindings for the gfx950 kernels.
// Every entry validates dtype/contiguity/device and launches on the
// current HIP stream.  No fallbacks: errors raise.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

hipError_t lfmq_pack(const void*, void*, long, int, int, hipStream_t);
hipError_t lfmq_unpack(const void*, void*, long, int, int, hipStream_t);
hipError_t lfmq_dq_gemv(const void*, const void*, const void*, const void*,
                        void*, const void*, const void*, int, int, int, int,
                        int, float, int, hipStream_t);
hipError_t lfmq_dq_gemm(const void*, const void*, const void*, const void*,
                        void*, const void*, int, int, int, int, int,
                        hipStream_t);
hipError_t lfmq_rmsnorm(const void*, const void*, const void*, void*, void*,
                        int, int, float, hipStream_t);
hipError_t lfmq_rope(void*, void*, const void*, const void*, int, int, int,
                     int, int, hipStream_t);
hipError_t lfmq_silu_mul(const void*, const void*, void*, long, hipStream_t);
hipError_t lfmq_silu_mul_fused(const void*, void*, long, int, hipStream_t);
hipError_t lfmq_residual_add(const void*, const void*, void*, long,
                             hipStream_t);
hipError_t lfmq_kv_append(void*, void*, const void*, const void*, int, int,
                          int, int, int, int, hipStream_t);
hipError_t lfmq_attn_decode(const void*, const void*, const void*, void*,
                            void*, int, int, int, int, int, const void*, int,
                            float, hipStream_t);
hipError_t lfmq_attn_decode_fused(const void*, void*, void*, void*, void*,
                                  const void*, int, int, int, int, int,
                                  const void*, int, float, void*,
                                  hipStream_t);
hipError_t lfmq_attn_prefill(const void*, const void*, const void*, void*,
                             int, int, int, int, int, int, int, float,
                             hipStream_t);
hipError_t lfmq_hessian_accum(const void*, void*, int, int, hipStream_t);
hipError_t lfmq_dq_gemv_gateup(const void*, const void*, const void*,
                               const void*, void*, int, int, int, int, int,
                               float, int, hipStream_t);
hipError_t lfmq_qkv_rope_append(const void*, void*, void*, void*, const void*,
                                const void*, int, int, int, int, int, int,
                                int, const void*, hipStream_t);
hipError_t lfmq_advance_record(void*, void*, const void*, void*, int, int,
                               hipStream_t);
hipError_t lfmq_grouped_dq_gemv(const void*, const void*, const void*,
                                const void*, void*, const void*, const void*,
                                const void*, int, int, int, int, int, int,
                                int, bool, hipStream_t);
hipError_t lfmq_grouped_dq_gemv_scatter(const void*, const void*, const void*,
                                        const void*, void*, const void*,
                                        const void*, const void*, const void*,
                                        int, int, int, int, int, int,
                                        hipStream_t);
hipError_t lfmq_moe_route(const void*, void*, void*, void*, void*, int, int,
                          int, hipStream_t);
hipError_t lfmq_argmax_rows(const void*, void*, void*, void*, int, int,
                            hipStream_t);
hipError_t lfmq_read_probe(const void*, void*, long, long, int, int, int,
                           hipStream_t);
#include "engine_args.h"

namespace {

#define CHECK_HIP(call)                                                     \
  do {                                                                      \
    hipError_t e = (call);                                                  \
    TORCH_CHECK(e == hipSuccess, "lfmq HIP error: ", hipGetErrorString(e)); \
  } while (0)

void check_t(const torch::Tensor& t, torch::ScalarType dt, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == dt, name, " has wrong dtype ",
              t.scalar_type(), " expected ", dt);
}

hipStream_t stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

torch::Tensor pack(torch::Tensor codes, long bits) {
  check_t(codes, torch::kInt32, "codes");
  const long n = codes.size(0), k = codes.size(1);
  const long per = 32 / bits;
  TORCH_CHECK(k % per == 0, "k must be divisible by ", per);
  auto out = torch::empty({n, k / per}, codes.options());
  CHECK_HIP(lfmq_pack(codes.data_ptr(), out.data_ptr(), n, (int)k, (int)bits,
                      stream()));
  return out;
}

torch::Tensor unpack(torch::Tensor packed, long bits, long k) {
  check_t(packed, torch::kInt32, "packed");
  const long n = packed.size(0);
  auto out = torch::empty({n, k}, packed.options());
  CHECK_HIP(lfmq_unpack(packed.data_ptr(), out.data_ptr(), n, (int)k,
                        (int)bits, stream()));
  return out;
}

torch::Tensor dq_linear(torch::Tensor x, torch::Tensor qweight,
                        torch::Tensor scales, torch::Tensor qzeros, long bits,
                        long g, c10::optional<torch::Tensor> bias,
                        c10::optional<torch::Tensor> resid, double rms_eps,
                        long cols_hint) {
  check_t(x, torch::kBFloat16, "x");
  check_t(qweight, torch::kInt32, "qweight");
  check_t(scales, torch::kBFloat16, "scales");
  check_t(qzeros, torch::kInt32, "qzeros");
  TORCH_CHECK(x.dim() == 2, "x must be [m, k]");
  const int m = (int)x.size(0), k = (int)x.size(1);
  const int n = (int)qweight.size(0);
  TORCH_CHECK(qweight.size(1) == (long)k * bits / 32, "qweight/k mismatch");
  const void* bptr = nullptr;
  if (bias.has_value()) {
    check_t(*bias, torch::kBFloat16, "bias");
    bptr = bias->data_ptr();
  }
  const void* rptr = nullptr;
  if (resid.has_value()) {
    check_t(*resid, torch::kBFloat16, "resid");
    rptr = resid->data_ptr();
  }
  auto y = torch::empty({m, n}, x.options());
  if (m <= 16) {
    CHECK_HIP(lfmq_dq_gemv(qweight.data_ptr(), scales.data_ptr(),
                           qzeros.data_ptr(), x.data_ptr(), y.data_ptr(), bptr,
                           rptr, m, n, k, (int)g, (int)bits, (float)rms_eps,
                           (int)cols_hint, stream()));
  } else {
    TORCH_CHECK(rptr == nullptr && rms_eps < 0,
                "fused rms/resid is a decode (m<=16) path");
    CHECK_HIP(lfmq_dq_gemm(qweight.data_ptr(), scales.data_ptr(),
                           qzeros.data_ptr(), x.data_ptr(), y.data_ptr(), bptr,
                           m, n, k, (int)g, (int)bits, stream()));
  }
  return y;
}

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor w, double eps,
                      c10::optional<torch::Tensor> residual) {
  check_t(x, torch::kBFloat16, "x");
  check_t(w, torch::kBFloat16, "w");
  const int hidden = (int)x.size(-1);
  const int rows = (int)(x.numel() / hidden);
  auto y = torch::empty_like(x);
  const void* rptr = nullptr;
  if (residual.has_value()) {
    check_t(*residual, torch::kBFloat16, "residual");
    rptr = residual->data_ptr();
  }
  CHECK_HIP(lfmq_rmsnorm(x.data_ptr(), rptr, w.data_ptr(), y.data_ptr(),
                         nullptr, rows, hidden, (float)eps, stream()));
  return y;
}

std::vector<torch::Tensor> rmsnorm_residual(torch::Tensor x,
                                            torch::Tensor residual,
                                            torch::Tensor w, double eps) {
  check_t(x, torch::kBFloat16, "x");
  check_t(residual, torch::kBFloat16, "residual");
  check_t(w, torch::kBFloat16, "w");
  const int hidden = (int)x.size(-1);
  const int rows = (int)(x.numel() / hidden);
  auto y = torch::empty_like(x);
  auto res_out = torch::empty_like(x);
  CHECK_HIP(lfmq_rmsnorm(x.data_ptr(), residual.data_ptr(), w.data_ptr(),
                         y.data_ptr(), res_out.data_ptr(), rows, hidden,
                         (float)eps, stream()));
  return {y, res_out};
}

void rope(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
          torch::Tensor inv_freq) {
  check_t(q, torch::kBFloat16, "q");
  check_t(k, torch::kBFloat16, "k");
  check_t(positions, torch::kInt32, "positions");
  check_t(inv_freq, torch::kFloat32, "inv_freq");
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4, "q/k must be [b, s, h, d]");
  const int b = (int)q.size(0), s = (int)q.size(1);
  const int hq = (int)q.size(2), d = (int)q.size(3);
  const int hkv = (int)k.size(2);
  TORCH_CHECK(positions.numel() == s, "positions must have s entries");
  CHECK_HIP(lfmq_rope(q.data_ptr(), k.data_ptr(), positions.data_ptr(),
                      inv_freq.data_ptr(), b * s, s, hq, hkv, d, stream()));
}

torch::Tensor silu_mul(torch::Tensor gate, torch::Tensor up) {
  check_t(gate, torch::kBFloat16, "gate");
  check_t(up, torch::kBFloat16, "up");
  auto y = torch::empty_like(gate);
  CHECK_HIP(lfmq_silu_mul(gate.data_ptr(), up.data_ptr(), y.data_ptr(),
                          gate.numel(), stream()));
  return y;
}

torch::Tensor silu_mul_fused(torch::Tensor gate_up) {
  check_t(gate_up, torch::kBFloat16, "gate_up");
  const long inter = gate_up.size(-1) / 2;
  const long rows = gate_up.numel() / (2 * inter);
  auto sizes = gate_up.sizes().vec();
  sizes.back() = inter;
  auto y = torch::empty(sizes, gate_up.options());
  CHECK_HIP(lfmq_silu_mul_fused(gate_up.data_ptr(), y.data_ptr(), rows,
                                (int)inter, stream()));
  return y;
}

torch::Tensor residual_add(torch::Tensor a, torch::Tensor b) {
  check_t(a, torch::kBFloat16, "a");
  check_t(b, torch::kBFloat16, "b");
  auto y = torch::empty_like(a);
  CHECK_HIP(lfmq_residual_add(a.data_ptr(), b.data_ptr(), y.data_ptr(),
                              a.numel(), stream()));
  return y;
}

void kv_append(torch::Tensor kc, torch::Tensor vc, torch::Tensor kn,
               torch::Tensor vn, long pos0) {
  check_t(kc, torch::kBFloat16, "k_cache");
  check_t(vc, torch::kBFloat16, "v_cache");
  check_t(kn, torch::kBFloat16, "k_new");
  check_t(vn, torch::kBFloat16, "v_new");
  // caches [b, hkv, tmax, d]; new [b, s, hkv, d] (token-major)
  const int b = (int)kc.size(0), hkv = (int)kc.size(1);
  const int tmax = (int)kc.size(2), d = (int)kc.size(3);
  const int s = (int)kn.size(1);
  TORCH_CHECK(kn.size(0) == b && kn.size(2) == hkv && kn.size(3) == d,
              "k_new shape mismatch");
  CHECK_HIP(lfmq_kv_append(kc.data_ptr(), vc.data_ptr(), kn.data_ptr(),
                           vn.data_ptr(), b, s, hkv, d, tmax, (int)pos0,
                           stream()));
}

torch::Tensor attn_decode(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                          long seq_len, double scale,
                          c10::optional<torch::Tensor> t_dev) {
  check_t(q, torch::kBFloat16, "q");
  check_t(kc, torch::kBFloat16, "k_cache");
  check_t(vc, torch::kBFloat16, "v_cache");
  const int b = (int)q.size(0), hq = (int)q.size(1), d = (int)q.size(3);
  const int hkv = (int)kc.size(1), tmax = (int)kc.size(2);
  // graph mode (t_dev set): seq_len is MAX T; grid fixed, T read on device
  const int nchunks = ((int)seq_len + 255) / 256;
  auto scratch = torch::empty({(long)b * hq * nchunks * (d + 2)},
                              q.options().dtype(torch::kFloat32));
  auto o = torch::empty_like(q);
  const void* tptr = nullptr;
  if (t_dev.has_value()) {
    check_t(*t_dev, torch::kInt32, "t_dev");
    tptr = t_dev->data_ptr();
  }
  CHECK_HIP(lfmq_attn_decode(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                             o.data_ptr(), scratch.data_ptr(), b, hq, hkv,
                             tmax, (int)seq_len, tptr, d, (float)scale,
                             stream()));
  return o;
}

// EXPERIMENTAL flash prefill (csrc/attn_prefill.hip): q [b,s,hq*d] roped,
// caches [b,hkv,tmax,d] holding rows [0, pos0+s).  d == 128 only.
torch::Tensor attn_prefill(torch::Tensor q, torch::Tensor kc,
                           torch::Tensor vc, long hq, long pos0,
                           double scale) {
  check_t(q, torch::kBFloat16, "q");
  check_t(kc, torch::kBFloat16, "k_cache");
  check_t(vc, torch::kBFloat16, "v_cache");
  const int b = (int)q.size(0), s = (int)q.size(1);
  const int hkv = (int)kc.size(1), tmax = (int)kc.size(2);
  const int d = (int)kc.size(3);
  TORCH_CHECK(q.size(2) == hq * d, "q last dim != hq*d");
  TORCH_CHECK(pos0 + s <= tmax, "cache too small");
  auto o = torch::empty_like(q);
  CHECK_HIP(lfmq_attn_prefill(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                              o.data_ptr(), b, s, (int)hq, hkv, tmax,
                              (int)pos0, d, (float)scale, stream()));
  return o;
}

torch::Tensor dq_gemv_gateup(torch::Tensor x, torch::Tensor qweight,
                             torch::Tensor scales, torch::Tensor qzeros,
                             long bits, long g, double rms_eps,
                             long cols_hint) {
  check_t(x, torch::kBFloat16, "x");
  check_t(qweight, torch::kInt32, "qweight");
  const int m = (int)x.size(0), k = (int)x.size(1);
  const int n = (int)qweight.size(0);
  auto y = torch::empty({m, n / 2}, x.options());
  CHECK_HIP(lfmq_dq_gemv_gateup(qweight.data_ptr(), scales.data_ptr(),
                                qzeros.data_ptr(), x.data_ptr(), y.data_ptr(),
                                m, n, k, (int)g, (int)bits, (float)rms_eps,
                                (int)cols_hint, stream()));
  return y;
}

torch::Tensor qkv_rope_append(torch::Tensor qkv, torch::Tensor kc,
                              torch::Tensor vc, torch::Tensor positions,
                              torch::Tensor inv_freq, long hq, long pos0,
                              c10::optional<torch::Tensor> pos_dev) {
  check_t(qkv, torch::kBFloat16, "qkv");
  check_t(kc, torch::kBFloat16, "k_cache");
  check_t(vc, torch::kBFloat16, "v_cache");
  check_t(positions, torch::kInt32, "positions");
  check_t(inv_freq, torch::kFloat32, "inv_freq");
  const int b = (int)kc.size(0), hkv = (int)kc.size(1);
  const int tmax = (int)kc.size(2), d = (int)kc.size(3);
  const int s = (int)(qkv.numel() / ((hq + 2 * hkv) * d)) / b;
  auto q_out = torch::empty({b, s, hq, (long)d}, qkv.options());
  const void* pdev = nullptr;
  if (pos_dev.has_value()) {
    check_t(*pos_dev, torch::kInt32, "pos_dev");
    pdev = pos_dev->data_ptr();
  }
  CHECK_HIP(lfmq_qkv_rope_append(qkv.data_ptr(), q_out.data_ptr(),
                                 kc.data_ptr(), vc.data_ptr(),
                                 positions.data_ptr(), inv_freq.data_ptr(), b,
                                 s, (int)hq, hkv, d, tmax, (int)pos0, pdev,
                                 stream()));
  return q_out;
}

torch::Tensor greedy_sample(torch::Tensor logits) {
  check_t(logits, torch::kBFloat16, "logits");
  TORCH_CHECK(logits.dim() == 2, "logits must be [b, vocab]");
  const int b = (int)logits.size(0), vocab = (int)logits.size(1);
  auto out = torch::empty({b}, logits.options().dtype(torch::kInt64));
  auto sf = torch::empty({b * 64}, logits.options().dtype(torch::kFloat32));
  auto si = torch::empty({b * 64}, logits.options().dtype(torch::kInt32));
  CHECK_HIP(lfmq_argmax_rows(logits.data_ptr(), out.data_ptr(), sf.data_ptr(),
                             si.data_ptr(), b, vocab, stream()));
  return out;
}

void read_probe(torch::Tensor src, torch::Tensor sink, long row_bytes,
                long n_rows, long nt, long streams, long mode) {
  CHECK_HIP(lfmq_read_probe(src.data_ptr(), sink.data_ptr(), row_bytes,
                            n_rows, (int)nt, (int)streams, (int)mode,
                            stream()));
}

torch::Tensor grouped_dq_gateup(torch::Tensor x, torch::Tensor src_tok,
                                torch::Tensor offsets, torch::Tensor qweight,
                                torch::Tensor scales, torch::Tensor qzeros,
                                long bits, long g,
                                c10::optional<torch::Tensor> active) {
  check_t(x, torch::kBFloat16, "x");
  check_t(qweight, torch::kInt32, "qweight");
  check_t(src_tok, torch::kInt32, "src_tok");
  check_t(offsets, torch::kInt32, "offsets");
  const int E = (int)qweight.size(0), n = (int)qweight.size(1);
  const int k = (int)x.size(1);
  const long slots = src_tok.numel();
  const void* act = nullptr;
  int grid_e = E;
  if (active.has_value()) {
    act = active->data_ptr();
    grid_e = (int)std::min((long)E, slots);
  }
  auto y = torch::empty({slots, (long)(n / 2)}, x.options());
  CHECK_HIP(lfmq_grouped_dq_gemv(qweight.data_ptr(), scales.data_ptr(),
                                 qzeros.data_ptr(), x.data_ptr(), y.data_ptr(),
                                 src_tok.data_ptr(), offsets.data_ptr(), act,
                                 grid_e, E, n, k, (int)g, (int)bits,
                                 (int)x.size(0), true, stream()));
  return y;
}

// one-kernel MoE routing: logits [t, E] bf16 -> (src_tok, slot_w, offsets)
std::vector<torch::Tensor> moe_route(torch::Tensor logits, long topk) {
  check_t(logits, torch::kBFloat16, "logits");
  const int t = (int)logits.size(0), E = (int)logits.size(1);
  auto src_tok = torch::empty({(long)t * topk},
                              logits.options().dtype(torch::kInt32));
  auto slot_w = torch::empty({(long)t * topk},
                             logits.options().dtype(torch::kFloat32));
  auto offsets = torch::empty({(long)E + 1},
                              logits.options().dtype(torch::kInt32));
  auto active = torch::empty({(long)E},
                             logits.options().dtype(torch::kInt32));
  CHECK_HIP(lfmq_moe_route(logits.data_ptr(), src_tok.data_ptr(),
                           slot_w.data_ptr(), offsets.data_ptr(),
                           active.data_ptr(), t, E, (int)topk, stream()));
  return {src_tok, slot_w, offsets, active};
}

// w2 grouped GEMV + fused weighted scatter into y_f32 [t, n] (caller zeroed)
void grouped_dq_linear_scatter(torch::Tensor x, torch::Tensor offsets,
                               torch::Tensor dst_tok, torch::Tensor slot_w,
                               torch::Tensor y_f32, torch::Tensor qweight,
                               torch::Tensor scales, torch::Tensor qzeros,
                               long bits, long g,
                               c10::optional<torch::Tensor> active) {
  check_t(x, torch::kBFloat16, "x");
  check_t(y_f32, torch::kFloat32, "y_f32");
  check_t(dst_tok, torch::kInt32, "dst_tok");
  check_t(slot_w, torch::kFloat32, "slot_w");
  check_t(offsets, torch::kInt32, "offsets");
  const int E = (int)qweight.size(0), n = (int)qweight.size(1);
  const int k = (int)x.size(1);
  TORCH_CHECK(y_f32.size(1) == n, "y_f32 width mismatch");
  const void* act = nullptr;
  int grid_e = E;
  if (active.has_value()) {
    act = active->data_ptr();
    grid_e = std::min((long)E, dst_tok.numel());
  }
  CHECK_HIP(lfmq_grouped_dq_gemv_scatter(
      qweight.data_ptr(), scales.data_ptr(), qzeros.data_ptr(), x.data_ptr(),
      y_f32.data_ptr(), offsets.data_ptr(), dst_tok.data_ptr(),
      slot_w.data_ptr(), act, grid_e, n, k, (int)g, (int)bits,
      (int)x.size(0), stream()));
}

torch::Tensor grouped_dq_linear(torch::Tensor x, torch::Tensor offsets,
                                torch::Tensor qweight, torch::Tensor scales,
                                torch::Tensor qzeros, long bits, long g) {
  check_t(x, torch::kBFloat16, "x");
  check_t(qweight, torch::kInt32, "qweight");
  check_t(offsets, torch::kInt32, "offsets");
  const int E = (int)qweight.size(0), n = (int)qweight.size(1);
  const int k = (int)x.size(1);
  auto y = torch::empty({x.size(0), (long)n}, x.options());
  CHECK_HIP(lfmq_grouped_dq_gemv(qweight.data_ptr(), scales.data_ptr(),
                                 qzeros.data_ptr(), x.data_ptr(), y.data_ptr(),
                                 nullptr, offsets.data_ptr(), nullptr, E, E,
                                 n, k, (int)g, (int)bits, (int)x.size(0),
                                 false, stream()));
  return y;
}

void advance_record(torch::Tensor state, torch::Tensor tok_buf,
                    torch::Tensor sampled, torch::Tensor ring) {
  check_t(state, torch::kInt32, "state");
  const int b = (int)tok_buf.size(0);
  const int ring_len = (int)ring.size(-1);
  CHECK_HIP(lfmq_advance_record(state.data_ptr(), tok_buf.data_ptr(),
                                sampled.data_ptr(), ring.data_ptr(), b,
                                ring_len, stream()));
}

torch::Tensor attn_decode_fused(torch::Tensor qkv, torch::Tensor kc,
                                torch::Tensor vc, long seq_len, double scale,
                                torch::Tensor inv_freq, long hq,
                                c10::optional<torch::Tensor> t_dev,
                                c10::optional<torch::Tensor> tickets) {
  check_t(qkv, torch::kBFloat16, "qkv");
  check_t(kc, torch::kBFloat16, "k_cache");
  check_t(vc, torch::kBFloat16, "v_cache");
  check_t(inv_freq, torch::kFloat32, "inv_freq");
  const int b = (int)kc.size(0), hkv = (int)kc.size(1);
  const int tmax = (int)kc.size(2), d = (int)kc.size(3);
  const int nchunks = ((int)seq_len + 63) / 64;  // fused path: FCH = 64
  auto scratch = torch::empty({(long)b * hq * nchunks * (d + 2)},
                              qkv.options().dtype(torch::kFloat32));
  auto o = torch::empty({(long)b, 1L, hq * (long)d}, qkv.options());
  const void* tptr = nullptr;
  if (t_dev.has_value()) tptr = t_dev->data_ptr();
  void* tk = nullptr;
  if (tickets.has_value()) {
    check_t(*tickets, torch::kInt32, "tickets");
    TORCH_CHECK(tickets->numel() >= (long)b * hq, "tickets too small");
    tk = tickets->data_ptr();
  }
  CHECK_HIP(lfmq_attn_decode_fused(qkv.data_ptr(), kc.data_ptr(),
                                   vc.data_ptr(), o.data_ptr(),
                                   scratch.data_ptr(), inv_freq.data_ptr(), b,
                                   (int)hq, hkv, tmax, (int)seq_len, tptr, d,
                                   (float)scale, tk, stream()));
  return o;
}

// persistent decode engine: one launch = one whole decode step (layers
// only; lm_head/sampling stay separate).  Granule buffers must be zeroed
// by the CALLER before every call (lfmq/models/engine.py does, inside the
// captured graph).  Returns nothing; output lands in xout.
void engine_step(torch::Tensor stream_t, long stream_per_cu, long nslot_ring,
                 long n_layers, long hidden, long inter, long nqkv, long hq,
                 long hkv, long log2g, double rms_eps, double attn_scale,
                 torch::Tensor inv_freq, torch::Tensor kc, torch::Tensor vc,
                 torch::Tensor state, torch::Tensor tok, torch::Tensor embed,
                 torch::Tensor final_w, torch::Tensor xout,
                 torch::Tensor g_qkv, torch::Tensor g_attn, torch::Tensor g_o,
                 torch::Tensor g_h, torch::Tensor g_down, torch::Tensor g_scr,
                 torch::Tensor tmo, c10::optional<torch::Tensor> prof,
                 long debug) {
  check_t(kc, torch::kBFloat16, "k_cache");
  check_t(vc, torch::kBFloat16, "v_cache");
  check_t(embed, torch::kBFloat16, "embed");
  check_t(final_w, torch::kBFloat16, "final_w");
  check_t(xout, torch::kBFloat16, "xout");
  check_t(inv_freq, torch::kFloat32, "inv_freq");
  check_t(state, torch::kInt32, "state");
  TORCH_CHECK(tok.scalar_type() == torch::kInt64, "tok must be int64");
  TORCH_CHECK(stream_t.scalar_type() == torch::kUInt8, "stream uint8");
  // kc: [L, 1, hkv, tmax, d] (stacked) or list-stacked contiguous
  const int tmax = (int)kc.size(-2), d = (int)kc.size(-1);
  EngineArgs a{};
  a.stream = (const char*)stream_t.data_ptr();
  a.stream_per_cu = (unsigned long long)stream_per_cu;
  a.n_slots_total = (int)((stream_per_cu + 16383) / 16384);
  a.nslot_ring = (int)nslot_ring;
  a.n_layers = (int)n_layers;
  a.hidden = (int)hidden;
  a.inter = (int)inter;
  a.nqkv = (int)nqkv;
  a.hq = (int)hq;
  a.hkv = (int)hkv;
  a.log2g = (int)log2g;
  a.rms_eps = (float)rms_eps;
  a.attn_scale = (float)attn_scale;
  a.inv_freq = (const float*)inv_freq.data_ptr();
  a.kc = (const unsigned short*)kc.data_ptr();
  a.vc = (const unsigned short*)vc.data_ptr();
  a.kv_layer_stride = (unsigned long long)hkv * tmax * d;
  a.tmax = tmax;
  a.nch_max = (tmax + 255) / 256;
  TORCH_CHECK(a.nch_max <= 64, "engine needs tmax <= 16384");
  a.T_dev = (const int*)state.data_ptr() + 1;  // state = {pos0, T, step}
  a.tok = (const long long*)tok.data_ptr();
  a.embed = (const unsigned short*)embed.data_ptr();
  a.final_w = (const unsigned short*)final_w.data_ptr();
  a.xout = (unsigned short*)xout.data_ptr();
  a.g_qkv = g_qkv.data_ptr();
  a.g_attn = g_attn.data_ptr();
  a.g_o = g_o.data_ptr();
  a.g_h = g_h.data_ptr();
  a.g_down = g_down.data_ptr();
  a.g_scr = g_scr.data_ptr();
  a.tmo = tmo.data_ptr();
  a.prof = prof.has_value()
               ? (unsigned long long*)prof->data_ptr()
               : nullptr;
  a.debug = (unsigned)debug;
  CHECK_HIP(lfmq_engine_step(&a, d, stream()));
}

void hessian_accum(torch::Tensor h, torch::Tensor x) {
  check_t(h, torch::kFloat32, "H");
  check_t(x, torch::kBFloat16, "x");
  const int t = (int)x.size(0), k = (int)x.size(1);
  TORCH_CHECK(h.size(0) == k && h.size(1) == k, "H must be [k, k]");
  CHECK_HIP(lfmq_hessian_accum(x.data_ptr(), h.data_ptr(), t, k, stream()));
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("pack", &pack, "pack int codes");
  m.def("unpack", &unpack, "unpack int codes");
  m.def("dq_linear", &dq_linear, "fused dequant GEMV/GEMM");
  m.def("rmsnorm", &rmsnorm, "fused rmsnorm");
  m.def("rmsnorm_residual", &rmsnorm_residual, "fused rmsnorm + residual");
  m.def("rope", &rope, "fused rope (in-place)");
  m.def("silu_mul", &silu_mul, "silu(gate) * up");
  m.def("silu_mul_fused", &silu_mul_fused, "silu-mul on fused gate_up");
  m.def("residual_add", &residual_add, "bf16 add");
  m.def("kv_append", &kv_append, "KV cache append");
  m.def("attn_decode", &attn_decode, "GQA decode attention");
  m.def("attn_prefill", &attn_prefill,
        "EXPERIMENTAL flash prefill attention (LFMQ_FLASH_PREFILL)");
  m.def("hessian_accum", &hessian_accum, "H += 2 X^T X");
  m.def("dq_gemv_gateup", &dq_gemv_gateup, "fused gate_up GEMV + silu*mul");
  m.def("qkv_rope_append", &qkv_rope_append, "fused rope + KV append");
  m.def("greedy_sample", &greedy_sample, "row argmax");
  m.def("read_probe", &read_probe, "bw probe");
  m.def("advance_record", &advance_record, "graph decode-state advance");
  m.def("attn_decode_fused", &attn_decode_fused,
        "fused rope+KV-append+attention decode");
  m.def("grouped_dq_gateup", &grouped_dq_gateup, "MoE grouped gate_up+silu");
  m.def("grouped_dq_linear", &grouped_dq_linear, "MoE grouped linear");
  m.def("moe_route", &moe_route, "one-kernel MoE top-k routing");
  m.def("grouped_dq_linear_scatter", &grouped_dq_linear_scatter,
        "MoE w2 grouped GEMV + fused weighted scatter");
  m.def("engine_step", &engine_step, "persistent decode engine step");
}
