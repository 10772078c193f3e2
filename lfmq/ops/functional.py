"""Device-typed op entry points used by the model runtime.

On a CUDA (ROCm) device every call goes to the hand-written gfx950 HIP
kernel — a missing extension raises (lfmq.ops.hip_ext), it never falls
back to eager torch.  On CPU the exact reference twin runs (unit tests +
the OPT int8 CPU decode config).  This is a device split, not a backend
dispatch: per device there is exactly one implementation.
"""
from __future__ import annotations

import math
import os
from typing import Optional

import torch

from . import hip_ext, ref

# Debug numeric guard: LFMQ_NANCHECK=1 asserts every op
# output is finite — catches kernel-layout or quantization bugs at the
# op that produced them instead of 30 layers later.
_NANCHECK = os.environ.get("LFMQ_NANCHECK", "0") == "1"


def _guard(t: torch.Tensor, name: str) -> torch.Tensor:
    if _NANCHECK and not bool(torch.isfinite(t.float()).all()):
        raise FloatingPointError(f"lfmq op {name} produced non-finite values")
    return t


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5,
            residual: Optional[torch.Tensor] = None) -> torch.Tensor:
    if x.is_cuda:
        return hip_ext().rmsnorm(x, weight, eps, residual)
    return ref.rmsnorm(x, weight, eps, residual)


def rmsnorm_residual(x: torch.Tensor, residual: torch.Tensor,
                     weight: torch.Tensor, eps: float = 1e-5):
    """Returns (normed, new_residual = x + residual) in one pass."""
    if x.is_cuda:
        return hip_ext().rmsnorm_residual(x, residual, weight, eps)
    new_res = ref.residual_add(x, residual)
    return ref.rmsnorm(new_res, weight, eps), new_res


_INV_FREQ_CACHE: dict = {}
# zero-initialized arrival tickets for the in-launch attention combine:
# the winning chunk resets its entry, so one buffer per (n, device) stays
# valid across steps and graph replays
_TICKETS_CACHE: dict = {}


def _tickets(n: int, device) -> "torch.Tensor":
    key = (n, str(device))
    t = _TICKETS_CACHE.get(key)
    if t is None:
        t = torch.zeros(n, dtype=torch.int32, device=device)
        _TICKETS_CACHE[key] = t
    return t


def _inv_freq(d: int, theta: float, device) -> torch.Tensor:
    key = (d, theta, str(device))
    t = _INV_FREQ_CACHE.get(key)
    if t is None:
        t = ref.rope_freqs(d, theta).to(device)
        _INV_FREQ_CACHE[key] = t
    return t


def apply_rope(q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor,
               theta: float = 500000.0):
    """q: [b, s, hq, d], k: [b, s, hkv, d] — rotated in place on GPU."""
    if q.is_cuda:
        q = q.contiguous()
        k = k.contiguous()
        hip_ext().rope(q, k, positions.to(torch.int32),
                       _inv_freq(q.shape[-1], theta, q.device))
        return q, k
    return ref.apply_rope(q, positions, theta), ref.apply_rope(k, positions, theta)


def dq_linear(x: torch.Tensor, qweight: torch.Tensor, scales: torch.Tensor,
              qzeros: torch.Tensor, bits: int, group_size: int,
              bias: Optional[torch.Tensor] = None,
              resid: Optional[torch.Tensor] = None,
              rms_eps: float = -1.0) -> torch.Tensor:
    """Fused dequant-GEMM y = x @ dequant(W)^T (+bias) (+residual)
    (rms_eps >= 0: fused-RMSNorm input, decode m<=4 GPU path only)."""
    if x.is_cuda:
        x2 = x.reshape(-1, x.shape[-1])
        r2 = resid.reshape(-1, resid.shape[-1]).contiguous() if resid is not None else None
        y = hip_ext().dq_linear(x2, qweight, scales, qzeros, bits, group_size,
                                bias, r2, rms_eps, 0)
        return _guard(y, "dq_linear").reshape(*x.shape[:-1], y.shape[-1])
    assert resid is None and rms_eps < 0, "fused rms/resid is GPU-only"
    return ref.dq_linear(x, qweight, scales, qzeros, bits, group_size, bias).to(x.dtype)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if gate.is_cuda:
        return hip_ext().silu_mul(gate.contiguous(), up.contiguous())
    return ref.silu_mul(gate, up)


def silu_mul_gateup(gate_up: torch.Tensor, inter: int) -> torch.Tensor:
    """silu(gu[..., :inter]) * gu[..., inter:] without materializing the
    split halves (one fused kernel on GPU)."""
    if gate_up.is_cuda:
        return hip_ext().silu_mul_fused(gate_up.contiguous())
    gate, up = gate_up.split([inter, inter], dim=-1)
    return ref.silu_mul(gate, up)


def attn_decode(q, k_cache, v_cache, seq_len: int, scale: Optional[float] = None):
    if q.is_cuda:
        return _guard(hip_ext().attn_decode(
            q, k_cache, v_cache, seq_len,
            scale or 1.0 / math.sqrt(q.shape[-1]), None), "attn_decode")
    return ref.attn_decode(q, k_cache, v_cache, seq_len, scale)


def attn_prefill(q, k, v, causal: bool = True, scale: Optional[float] = None):
    """Prefill attention. GPU: explicit bf16 matmul chain via rocBLAS (the
    PyTorch-ROCm layer; never SDPA, which can route to aotriton/Triton).
    GQA is handled by folding the q-head repeat into the row dimension
    (no K/V repeat_interleave copies), and the query axis is processed in
    chunks so the fp32 score tile stays bounded (a full 2048-seq layer
    would otherwise materialize a 537 MB score tensor)."""
    if q.is_cuda:
        b, hq, s, d = q.shape
        hkv = k.shape[1]
        rep = hq // hkv
        scale = scale or 1.0 / math.sqrt(d)
        kt = k.transpose(-1, -2)                      # [b, hkv, d, s]
        out = torch.empty_like(q)
        CH = 512 if s > 512 else s
        for q0 in range(0, s, CH):
            q1 = min(s, q0 + CH)
            # [b, hkv, rep*(q1-q0), d]: repeat folded into the row dim
            qc = (q.view(b, hkv, rep, s, d)[:, :, :, q0:q1]
                  .reshape(b, hkv, rep * (q1 - q0), d))
            sc = torch.matmul(qc, kt).float() * scale  # [b,hkv,rep*cq,s]
            if causal:
                t_idx = torch.arange(s, device=q.device)
                rows = torch.arange(q0, q1, device=q.device)
                m = (t_idx[None, :] > rows[:, None])   # [cq, s]
                sc = sc.view(b, hkv, rep, q1 - q0, s).masked_fill(
                    m[None, None, None], float("-inf")).view(
                    b, hkv, rep * (q1 - q0), s)
            p = torch.softmax(sc, dim=-1).to(q.dtype)
            oc = torch.matmul(p, v)                    # [b,hkv,rep*cq,d]
            out.view(b, hkv, rep, s, d)[:, :, :, q0:q1] = (
                oc.view(b, hkv, rep, q1 - q0, d))
        return out
    return ref.attn_prefill(q, k, v, causal, scale)


def attn_prefill_fused(q: torch.Tensor, k_cache: torch.Tensor,
                       v_cache: torch.Tensor, pos0: int,
                       scale: Optional[float] = None) -> torch.Tensor:
    """GPU flash prefill (csrc/attn_prefill.hip): one MFMA kernel instead
    of the rocBLAS matmul chain (measured 3.3-6.3x faster at s 256-2048,
    rel_err ~4e-3 vs fp32; tools/prefill_flash_bench.py).  q [b, s, hq, d]
    already roped (the qkv_rope_append output); caches [b, hkv, Tmax, d]
    hold rows [0, pos0+s).  d == 128 only.  Default GPU prefill path;
    LFMQ_FLASH_PREFILL=0 restores the matmul chain."""
    b, s, hq, d = q.shape
    scale = scale or 1.0 / math.sqrt(d)
    o = _guard(hip_ext().attn_prefill(q.reshape(b, s, hq * d), k_cache,
                                      v_cache, hq, pos0, scale),
               "attn_prefill_fused")
    return o.view(b, s, hq, d)


USE_FLASH_PREFILL = os.environ.get("LFMQ_FLASH_PREFILL", "1") == "1"


def qkv_rope_append(qkv: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, positions: torch.Tensor,
                    pos0: int, hq: int, theta: float = 500000.0) -> torch.Tensor:
    """GPU-only fused decode epilogue of the qkv projection: rope(q),
    rope(k)+append, append(v) in one launch.  qkv: [b, s, (hq+2hkv)*d];
    returns q [b, s, hq, d] (== [b, hq, 1, d] layout at s == 1)."""
    d = k_cache.shape[-1]
    return hip_ext().qkv_rope_append(qkv.contiguous(), k_cache, v_cache,
                                     positions.to(torch.int32),
                                     _inv_freq(d, theta, qkv.device), hq,
                                     pos0, None)


def qkv_rope_append_graph(qkv: torch.Tensor, k_cache: torch.Tensor,
                          v_cache: torch.Tensor, state: torch.Tensor,
                          hq: int, theta: float = 500000.0) -> torch.Tensor:
    """hipGraph decode variant: position read from state[0] on device."""
    d = k_cache.shape[-1]
    pos_dev = state[0:1]
    return hip_ext().qkv_rope_append(qkv.contiguous(), k_cache, v_cache,
                                     pos_dev, _inv_freq(d, theta, qkv.device),
                                     hq, 0, pos_dev)


def attn_decode_graph(q, k_cache, v_cache, state, max_t: int, scale: float):
    """hipGraph decode variant: T read from state[1]; grid sized for max_t."""
    return hip_ext().attn_decode(q, k_cache, v_cache, max_t, scale, state[1:2])


def attn_decode_fused(qkv, k_cache, v_cache, state_or_T, max_t: int,
                      scale: float, hq: int, theta: float = 500000.0):
    """One launch: rope(q,k) off the raw qkv buffer, KV-cache append,
    chunked decode attention.  state_or_T: device state tensor (graph
    mode, T = state[1]) or a host int T."""
    d = k_cache.shape[-1]
    inv = _inv_freq(d, theta, qkv.device)
    tk = _tickets(k_cache.shape[0] * hq, qkv.device)
    if torch.is_tensor(state_or_T):
        return hip_ext().attn_decode_fused(qkv.contiguous(), k_cache, v_cache,
                                           max_t, scale, inv, hq,
                                           state_or_T[1:2], tk)
    return hip_ext().attn_decode_fused(qkv.contiguous(), k_cache, v_cache,
                                       int(state_or_T), scale, inv, hq, None,
                                       tk)


def dq_gemv_gateup(x: torch.Tensor, qweight: torch.Tensor,
                   scales: torch.Tensor, qzeros: torch.Tensor, bits: int,
                   group_size: int, rms_eps: float = -1.0) -> torch.Tensor:
    """GPU-only fused gate_up GEMV + silu·mul epilogue (m <= 4);
    rms_eps >= 0 additionally fuses the RMSNorm of the input."""
    x2 = x.reshape(-1, x.shape[-1])
    y = hip_ext().dq_gemv_gateup(x2, qweight, scales, qzeros, bits,
                                 group_size, rms_eps, 0)
    return y.reshape(*x.shape[:-1], y.shape[-1])


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    """argmax over the last dim of [b, vocab] logits -> int64 [b]."""
    if logits.is_cuda:
        return hip_ext().greedy_sample(logits.contiguous())
    return logits.argmax(dim=-1)


def kv_append(k_cache, v_cache, k_new, v_new, pos: int) -> None:
    """caches [b, hkv, T_max, d]; k_new/v_new TOKEN-major [b, s, hkv, d]
    (as the qkv split produces them — saves the transpose copies)."""
    if k_cache.is_cuda:
        hip_ext().kv_append(k_cache, v_cache, k_new.contiguous(),
                            v_new.contiguous(), pos)
        return
    ref.kv_append(k_cache, v_cache, k_new.transpose(1, 2), v_new.transpose(1, 2), pos)
