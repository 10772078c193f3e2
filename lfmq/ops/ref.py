"""CPU reference twins of every kernel in csrc/ — the unit-test oracle.

Exact, simple torch-CPU (fp32) implementations.  Every HIP kernel's GPU
parity test (tests/test_gpu_kernels.py, -m gpu) compares against one of
these.  The OPT-125m int8 CPU decode path (BASELINE.json config #1) also
runs on these — that is a spec'd CPU path, not a dispatch fallback.

Packed layout v1 (FORMAT_VERSION = 1, lfmq/config.py):
  qweight: uint32 [n, k/8]   — nibble i (little-endian: bits 4i..4i+3) of
                                word w holds the code for k = 8*w + i.
                                int8: uint32 [n, k/4], byte i of word w is
                                the code for k = 4*w + i.
  scales:  bf16   [n, ceil(k/g)]
  zeros:   uint32 [n, ceil(ceil(k/g)/8)] int4-packed zero-point codes
           (int8: [n, ceil(ceil(k/g)/4)] byte-packed).
  Symmetric mode packs codes offset by 2^(bits-1) (e.g. int4 sym code =
  clamp(round(w/s), -8, 7) + 8) and records zeros = 2^(bits-1).

Dequant everywhere: w = (code - zero) * scale, computed in fp32.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


# ---------------------------------------------------------------------------
# group-wise quantize / dequantize (K12 rtn_quant reference)
# ---------------------------------------------------------------------------

def _group_view(w: torch.Tensor, group_size: int) -> Tuple[torch.Tensor, int]:
    """[n, k] -> [n, n_groups, g].  group_size -1 => one group per row."""
    n, k = w.shape
    g = k if group_size == -1 else group_size
    if k % g != 0:
        raise ValueError(f"k={k} not divisible by group_size={g}")
    return w.reshape(n, k // g, g), g


def quant_groupwise(
    w: torch.Tensor, bits: int, group_size: int, sym: bool
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Round-to-nearest group-wise quantization.

    w: fp32 [n, k].  Returns (codes int32 [n,k] in [0, 2^bits-1],
    scales fp32 [n, n_groups], zeros int32 [n, n_groups]).
    """
    w = w.float()
    wg, g = _group_view(w, group_size)
    qmax = (1 << bits) - 1
    if sym:
        absmax = wg.abs().amax(dim=-1)
        scale = absmax / ((qmax - 1) / 2)  # int4: /7.0, int8: /127.0
        scale = torch.where(scale == 0, torch.ones_like(scale), scale)
        zero = torch.full_like(scale, (qmax + 1) // 2, dtype=torch.int32)
        q = torch.round(wg / scale.unsqueeze(-1)) + zero.unsqueeze(-1)
    else:
        wmin = wg.amin(dim=-1)
        wmax = wg.amax(dim=-1)
        scale = (wmax - wmin) / qmax
        scale = torch.where(scale == 0, torch.ones_like(scale), scale)
        zero = torch.clamp(torch.round(-wmin / scale), 0, qmax).to(torch.int32)
        q = torch.round(wg / scale.unsqueeze(-1)) + zero.unsqueeze(-1)
    q = torch.clamp(q, 0, qmax).to(torch.int32)
    n, k = w.shape
    return q.reshape(n, k), scale, zero


def quant_per_tensor_int8(w: torch.Tensor) -> Tuple[torch.Tensor, float]:
    """Per-tensor symmetric int8 (config #1).  Returns (codes int8, scale)."""
    w = w.float()
    absmax = float(w.abs().max())
    scale = absmax / 127.0 if absmax > 0 else 1.0
    q = torch.clamp(torch.round(w / scale), -128, 127).to(torch.int8)
    return q, scale


def dequant_groupwise(
    codes: torch.Tensor, scales: torch.Tensor, zeros: torch.Tensor, group_size: int
) -> torch.Tensor:
    """codes int32 [n,k], scales fp32 [n,G], zeros int32 [n,G] -> fp32 [n,k]."""
    n, k = codes.shape
    g = k if group_size == -1 else group_size
    cg = codes.reshape(n, k // g, g).float()
    return ((cg - zeros.unsqueeze(-1).float()) * scales.unsqueeze(-1).float()).reshape(n, k)


# ---------------------------------------------------------------------------
# K2 pack / unpack
# ---------------------------------------------------------------------------

def pack_codes(codes: torch.Tensor, bits: int) -> torch.Tensor:
    """int32 codes [n, k] in [0, 2^bits-1] -> uint32-packed [n, k*bits/32].

    Little-endian within the word: element i of a word sits at bits
    [i*bits, (i+1)*bits).  k must be divisible by 32/bits.
    """
    n, k = codes.shape
    per = 32 // bits
    if k % per != 0:
        raise ValueError(f"k={k} not divisible by {per}")
    c = codes.to(torch.int64).reshape(n, k // per, per)
    shifts = torch.arange(per, dtype=torch.int64, device=codes.device) * bits
    packed = (c << shifts).sum(dim=-1)
    return (packed & 0xFFFFFFFF).to(torch.int32)  # stored as int32, bits are uint32


def unpack_codes(packed: torch.Tensor, bits: int, k: int) -> torch.Tensor:
    """Inverse of pack_codes -> int32 [n, k]."""
    n = packed.shape[0]
    per = 32 // bits
    p = packed.to(torch.int64) & 0xFFFFFFFF
    shifts = torch.arange(per, dtype=torch.int64, device=packed.device) * bits
    mask = (1 << bits) - 1
    c = (p.unsqueeze(-1) >> shifts) & mask
    return c.reshape(n, -1)[:, :k].to(torch.int32)


def pack_zeros(zeros: torch.Tensor, bits: int) -> torch.Tensor:
    """int32 zeros [n, G] -> packed uint32 [n, ceil(G*bits/32)], zero-padded."""
    n, G = zeros.shape
    per = 32 // bits
    pad = (-G) % per
    if pad:
        zeros = torch.cat([zeros, zeros.new_zeros(n, pad)], dim=1)
    return pack_codes(zeros, bits)


def unpack_zeros(packed: torch.Tensor, bits: int, G: int) -> torch.Tensor:
    return unpack_codes(packed, bits, G)


# ---------------------------------------------------------------------------
# packed-linear forward (K3/K4 reference): y = x @ dequant(W)^T
# ---------------------------------------------------------------------------

def dequant_weight(
    qweight: torch.Tensor, scales: torch.Tensor, qzeros: torch.Tensor,
    bits: int, group_size: int, k: int,
    g_idx: "Optional[torch.Tensor]" = None,
) -> torch.Tensor:
    """Packed tensors -> fp32 weight [n, k] (W[n,k]: out x in, torch Linear
    layout).  g_idx (act-order GPTQ): per-column group index."""
    n = qweight.shape[0]
    G = scales.shape[1]
    codes = unpack_codes(qweight, bits, k)
    zeros = unpack_zeros(qzeros, bits, G)
    if g_idx is not None:
        gi = g_idx.long()
        return ((codes.float() - zeros.float()[:, gi])
                * scales.float()[:, gi])
    g = k if group_size == -1 else group_size
    return dequant_groupwise(codes, scales.float(), zeros, g)


def dq_linear(
    x: torch.Tensor, qweight: torch.Tensor, scales: torch.Tensor,
    qzeros: torch.Tensor, bits: int, group_size: int,
    bias: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Reference of the fused dequant-GEMM: y[m,n] = x[m,k] · W[n,k]^T."""
    k = x.shape[-1]
    w = dequant_weight(qweight, scales, qzeros, bits, group_size, k)
    y = x.float() @ w.t()
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


# ---------------------------------------------------------------------------
# K1 hessian accumulate
# ---------------------------------------------------------------------------

def hessian_accum(h: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """H += 2 · Xᵀ X.  x: [ntok, k] (any float dtype), h: fp32 [k, k]."""
    xf = x.float()
    h += 2.0 * (xf.t() @ xf)
    return h


# ---------------------------------------------------------------------------
# K6 rmsnorm (+ optional fused residual add)
# ---------------------------------------------------------------------------

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5,
            residual: Optional[torch.Tensor] = None) -> torch.Tensor:
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * weight.float()
    return y.to(x.dtype)


# ---------------------------------------------------------------------------
# K7 rope (NeoX non-interleaved halves, Llama-3 θ=500000)
# ---------------------------------------------------------------------------

def rope_freqs(head_dim: int, theta: float = 500000.0) -> torch.Tensor:
    return 1.0 / (theta ** (torch.arange(0, head_dim, 2).float() / head_dim))


def apply_rope(x: torch.Tensor, positions: torch.Tensor,
               theta: float = 500000.0) -> torch.Tensor:
    """x: [..., seq, n_heads, head_dim]; positions: [seq] int."""
    d = x.shape[-1]
    inv = rope_freqs(d, theta)                      # [d/2]
    ang = positions.float().unsqueeze(-1) * inv     # [seq, d/2]
    cos = ang.cos()[..., None, :]                   # [seq, 1, d/2]
    sin = ang.sin()[..., None, :]
    xf = x.float()
    x1, x2 = xf[..., : d // 2], xf[..., d // 2:]
    out = torch.cat([x1 * cos - x2 * sin, x1 * sin + x2 * cos], dim=-1)
    return out.to(x.dtype)


# ---------------------------------------------------------------------------
# K8 attention decode (single/few token vs KV cache, GQA)
# ---------------------------------------------------------------------------

def attn_decode(q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
                seq_len: int, scale: Optional[float] = None) -> torch.Tensor:
    """q: [b, hq, 1, d]; caches: [b, hkv, T_max, d]; attends to [0, seq_len)."""
    b, hq, one, d = q.shape
    hkv = k_cache.shape[1]
    rep = hq // hkv
    scale = scale or 1.0 / math.sqrt(d)
    kc = k_cache[:, :, :seq_len].float()
    vc = v_cache[:, :, :seq_len].float()
    kc = kc.repeat_interleave(rep, dim=1)
    vc = vc.repeat_interleave(rep, dim=1)
    s = torch.einsum("bhqd,bhtd->bhqt", q.float(), kc) * scale
    p = torch.softmax(s, dim=-1)
    o = torch.einsum("bhqt,bhtd->bhqd", p, vc)
    return o.to(q.dtype)


def attn_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 causal: bool = True, scale: Optional[float] = None) -> torch.Tensor:
    """q: [b, hq, s, d]; k,v: [b, hkv, s, d].  Materializing reference."""
    b, hq, s, d = q.shape
    hkv = k.shape[1]
    rep = hq // hkv
    scale = scale or 1.0 / math.sqrt(d)
    kf = k.float().repeat_interleave(rep, dim=1)
    vf = v.float().repeat_interleave(rep, dim=1)
    sc = torch.einsum("bhqd,bhtd->bhqt", q.float(), kf) * scale
    if causal:
        mask = torch.full((s, s), float("-inf")).triu(1)
        sc = sc + mask
    p = torch.softmax(sc, dim=-1)
    o = torch.einsum("bhqt,bhtd->bhqd", p, vf)
    return o.to(q.dtype)


# ---------------------------------------------------------------------------
# K9 silu·mul, K10 kv append, K11 residual add
# ---------------------------------------------------------------------------

def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    gf = gate.float()
    return (gf * torch.sigmoid(gf) * up.float()).to(gate.dtype)


def kv_append(k_cache: torch.Tensor, v_cache: torch.Tensor,
              k_new: torch.Tensor, v_new: torch.Tensor, pos: int) -> None:
    """caches [b, hkv, T_max, d]; new [b, hkv, s, d]; writes at [pos, pos+s)."""
    s = k_new.shape[2]
    k_cache[:, :, pos:pos + s] = k_new.to(k_cache.dtype)
    v_cache[:, :, pos:pos + s] = v_new.to(v_cache.dtype)


def residual_add(x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    return (x.float() + y.float()).to(x.dtype)


# ---------------------------------------------------------------------------
# K5 grouped dequant GEMM (MoE experts) reference
# ---------------------------------------------------------------------------

def grouped_dq_linear(
    x: torch.Tensor, expert_ids: torch.Tensor,
    qweight: torch.Tensor, scales: torch.Tensor, qzeros: torch.Tensor,
    bits: int, group_size: int,
) -> torch.Tensor:
    """x: [t, k] tokens sorted by expert; expert_ids: [t] int; packed weights
    stacked per expert: qweight [E, n, k*bits/32] etc.  Returns [t, n]."""
    t, k = x.shape
    E, n = qweight.shape[0], qweight.shape[1]
    out = torch.zeros(t, n, dtype=x.dtype)
    for e in range(E):
        m = expert_ids == e
        if m.any():
            out[m] = dq_linear(x[m], qweight[e], scales[e], qzeros[e], bits, group_size)
    return out
