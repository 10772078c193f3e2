"""Semantics-pinning tests for the CPU reference twins (the kernel oracle)."""
import math

import pytest
import torch

from lfmq.ops import ref


def test_pack_unpack_roundtrip_int4():
    codes = torch.randint(0, 16, (8, 64), dtype=torch.int32)
    packed = ref.pack_codes(codes, 4)
    assert packed.shape == (8, 8)
    assert torch.equal(ref.unpack_codes(packed, 4, 64), codes)


def test_pack_unpack_roundtrip_int8():
    codes = torch.randint(0, 256, (5, 32), dtype=torch.int32)
    packed = ref.pack_codes(codes, 8)
    assert packed.shape == (5, 8)
    assert torch.equal(ref.unpack_codes(packed, 8, 32), codes)


def test_pack_nibble_order_little_endian():
    # word 0 of row 0 must hold codes[0..7], code i at bits 4i..4i+3
    codes = torch.arange(8, dtype=torch.int32).reshape(1, 8)
    packed = ref.pack_codes(codes, 4)
    word = packed[0, 0].item() & 0xFFFFFFFF
    for i in range(8):
        assert (word >> (4 * i)) & 0xF == i


def test_pack_rejects_bad_k():
    with pytest.raises(ValueError):
        ref.pack_codes(torch.zeros(2, 7, dtype=torch.int32), 4)


@pytest.mark.parametrize("sym", [False, True])
@pytest.mark.parametrize("bits", [4, 8])
def test_quant_dequant_groupwise(bits, sym):
    w = torch.randn(16, 256)
    codes, s, z = ref.quant_groupwise(w, bits, 128, sym)
    assert codes.min() >= 0 and codes.max() <= (1 << bits) - 1
    dq = ref.dequant_groupwise(codes, s, z, 128)
    # error bounded by half a quantization step per element
    step = s.repeat_interleave(128, dim=1)
    assert ((w - dq).abs() <= step * 0.5 + 1e-6).all()


def test_quant_groupwise_rejects_ragged():
    with pytest.raises(ValueError):
        ref.quant_groupwise(torch.randn(4, 100), 4, 64, False)


def test_quant_per_tensor_int8():
    w = torch.randn(8, 32)
    q, s = ref.quant_per_tensor_int8(w)
    assert q.dtype == torch.int8
    assert ((w - q.float() * s).abs() <= s * 0.5 + 1e-7).all()


def test_zeros_pack_roundtrip_padding():
    z = torch.randint(0, 16, (4, 5), dtype=torch.int32)  # G=5 not mult of 8
    packed = ref.pack_zeros(z, 4)
    assert torch.equal(ref.unpack_zeros(packed, 4, 5), z)


def test_dq_linear_matches_dense():
    torch.manual_seed(1)
    w = torch.randn(32, 128)
    codes, s, z = ref.quant_groupwise(w, 4, 64, False)
    qw = ref.pack_codes(codes, 4)
    qz = ref.pack_zeros(z, 4)
    x = torch.randn(4, 128)
    y = ref.dq_linear(x, qw, s, qz, 4, 64)
    wdq = ref.dequant_groupwise(codes, s, z, 64)
    assert torch.allclose(y, x @ wdq.t(), atol=1e-4)


def test_hessian_accum():
    x = torch.randn(64, 32)
    h = torch.zeros(32, 32)
    ref.hessian_accum(h, x)
    assert torch.allclose(h, 2 * x.t() @ x, atol=1e-4)
    assert torch.allclose(h, h.t(), atol=1e-5)  # symmetric


def test_rmsnorm():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    y = ref.rmsnorm(x, w, 1e-5)
    expect = x / (x.pow(2).mean(-1, keepdim=True) + 1e-5).sqrt() * w
    assert torch.allclose(y, expect, atol=1e-5)


def test_rmsnorm_fused_residual():
    x, r = torch.randn(4, 64), torch.randn(4, 64)
    w = torch.randn(64)
    assert torch.allclose(ref.rmsnorm(x, w, 1e-5, residual=r),
                          ref.rmsnorm(x + r, w, 1e-5), atol=1e-5)


def test_rope_norm_preserving_and_pos0_identity():
    x = torch.randn(2, 3, 4, 64)  # [b, s, h, d]
    pos = torch.arange(3)
    y = ref.apply_rope(x, pos)
    # position 0 is the identity rotation
    assert torch.allclose(y[:, 0], x[:, 0], atol=1e-6)
    # rotation preserves pairwise norms
    def pair_norm(t):
        d = t.shape[-1] // 2
        return t[..., :d] ** 2 + t[..., d:] ** 2
    assert torch.allclose(pair_norm(y), pair_norm(x), atol=1e-4)


def test_rope_dot_product_depends_on_relative_pos():
    d = 64
    q = torch.randn(1, 1, 1, d)
    k = torch.randn(1, 1, 1, d)
    def dot(p_q, p_k):
        qr = ref.apply_rope(q, torch.tensor([p_q]))
        kr = ref.apply_rope(k, torch.tensor([p_k]))
        return float((qr * kr).sum())
    assert abs(dot(5, 3) - dot(12, 10)) < 1e-3  # same relative distance


def test_attn_decode_matches_prefill_last_row():
    b, hq, hkv, t, d = 2, 8, 2, 10, 32
    q_all = torch.randn(b, hq, t, d)
    k = torch.randn(b, hkv, t, d)
    v = torch.randn(b, hkv, t, d)
    full = ref.attn_prefill(q_all, k, v, causal=True)
    kc = torch.zeros(b, hkv, 16, d)
    vc = torch.zeros(b, hkv, 16, d)
    kc[:, :, :t] = k
    vc[:, :, :t] = v
    dec = ref.attn_decode(q_all[:, :, -1:], kc, vc, seq_len=t)
    assert torch.allclose(dec, full[:, :, -1:], atol=1e-5)


def test_silu_mul():
    g, u = torch.randn(4, 16), torch.randn(4, 16)
    assert torch.allclose(ref.silu_mul(g, u),
                          torch.nn.functional.silu(g) * u, atol=1e-6)


def test_kv_append():
    kc = torch.zeros(1, 2, 8, 4)
    vc = torch.zeros(1, 2, 8, 4)
    kn = torch.randn(1, 2, 3, 4)
    vn = torch.randn(1, 2, 3, 4)
    ref.kv_append(kc, vc, kn, vn, pos=2)
    assert torch.equal(kc[:, :, 2:5], kn)
    assert kc[:, :, :2].abs().sum() == 0


def test_grouped_dq_linear():
    E, n, k, t = 3, 16, 64, 10
    ws = torch.randn(E, n, k)
    qws, ss, zs = [], [], []
    for e in range(E):
        c, s, z = ref.quant_groupwise(ws[e], 4, 32, False)
        qws.append(ref.pack_codes(c, 4)); ss.append(s); zs.append(ref.pack_zeros(z, 4))
    qw, s, z = torch.stack(qws), torch.stack(ss), torch.stack(zs)
    x = torch.randn(t, k)
    eid = torch.randint(0, E, (t,))
    y = ref.grouped_dq_linear(x, eid, qw, s, z, 4, 32)
    for i in range(t):
        yi = ref.dq_linear(x[i:i+1], qw[eid[i]], s[eid[i]], z[eid[i]], 4, 32)
        assert torch.allclose(y[i:i+1], yi, atol=1e-5)


def test_nancheck_guard(monkeypatch):
    """LFMQ_NANCHECK flag wiring (SURVEY §5.2 numeric guards)."""
    import importlib
    import lfmq.ops.functional as F
    monkeypatch.setattr(F, "_NANCHECK", True)
    t = torch.tensor([1.0, float("nan")])
    with pytest.raises(FloatingPointError):
        F._guard(t, "test")
    F._guard(torch.ones(3), "ok")
