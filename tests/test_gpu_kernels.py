"""GPU parity tests: every HIP kernel vs its CPU reference twin.

All run in ONE gpurun call (`pytest -m gpu`).  Tolerances: exact for
pack/unpack/kv_append; bf16-accumulation tolerances vs the fp32 reference
for the compute kernels.  Inputs are random and ASYMMETRIC (guide §3:
symmetric operands mask transposed-output bugs).
"""
import math

import pytest
import torch

from lfmq.ops import ref

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def ext():
    from lfmq.ops import hip_ext
    return hip_ext()


def rel_err(a, b):
    a, b = a.float().cpu(), b.float().cpu()
    return float((a - b).abs().max() / (b.abs().max() + 1e-8))


# ---------------------------------------------------------------- pack
@pytest.mark.parametrize("bits", [4, 8])
def test_pack_unpack_gpu(bits):
    codes = torch.randint(0, 1 << bits, (64, 256), dtype=torch.int32)
    gpu = ext().pack(codes.to(DEV), bits)
    assert torch.equal(gpu.cpu(), ref.pack_codes(codes, bits))
    back = ext().unpack(gpu, bits, 256)
    assert torch.equal(back.cpu(), codes)


# ------------------------------------------------------------- mfma map
def test_mfma_layout_probe_via_hessian():
    """hessian_accum is a pure MFMA kernel: if the assumed 16x16x32
    fragment layout (A/B: lane&15 row/col, 8 contiguous k at (lane>>4)*8;
    C/D: col=lane&15, row=(lane>>4)*4+reg) is wrong, this fails."""
    t, k = 64, 128
    x = (torch.randn(t, k) * torch.linspace(0.5, 2.0, k)).to(torch.bfloat16)
    h = torch.zeros(k, k, dtype=torch.float32)
    hg = h.clone().to(DEV)
    ext().hessian_accum(hg, x.to(DEV))
    ref.hessian_accum(h, x.float())
    assert rel_err(hg, h) < 2e-2


@pytest.mark.parametrize("t,k", [(300, 256), (64, 4096), (1000, 512)])
def test_hessian_accum(t, k):
    x = torch.randn(t, k).to(torch.bfloat16)
    h0 = torch.randn(k, k).float()
    hg = h0.clone().to(DEV)
    ext().hessian_accum(hg, x.to(DEV))
    hc = h0.clone()
    ref.hessian_accum(hc, x.float())
    assert rel_err(hg, hc) < 2e-2


# ---------------------------------------------------------------- gemv
@pytest.mark.parametrize("bits", [4, 8])
@pytest.mark.parametrize("m", [1, 2, 4, 8, 16])
@pytest.mark.parametrize("n,k,g", [(256, 2048, 128), (37, 512, 64),
                                   (129, 4096, 128), (64, 96, 32)])
def test_dq_gemv(bits, m, n, k, g):
    torch.manual_seed(bits * 1000 + m * 100 + n)
    w = torch.randn(n, k)
    codes, scales, zeros = ref.quant_groupwise(w, bits, g, sym=False)
    qw = ref.pack_codes(codes, bits)
    qz = ref.pack_zeros(zeros, bits)
    x = torch.randn(m, k).to(torch.bfloat16)
    y_ref = ref.dq_linear(x.float(), qw, scales.to(torch.bfloat16).float(),
                          qz, bits, g)
    y = ext().dq_linear(x.to(DEV), qw.to(DEV), scales.to(torch.bfloat16).to(DEV),
                        qz.to(DEV), bits, g, None, None, -1.0, 0)
    assert y.shape == (m, n)
    assert rel_err(y, y_ref) < 2e-2, f"rel={rel_err(y, y_ref)}"


def test_dq_gemv_bias():
    n, k, g = 128, 256, 64
    w = torch.randn(n, k)
    codes, scales, zeros = ref.quant_groupwise(w, 4, g, False)
    qw, qz = ref.pack_codes(codes, 4), ref.pack_zeros(zeros, 4)
    bias = torch.randn(n).to(torch.bfloat16)
    x = torch.randn(2, k).to(torch.bfloat16)
    y_ref = ref.dq_linear(x.float(), qw, scales.to(torch.bfloat16).float(), qz,
                          4, g, bias.float())
    y = ext().dq_linear(x.to(DEV), qw.to(DEV), scales.to(torch.bfloat16).to(DEV),
                        qz.to(DEV), 4, g, bias.to(DEV), None, -1.0, 0)
    assert rel_err(y, y_ref) < 2e-2


# ---------------------------------------------------------------- gemm
@pytest.mark.parametrize("bits", [4, 8])
@pytest.mark.parametrize("m,n,k,g", [(64, 256, 512, 128), (200, 300, 256, 64),
                                     (128, 128, 128, 64), (513, 100, 1024, 128)])
def test_dq_gemm(bits, m, n, k, g):
    torch.manual_seed(m + n)
    w = torch.randn(n, k)
    codes, scales, zeros = ref.quant_groupwise(w, bits, g, False)
    qw, qz = ref.pack_codes(codes, bits), ref.pack_zeros(zeros, bits)
    x = torch.randn(m, k).to(torch.bfloat16)
    y_ref = ref.dq_linear(x.float(), qw, scales.to(torch.bfloat16).float(),
                          qz, bits, g)
    y = ext().dq_linear(x.to(DEV), qw.to(DEV), scales.to(torch.bfloat16).to(DEV),
                        qz.to(DEV), bits, g, None, None, -1.0, 0)
    assert rel_err(y, y_ref) < 3e-2, f"rel={rel_err(y, y_ref)}"


# ------------------------------------------------------------- rmsnorm
@pytest.mark.parametrize("rows,hidden", [(1, 4096), (17, 256), (64, 8192)])
def test_rmsnorm(rows, hidden):
    x = torch.randn(rows, hidden).to(torch.bfloat16)
    w = torch.randn(hidden).to(torch.bfloat16)
    y = ext().rmsnorm(x.to(DEV), w.to(DEV), 1e-5, None)
    y_ref = ref.rmsnorm(x, w, 1e-5)
    assert rel_err(y, y_ref) < 2e-2


def test_rmsnorm_residual():
    x = torch.randn(9, 512).to(torch.bfloat16)
    r = torch.randn(9, 512).to(torch.bfloat16)
    w = torch.randn(512).to(torch.bfloat16)
    y, nr = ext().rmsnorm_residual(x.to(DEV), r.to(DEV), w.to(DEV), 1e-5)
    y_ref = ref.rmsnorm(x, w, 1e-5, residual=r)
    nr_ref = ref.residual_add(x, r)
    assert rel_err(y, y_ref) < 2e-2
    assert rel_err(nr, nr_ref) < 1e-2


# ----------------------------------------------------------------- rope
@pytest.mark.parametrize("s,pos0", [(1, 77), (16, 0)])
def test_rope(s, pos0):
    b, hq, hkv, d = 2, 8, 2, 128
    q = torch.randn(b, s, hq, d).to(torch.bfloat16)
    k = torch.randn(b, s, hkv, d).to(torch.bfloat16)
    pos = torch.arange(pos0, pos0 + s)
    qg, kg = q.to(DEV), k.to(DEV)
    inv = ref.rope_freqs(d, 500000.0).to(DEV)
    ext().rope(qg, kg, pos.to(torch.int32).to(DEV), inv)
    q_ref = ref.apply_rope(q.float(), pos)
    k_ref = ref.apply_rope(k.float(), pos)
    assert rel_err(qg, q_ref) < 2e-2
    assert rel_err(kg, k_ref) < 2e-2


# ---------------------------------------------------------- elementwise
def test_silu_mul():
    g = torch.randn(33, 1024).to(torch.bfloat16)
    u = torch.randn(33, 1024).to(torch.bfloat16)
    y = ext().silu_mul(g.to(DEV), u.to(DEV))
    assert rel_err(y, ref.silu_mul(g, u)) < 2e-2


def test_silu_mul_fused():
    gu = torch.randn(8, 512).to(torch.bfloat16)
    y = ext().silu_mul_fused(gu.to(DEV))
    g, u = gu.split([256, 256], dim=-1)
    assert rel_err(y, ref.silu_mul(g.contiguous(), u.contiguous())) < 2e-2


def test_residual_add():
    a = torch.randn(7, 300).to(torch.bfloat16)
    b = torch.randn(7, 300).to(torch.bfloat16)
    y = ext().residual_add(a.to(DEV), b.to(DEV))
    assert rel_err(y, ref.residual_add(a, b)) < 1e-2


def test_kv_append():
    b, s, hkv, d, tmax = 2, 3, 4, 128, 64
    kc = torch.zeros(b, hkv, tmax, d).to(torch.bfloat16).to(DEV)
    vc = torch.zeros_like(kc)
    kn = torch.randn(b, s, hkv, d).to(torch.bfloat16)
    vn = torch.randn(b, s, hkv, d).to(torch.bfloat16)
    ext().kv_append(kc, vc, kn.to(DEV), vn.to(DEV), 5)
    assert torch.equal(kc[:, :, 5:8].cpu(), kn.transpose(1, 2))
    assert torch.equal(vc[:, :, 5:8].cpu(), vn.transpose(1, 2))
    assert kc[:, :, :5].abs().sum().item() == 0


# ---------------------------------------------------------- attn decode
@pytest.mark.parametrize("b,T,d", [(1, 17, 128), (2, 256, 128), (1, 700, 128),
                                   (4, 64, 64), (1, 2048, 128)])
def test_attn_decode(b, T, d):
    hq, hkv = 8, 2
    tmax = T + 13
    torch.manual_seed(T)
    q = torch.randn(b, hq, 1, d).to(torch.bfloat16)
    kc = torch.randn(b, hkv, tmax, d).to(torch.bfloat16)
    vc = torch.randn(b, hkv, tmax, d).to(torch.bfloat16)
    scale = 1.0 / math.sqrt(d)
    o = ext().attn_decode(q.to(DEV), kc.to(DEV), vc.to(DEV), T, scale, None)
    o_ref = ref.attn_decode(q.float(), kc.float(), vc.float(), T, scale)
    assert rel_err(o, o_ref) < 2e-2, f"rel={rel_err(o, o_ref)}"


# ------------------------------------------------- fused decode kernels
def test_dq_gemv_gateup():
    torch.manual_seed(7)
    inter, k, g = 256, 512, 128
    w = torch.randn(2 * inter, k)
    codes, scales, zeros = ref.quant_groupwise(w, 4, g, False)
    qw, qz = ref.pack_codes(codes, 4), ref.pack_zeros(zeros, 4)
    x = torch.randn(1, k).to(torch.bfloat16)
    gu = ref.dq_linear(x.float(), qw, scales.to(torch.bfloat16).float(), qz, 4, g)
    y_ref = ref.silu_mul(gu[:, :inter].contiguous(), gu[:, inter:].contiguous())
    y = ext().dq_gemv_gateup(x.to(DEV), qw.to(DEV),
                             scales.to(torch.bfloat16).to(DEV), qz.to(DEV),
                             4, g, -1.0, 0)
    assert y.shape == (1, inter)
    assert rel_err(y, y_ref) < 2e-2


def test_qkv_rope_append():
    b, s, hq, hkv, d, tmax, pos0 = 2, 3, 8, 2, 128, 32, 5
    torch.manual_seed(9)
    qkv = torch.randn(b, s, (hq + 2 * hkv) * d).to(torch.bfloat16)
    kc = torch.zeros(b, hkv, tmax, d).to(torch.bfloat16).to(DEV)
    vc = torch.zeros_like(kc)
    pos = torch.arange(pos0, pos0 + s)
    inv = ref.rope_freqs(d, 500000.0).to(DEV)
    q = ext().qkv_rope_append(qkv.to(DEV), kc, vc, pos.to(torch.int32).to(DEV),
                              inv, hq, pos0, None)
    # reference
    qr, kr, vr = qkv.split([hq * d, hkv * d, hkv * d], dim=-1)
    q_ref = ref.apply_rope(qr.reshape(b, s, hq, d).float(), pos)
    k_ref = ref.apply_rope(kr.reshape(b, s, hkv, d).float(), pos)
    v_ref = vr.reshape(b, s, hkv, d)
    assert rel_err(q, q_ref) < 2e-2
    assert rel_err(kc[:, :, pos0:pos0 + s], k_ref.transpose(1, 2)) < 2e-2
    assert torch.equal(vc[:, :, pos0:pos0 + s].cpu(), v_ref.transpose(1, 2))
    assert kc[:, :, :pos0].abs().sum().item() == 0


def test_greedy_sample():
    torch.manual_seed(11)
    for b, vocab in [(1, 128256), (4, 512), (2, 50272)]:
        logits = torch.randn(b, vocab).to(torch.bfloat16)
        got = ext().greedy_sample(logits.to(DEV))
        want = logits.float().argmax(dim=-1)
        assert torch.equal(got.cpu(), want), (b, vocab)



def test_dq_gemv_rms_fused():
    """GEMV with fused RMSNorm input vs rmsnorm+gemv reference."""
    torch.manual_seed(17)
    n, k, g, eps = 64, 4096, 128, 1e-5
    w = torch.randn(n, k)
    codes, scales, zeros = ref.quant_groupwise(w, 4, g, False)
    qw, qz = ref.pack_codes(codes, 4), ref.pack_zeros(zeros, 4)
    x = torch.randn(1, k).to(torch.bfloat16)
    xn = ref.rmsnorm(x, torch.ones(k).to(torch.bfloat16), eps)
    y_ref = ref.dq_linear(xn.float(), qw, scales.to(torch.bfloat16).float(), qz, 4, g)
    y = ext().dq_linear(x.to(DEV), qw.to(DEV), scales.to(torch.bfloat16).to(DEV),
                        qz.to(DEV), 4, g, None, None, eps, 0)
    assert rel_err(y, y_ref) < 2e-2, f"rel={rel_err(y, y_ref)}"


def test_dq_gemv_resid_fused():
    torch.manual_seed(19)
    n, k, g = 128, 2048, 128
    w = torch.randn(n, k)
    codes, scales, zeros = ref.quant_groupwise(w, 4, g, False)
    qw, qz = ref.pack_codes(codes, 4), ref.pack_zeros(zeros, 4)
    x = torch.randn(2, k).to(torch.bfloat16)
    r = torch.randn(2, n).to(torch.bfloat16)
    y_ref = ref.dq_linear(x.float(), qw, scales.to(torch.bfloat16).float(),
                          qz, 4, g) + r.float()
    y = ext().dq_linear(x.to(DEV), qw.to(DEV), scales.to(torch.bfloat16).to(DEV),
                        qz.to(DEV), 4, g, None, r.to(DEV), -1.0, 0)
    assert rel_err(y, y_ref) < 2e-2


def test_grouped_dq_kernels():
    """K5 grouped MoE kernels vs the CPU grouped reference."""
    torch.manual_seed(23)
    E, n, k, g = 4, 64, 256, 64
    inter = n // 2
    ws = torch.randn(E, n, k)
    qws, scs, qzs = [], [], []
    for e in range(E):
        c, s, z = ref.quant_groupwise(ws[e], 4, g, False)
        qws.append(ref.pack_codes(c, 4)); scs.append(s.to(torch.bfloat16)); qzs.append(ref.pack_zeros(z, 4))
    qw, sc, qz = torch.stack(qws).to(DEV), torch.stack(scs).to(DEV), torch.stack(qzs).to(DEV)
    t, topk = 6, 2
    x = torch.randn(t, k).to(torch.bfloat16)
    ids = torch.randint(0, E, (t * topk,))
    order = torch.argsort(ids)
    src_tok = (order // topk).to(torch.int32)
    sorted_ids = ids[order]
    counts = torch.bincount(sorted_ids, minlength=E)
    offsets = torch.zeros(E + 1, dtype=torch.int32)
    offsets[1:] = counts.cumsum(0)
    import lfmq_hip
    h = lfmq_hip.grouped_dq_gateup(x.to(DEV), src_tok.to(DEV), offsets.to(DEV),
                                   qw, sc, qz, 4, g, None)
    # reference: per slot, expert gateup + silu
    h_ref = torch.zeros(t * topk, inter)
    for s_ in range(t * topk):
        e = int(sorted_ids[s_])
        gu = ref.dq_linear(x[src_tok[s_]].float().unsqueeze(0), qws[e],
                           scs[e].float(), qzs[e], 4, g)
        h_ref[s_] = ref.silu_mul(gu[:, :inter].contiguous(), gu[:, inter:].contiguous())
    assert rel_err(h, h_ref) < 2e-2

    # w2 pass (identity src_tok)
    w2s = torch.randn(E, k, inter)
    qw2, sc2, qz2 = [], [], []
    for e in range(E):
        c, s, z = ref.quant_groupwise(w2s[e], 4, 32, False)
        qw2.append(ref.pack_codes(c, 4)); sc2.append(s.to(torch.bfloat16)); qz2.append(ref.pack_zeros(z, 4))
    qw2t, sc2t, qz2t = torch.stack(qw2).to(DEV), torch.stack(sc2).to(DEV), torch.stack(qz2).to(DEV)
    hh = h_ref.to(torch.bfloat16)
    y = lfmq_hip.grouped_dq_linear(hh.to(DEV), offsets.to(DEV), qw2t, sc2t, qz2t, 4, 32)
    y_ref = torch.zeros(t * topk, k)
    for s_ in range(t * topk):
        e = int(sorted_ids[s_])
        y_ref[s_] = ref.dq_linear(hh[s_].float().unsqueeze(0), qw2[e],
                                  sc2[e].float(), qz2[e], 4, 32)
    assert rel_err(y, y_ref) < 2e-2


def test_attn_decode_fused_kernel():
    """Fused rope+append+attention vs the composed reference path."""
    import lfmq_hip
    b, hq, hkv, d, tmax, T = 2, 8, 2, 128, 96, 41
    theta = 500000.0
    torch.manual_seed(29)
    qkv = torch.randn(b, (hq + 2 * hkv) * d).to(torch.bfloat16)
    kc = torch.randn(b, hkv, tmax, d).to(torch.bfloat16)
    vc = torch.randn(b, hkv, tmax, d).to(torch.bfloat16)
    pos = torch.tensor([T - 1])
    # reference: rope q/k, append, attn over T
    qr, kr, vr = qkv.split([hq * d, hkv * d, hkv * d], dim=-1)
    q_ref = ref.apply_rope(qr.reshape(b, 1, hq, d).float(), pos, theta)
    k_ref = ref.apply_rope(kr.reshape(b, 1, hkv, d).float(), pos, theta)
    kc_ref, vc_ref = kc.float().clone(), vc.float().clone()
    kc_ref[:, :, T - 1] = k_ref[:, 0]
    vc_ref[:, :, T - 1] = vr.reshape(b, 1, hkv, d)[:, 0].float()
    import math as _m
    o_ref = ref.attn_decode(q_ref.transpose(1, 2), kc_ref, vc_ref, T,
                            1.0 / _m.sqrt(d))
    # fused kernel
    kcg, vcg = kc.to(DEV), vc.to(DEV)
    inv = ref.rope_freqs(d, theta).to(DEV)
    tk = torch.zeros(b * hq, dtype=torch.int32, device=DEV)
    o = lfmq_hip.attn_decode_fused(qkv.to(DEV), kcg, vcg, T,
                                   1.0 / _m.sqrt(d), inv, hq, None, tk)
    assert rel_err(o.reshape(b, hq, 1, d), o_ref) < 2e-2
    # cache rows written
    assert rel_err(kcg[:, :, T - 1], k_ref[:, 0].transpose(0, 1).transpose(0, 1)) < 2e-2
    assert rel_err(vcg[:, :, T - 1],
                   vr.reshape(b, hkv, d)) < 1e-6


def test_attn_decode_fused_multichunk():
    import lfmq_hip, math as _m
    b, hq, hkv, d, tmax, T = 1, 8, 2, 128, 600, 500
    torch.manual_seed(37)
    qkv = torch.randn(b, (hq + 2 * hkv) * d).to(torch.bfloat16)
    kc = torch.randn(b, hkv, tmax, d).to(torch.bfloat16)
    vc = torch.randn(b, hkv, tmax, d).to(torch.bfloat16)
    pos = torch.tensor([T - 1])
    qr, kr, vr = qkv.split([hq * d, hkv * d, hkv * d], dim=-1)
    q_ref = ref.apply_rope(qr.reshape(b, 1, hq, d).float(), pos)
    k_ref = ref.apply_rope(kr.reshape(b, 1, hkv, d).float(), pos)
    kc_ref, vc_ref = kc.float().clone(), vc.float().clone()
    kc_ref[:, :, T - 1] = k_ref[:, 0]
    vc_ref[:, :, T - 1] = vr.reshape(b, 1, hkv, d)[:, 0].float()
    o_ref = ref.attn_decode(q_ref.transpose(1, 2), kc_ref, vc_ref, T,
                            1.0 / _m.sqrt(d))
    inv = ref.rope_freqs(d, 500000.0).to(DEV)
    tk = torch.zeros(b * hq, dtype=torch.int32, device=DEV)
    o = lfmq_hip.attn_decode_fused(qkv.to(DEV), kc.to(DEV), vc.to(DEV), T,
                                   1.0 / _m.sqrt(d), inv, hq, None, tk)
    assert rel_err(o.reshape(b, hq, 1, d), o_ref) < 2e-2
    assert int(tk.abs().sum()) == 0, "winner must reset its ticket"
    # and the separate-combine path (tickets=None) must still match
    o2 = lfmq_hip.attn_decode_fused(qkv.to(DEV), kc.to(DEV), vc.to(DEV), T,
                                    1.0 / _m.sqrt(d), inv, hq, None, None)
    assert rel_err(o2.reshape(b, hq, 1, d), o_ref) < 2e-2


@pytest.mark.parametrize("m", [5, 8, 16])
@pytest.mark.parametrize("n,k,g", [(256, 2048, 128), (100, 512, 64),
                                   (48, 4096, 128)])
def test_dq_gemv_mfma_variant(m, n, k, g):
    """The MFMA-tiled batched GEMV (cols_hint=6) vs reference."""
    torch.manual_seed(m * 10 + n)
    w = torch.randn(n, k)
    codes, scales, zeros = ref.quant_groupwise(w, 4, g, False)
    qw, qz = ref.pack_codes(codes, 4), ref.pack_zeros(zeros, 4)
    x = torch.randn(m, k).to(torch.bfloat16)
    y_ref = ref.dq_linear(x.float(), qw, scales.to(torch.bfloat16).float(),
                          qz, 4, g)
    y = ext().dq_linear(x.to(DEV), qw.to(DEV), scales.to(torch.bfloat16).to(DEV),
                        qz.to(DEV), 4, g, None, None, -1.0, 6)
    assert rel_err(y, y_ref) < 2e-2, f"rel={rel_err(y, y_ref)}"


# ------------------------------------------------ flash prefill
def _prefill_oracle(q, kc, vc, pos0, T, scale):
    """fp32 attention of queries at positions [pos0, T) against cache
    rows [0, T) with causal mask.  q [b,s,hq,d]; caches [b,hkv,Tmax,d]."""
    b, s, hq, d = q.shape
    hkv = kc.shape[1]
    rep = hq // hkv
    kf = kc[:, :, :T].float().repeat_interleave(rep, dim=1)
    vf = vc[:, :, :T].float().repeat_interleave(rep, dim=1)
    qh = q.float().permute(0, 2, 1, 3)                       # [b,hq,s,d]
    sc = torch.einsum("bhqd,bhtd->bhqt", qh, kf) * scale
    t_idx = torch.arange(T)
    rows = torch.arange(pos0, pos0 + s)
    sc = sc.masked_fill((t_idx[None, :] > rows[:, None])[None, None],
                        float("-inf"))
    o = torch.einsum("bhqt,bhtd->bhqd", torch.softmax(sc, dim=-1), vf)
    return o.permute(0, 2, 1, 3)                             # [b,s,hq,d]


@pytest.mark.parametrize("b,s,pos0", [(1, 128, 0), (1, 96, 0), (2, 64, 0),
                                      (1, 64, 100), (1, 1, 37), (1, 300, 13)])
def test_attn_prefill_flash(b, s, pos0):
    torch.manual_seed(s * 7 + pos0)
    hq, hkv, d = 8, 2, 128
    T = pos0 + s
    tmax = T + 5
    scale = 1.0 / math.sqrt(d)
    q = torch.randn(b, s, hq, d, dtype=torch.bfloat16) * 0.5
    kc = torch.zeros(b, hkv, tmax, d, dtype=torch.bfloat16)
    vc = torch.zeros(b, hkv, tmax, d, dtype=torch.bfloat16)
    kc[:, :, :T] = torch.randn(b, hkv, T, d, dtype=torch.bfloat16) * 0.5
    vc[:, :, :T] = torch.randn(b, hkv, T, d, dtype=torch.bfloat16)
    o_ref = _prefill_oracle(q, kc, vc, pos0, T, scale)
    from lfmq.ops import functional as F
    o = F.attn_prefill_fused(q.to(DEV), kc.to(DEV), vc.to(DEV), pos0, scale)
    assert rel_err(o, o_ref) < 3e-2


def test_moe_route_kernel():
    """One-kernel MoE routing vs the torch reference chain."""
    import lfmq_hip
    torch.manual_seed(11)
    for t, E, k in ((1, 8, 2), (7, 8, 2), (33, 16, 4)):
        logits = torch.randn(t, E, dtype=torch.bfloat16, device=DEV)
        src_tok, slot_w, offsets, active = lfmq_hip.moe_route(logits, k)
        # reference
        w_ref, ids_ref = torch.topk(logits.float(), k, dim=-1)
        w_ref = torch.softmax(w_ref, dim=-1)
        off = offsets.cpu().tolist()
        assert off[0] == 0 and off[-1] == t * k
        counts = torch.bincount(ids_ref.reshape(-1).cpu(), minlength=E)
        for e in range(E):
            assert off[e + 1] - off[e] == int(counts[e]), f"expert {e} count"
            # every slot in the expert's range routes a token that picked e
            for s in range(off[e], off[e + 1]):
                tok = int(src_tok[s])
                j = (ids_ref[tok] == e).nonzero()
                assert j.numel() == 1
                assert abs(float(slot_w[s]) -
                           float(w_ref[tok, j[0, 0]])) < 1e-3
