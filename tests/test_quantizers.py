"""RTN / GPTQ / AWQ quantizer math tests against naive oracles."""
import pytest
import torch
import torch.nn as nn

from lfmq.config import QuantConfig
from lfmq.models.qlinear import QLinear
from lfmq.ops import ref
from lfmq.quant.gptq import gptq_quantize_layer
from lfmq.quant.awq import search_scale, apply_scale, _pseudo_quant


def _naive_gptq(w, h, cfg):
    """Per-column (block_size=1) GPTQ — the independent oracle."""
    n, k = w.shape
    g = cfg.group_size
    w = w.float().clone()
    h = h.float().clone()
    damp = cfg.damp_percent * torch.diag(h).mean()
    h += torch.eye(k) * damp
    hinv = torch.cholesky_inverse(torch.linalg.cholesky(h))
    U = torch.linalg.cholesky(hinv, upper=True)
    codes = torch.zeros(n, k, dtype=torch.int32)
    scales = torch.zeros(n, k // g)
    zeros = torch.zeros(n, k // g, dtype=torch.int32)
    qmax = (1 << cfg.bits) - 1
    for j in range(k):
        if j % g == 0:
            _, s_, z_ = ref.quant_groupwise(w[:, j:j + g], cfg.bits, g, cfg.sym)
            scales[:, j // g] = s_[:, 0]
            zeros[:, j // g] = z_[:, 0]
        s = scales[:, j // g]
        z = zeros[:, j // g]
        q = torch.clamp(torch.round(w[:, j] / s) + z, 0, qmax)
        codes[:, j] = q.to(torch.int32)
        dq = (q - z) * s
        e = (w[:, j] - dq) / U[j, j]
        if j + 1 < k:
            w[:, j + 1:] -= torch.outer(e, U[j, j + 1:])
    return codes, scales, zeros


@pytest.mark.parametrize("bits", [4, 8])
def test_gptq_blocked_equals_naive(bits):
    torch.manual_seed(3)
    n, k = 24, 128
    w = torch.randn(n, k)
    x = torch.randn(300, k)
    h = 2 * x.t() @ x
    cfg = QuantConfig(scheme="gptq", bits=bits, group_size=32, block_size=32)
    c1, s1, z1 = _naive_gptq(w, h, cfg)
    c2, s2, z2, gi = gptq_quantize_layer(w, h, cfg)
    assert gi is None
    assert torch.allclose(s1, s2, atol=1e-5)
    assert torch.equal(z1, z2)
    # rounding at block boundaries can differ by float assoc: allow tiny diff
    assert (c1 != c2).float().mean() < 0.01


def test_gptq_beats_rtn_on_correlated_inputs():
    torch.manual_seed(0)
    n, k = 32, 256
    w = torch.randn(n, k)
    # correlated calibration inputs (low-rank + noise)
    base = torch.randn(2000, 8) @ torch.randn(8, k)
    x = base + 0.1 * torch.randn(2000, k)
    h = 2 * x.t() @ x
    cfg = QuantConfig(scheme="gptq", bits=4, group_size=64)
    c, s, z, _ = gptq_quantize_layer(w, h, cfg)
    w_gptq = ref.dequant_groupwise(c, s, z, 64)
    c0, s0, z0 = ref.quant_groupwise(w, 4, 64, False)
    w_rtn = ref.dequant_groupwise(c0, s0, z0, 64)
    err_gptq = ((x @ (w - w_gptq).t()) ** 2).mean()
    err_rtn = ((x @ (w - w_rtn).t()) ** 2).mean()
    assert err_gptq < err_rtn * 0.9  # GPTQ must clearly win on task loss


def test_gptq_act_order_perm():
    torch.manual_seed(1)
    n, k, g = 8, 64, 16
    w = torch.randn(n, k)
    x = torch.randn(500, k) * torch.linspace(0.1, 3.0, k)
    h = 2 * x.t() @ x
    cfg = QuantConfig(scheme="gptq", bits=4, group_size=g, act_order=True)
    c, s, z, perm = gptq_quantize_layer(w, h, cfg)
    assert perm is not None and perm.shape == (k,)
    # perm is a permutation of 0..k-1, highest-|H| diagonal first
    assert torch.equal(perm.sort().values, torch.arange(k))
    # codes are in QUANT order (contiguous groups): column j of the
    # dequant corresponds to ORIGINAL input perm[j]
    gi = torch.arange(k) // g
    dq = (c.float() - z.float()[:, gi]) * s[:, gi]
    assert ((w[:, perm] - dq).abs().mean()) < 0.2


def test_gptq_dead_column():
    n, k = 4, 32
    w = torch.randn(n, k)
    x = torch.randn(100, k)
    x[:, 5] = 0  # dead input channel
    h = 2 * x.t() @ x
    cfg = QuantConfig(scheme="gptq", bits=4, group_size=16)
    c, s, z, _ = gptq_quantize_layer(w, h, cfg)  # must not raise/NaN
    assert torch.isfinite(s).all()


def test_awq_scale_fold_invariance():
    """(s·W)(x/s) == Wx — folding must be numerically invisible pre-quant."""
    torch.manual_seed(0)
    k, n = 32, 16
    lin = nn.Linear(k, n, bias=False, dtype=torch.float64)
    prev = nn.Linear(8, k, bias=True, dtype=torch.float64)
    x0 = torch.randn(10, 8, dtype=torch.float64)
    y_before = lin(torch.relu(prev(x0)))
    s = torch.rand(k, dtype=torch.float64) + 0.5

    class Blk(nn.Module):
        def __init__(self):
            super().__init__()
            self.prev = prev
            self.lin = lin
    blk = Blk()
    apply_scale(blk, ("linear_rows", "prev", (0, None)), [lin], s)
    y_after = lin(torch.relu(prev(x0)))
    assert torch.allclose(y_before, y_after, atol=1e-10)


def test_awq_search_improves_quant_error():
    torch.manual_seed(0)
    k, n = 128, 64
    w = torch.randn(n, k)
    # salient channels: a few input channels with large activations
    xscale = torch.ones(k)
    xscale[:8] = 20.0
    x = torch.randn(1000, k) * xscale
    cfg = QuantConfig(scheme="awq", bits=4, group_size=64)
    s = search_scale(x, [w], cfg)
    err_s = (((w - _pseudo_quant(w * s, 4, 64, False) / s)) ** 2 * x.pow(2).mean(0)).sum()
    err_0 = (((w - _pseudo_quant(w, 4, 64, False))) ** 2 * x.pow(2).mean(0)).sum()
    assert err_s < err_0


def test_qlinear_rtn_forward_matches_ref():
    lin = nn.Linear(128, 32, bias=True)
    cfg = QuantConfig(scheme="rtn", bits=4, group_size=64)
    q = QLinear.from_rtn(lin, cfg)
    x = torch.randn(4, 128)
    y = q(x)
    wdq = q.dequant()
    expect = x @ wdq.t() + lin.bias.float()
    assert torch.allclose(y.float(), expect, atol=2e-2, rtol=1e-2)


def test_qlinear_w8():
    lin = nn.Linear(64, 16, bias=False)
    cfg = QuantConfig(scheme="rtn", bits=8, group_size=32)
    q = QLinear.from_rtn(lin, cfg)
    x = torch.randn(4, 64)
    rel = (q(x).float() - lin(x).float()).abs().mean() / lin(x).float().abs().mean()
    assert rel < 0.05


def test_gptq_resume(tmp_path):
    """Per-block resume: a second run with resume_dir skips finished
    blocks and yields identical weights."""
    import lfmq
    from lfmq.models.llama import LlamaConfig
    from lfmq.quant.gptq import GPTQQuantizer
    from lfmq.quant.calib import CalibStream
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="gptq", bits=4, group_size=64,
                            calib_n_seqs=2, calib_seq_len=32)
    rd = str(tmp_path / "resume")
    calib = CalibStream(cfg.vocab_size, 32, 2, qcfg.calib_seed)
    m1 = lfmq.build_model(cfg, seed=6)
    m1 = GPTQQuantizer(qcfg).run(m1, calib=calib, resume_dir=rd)
    # second run from scratch model with the SAME resume dir: all blocks
    # restored, result identical
    calib2 = CalibStream(cfg.vocab_size, 32, 2, qcfg.calib_seed)
    m2 = lfmq.build_model(cfg, seed=6)
    m2 = GPTQQuantizer(qcfg).run(m2, calib=calib2, resume_dir=rd)
    for (k1, v1), (k2, v2) in zip(m1.state_dict().items(),
                                  m2.state_dict().items()):
        assert k1 == k2
        if "layers." in k1:
            assert torch.equal(v1, v2), k1


def test_awq_clip_search_reduces_error():
    from lfmq.quant.awq import search_clip, apply_clip
    torch.manual_seed(2)
    k, n = 128, 32
    w = torch.randn(n, k)
    w[:, 0] *= 30.0  # an outlier column stretches every row's range
    x = torch.randn(500, k)
    x[:, 0] *= 0.01  # ...but that channel barely matters to the output
    cfg = QuantConfig(scheme="awq", bits=4, group_size=-1)
    ratios = search_clip(x, w, cfg)
    assert (ratios < 1.0).any()  # clipping must engage
    xw = x.pow(2).mean(0)
    err_clip = 0.0
    lin = nn.Linear(k, n, bias=False)
    lin.weight.data = w.clone()
    apply_clip(lin, ratios)
    from lfmq.quant.awq import _pseudo_quant
    err_c = ((w - _pseudo_quant(lin.weight.float(), 4, -1, False)).pow(2) * xw).sum()
    err_0 = ((w - _pseudo_quant(w, 4, -1, False)).pow(2) * xw).sum()
    assert err_c < err_0


def test_awq_clip_e2e():
    import lfmq
    from lfmq.models.llama import LlamaConfig
    from lfmq.quant.calib import CalibStream
    cfg = LlamaConfig.tiny()
    qcfg = QuantConfig(scheme="awq", bits=4, group_size=64, awq_clip=True,
                       awq_grid=6, calib_n_seqs=2, calib_seq_len=32)
    calib = CalibStream(cfg.vocab_size, 32, 2, qcfg.calib_seed)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=15), qcfg, calib=calib)
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    assert torch.isfinite(qm(ids, qm.make_cache(1, 16)).float()).all()


def test_degenerate_weights_quantize_finite():
    """All-zero and all-constant weight rows must quantize (scale==0
    guard) and run finite; GPTQ must survive dead input columns (zeroed
    Hessian diagonal gets the dead-column patch)."""
    import lfmq
    from lfmq.models.llama import LlamaConfig
    cfg = LlamaConfig.tiny()
    m = lfmq.build_model(cfg, seed=0)
    with torch.no_grad():
        m.layers[0].self_attn.qkv_proj.weight.zero_()
        m.layers[1].mlp.down_proj.weight.fill_(0.25)
    qm = lfmq.quantize(m, lfmq.QuantConfig(scheme="rtn", bits=4,
                                           group_size=32))
    ids = torch.arange(8).reshape(1, 8)
    assert torch.isfinite(qm(ids, qm.make_cache(1, 16)).float()).all()

    m2 = lfmq.build_model(cfg, seed=1)
    with torch.no_grad():
        m2.embed_tokens.weight[:, :8].zero_()
    qm2 = lfmq.quantize(m2, lfmq.QuantConfig(
        scheme="gptq", bits=4, group_size=32, calib_n_seqs=1,
        calib_seq_len=32))
    assert torch.isfinite(qm2(ids, qm2.make_cache(1, 16)).float()).all()
