"""Property-based tests (hypothesis) for the packing/quantization layer
(SURVEY.md §4.1)."""
import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from lfmq.ops import ref


@settings(max_examples=40, deadline=None)
@given(n=st.integers(1, 32), kb=st.integers(1, 16),
       bits=st.sampled_from([4, 8]))
def test_pack_roundtrip_property(n, kb, bits):
    per = 32 // bits
    k = kb * per
    codes = torch.randint(0, 1 << bits, (n, k), dtype=torch.int32)
    assert torch.equal(ref.unpack_codes(ref.pack_codes(codes, bits), bits, k),
                       codes)


@settings(max_examples=30, deadline=None)
@given(n=st.integers(1, 16), groups=st.integers(1, 8),
       g=st.sampled_from([32, 64, 128]), bits=st.sampled_from([4, 8]),
       sym=st.booleans())
def test_quant_error_bound_property(n, groups, g, bits, sym):
    w = torch.randn(n, groups * g)
    codes, s, z = ref.quant_groupwise(w, bits, g, sym)
    dq = ref.dequant_groupwise(codes, s, z, g)
    step = s.repeat_interleave(g, dim=1)
    # asym: within half a step everywhere; sym: clamping can exceed at the
    # negative extreme by at most one step
    slack = 0.5 if not sym else 1.0
    assert ((w - dq).abs() <= step * slack + 1e-5).all()


@settings(max_examples=20, deadline=None)
@given(k=st.sampled_from([64, 96, 128, 256]), g=st.sampled_from([32, 64]))
def test_group_edge_rejection_property(k, g):
    import pytest
    if k % g == 0:
        ref.quant_groupwise(torch.randn(4, k), 4, g, False)
    else:
        with pytest.raises(ValueError):
            ref.quant_groupwise(torch.randn(4, k), 4, g, False)


@settings(max_examples=30, deadline=None)
@given(n=st.integers(1, 16), G=st.integers(1, 20),
       bits=st.sampled_from([4, 8]))
def test_pack_zeros_padding_property(n, G, bits):
    """pack_zeros pads G up to a whole uint32 word; the roundtrip must be
    exact for every non-multiple G (the checkpoint stores zeros packed)."""
    z = torch.randint(0, 1 << bits, (n, G), dtype=torch.int32)
    assert torch.equal(ref.unpack_zeros(ref.pack_zeros(z, bits), bits, G), z)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(1, 8), groups=st.integers(1, 4),
       g=st.sampled_from([32, 64]), bits=st.sampled_from([4, 8]))
def test_sym_zero_is_midpoint_property(n, groups, g, bits):
    """Symmetric mode: the zero code is the fixed range midpoint for every
    group (the fused kernels rely on it folding like any other zero), and
    a symmetric tensor dequantizes with a sign-symmetric error profile."""
    w = torch.randn(n, groups * g)
    _, _, z = ref.quant_groupwise(w, bits, g, sym=True)
    assert (z == (1 << bits) // 2).all()
    codes, s, z = ref.quant_groupwise(torch.cat([w, -w], 0), bits, g, True)
    dq = ref.dequant_groupwise(codes, s, z, g)
    err = (torch.cat([w, -w], 0) - dq)
    # clamping aside, err(w) == -err(-w): same scale, mirrored codes
    assert (err[: n * 1] + err[n:]).abs().max() <= s.max() + 1e-5
