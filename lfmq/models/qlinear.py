"""Quantized linear modules built on the packed layout (lfmq/ops/ref.py doc).

QLinear holds the packed buffers and forwards through the fused
dequant-GEMM (HIP on GPU, exact CPU twin otherwise).  TP variants:
ColumnParallelQLinear (output-dim shard, no collective unless gathered)
and RowParallelQLinear (input-dim shard + all-reduce on RCCL over xGMI).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from .. import dist as lfmq_dist
from ..config import QuantConfig
from ..ops import functional as F
from ..ops import ref


class QLinear(nn.Module):
    """Packed weight-quantized linear: y = x @ dequant(W)^T + b."""

    def __init__(self, in_features: int, out_features: int, bits: int,
                 group_size: int, bias: bool = False, dtype=torch.bfloat16):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.bits = bits
        g = in_features if group_size == -1 else group_size
        if in_features % g != 0:
            raise ValueError(f"in_features={in_features} % group_size={g} != 0")
        self.group_size = g
        per = 32 // bits
        G = in_features // g
        self.register_buffer("qweight", torch.zeros(out_features, in_features // per, dtype=torch.int32))
        self.register_buffer("scales", torch.zeros(out_features, G, dtype=dtype))
        self.register_buffer("qzeros", torch.zeros(out_features, (G + per - 1) // per, dtype=torch.int32))
        if bias:
            self.register_buffer("bias", torch.zeros(out_features, dtype=dtype))
        else:
            self.bias = None

    def extra_repr(self) -> str:
        return (f"in={self.in_features}, out={self.out_features}, "
                f"bits={self.bits}, g={self.group_size}")

    @torch.no_grad()
    def load_quantized(self, codes: torch.Tensor, scales: torch.Tensor,
                       zeros: torch.Tensor, bias: Optional[torch.Tensor] = None):
        """codes int32 [n,k], scales fp32/bf16 [n,G], zeros int32 [n,G].
        On GPU the hand-written HIP pack kernel (csrc/pack.hip) runs; on
        CPU the exact reference twin."""
        if codes.is_cuda:
            from ..ops import hip_ext
            ext = hip_ext()
            self.qweight.copy_(ext.pack(codes.contiguous(), self.bits))
            z = zeros.to(torch.int32).contiguous()
            per = 32 // self.bits
            pad = (-z.shape[1]) % per
            if pad:
                z = torch.cat([z, z.new_zeros(z.shape[0], pad)], dim=1)
            self.qzeros.copy_(ext.pack(z, self.bits))
        else:
            self.qweight.copy_(ref.pack_codes(codes, self.bits))
            self.qzeros.copy_(ref.pack_zeros(zeros.to(torch.int32), self.bits))
        self.scales.copy_(scales.to(self.scales.dtype))
        if bias is not None:
            self.bias.copy_(bias.to(self.scales.dtype))

    @classmethod
    @torch.no_grad()
    def from_rtn(cls, linear: nn.Linear, cfg: QuantConfig) -> "QLinear":
        w = linear.weight.detach().float()
        codes, scales, zeros = ref.quant_groupwise(w, cfg.bits, cfg.group_size, cfg.sym)
        q = cls(linear.in_features, linear.out_features, cfg.bits,
                cfg.group_size if cfg.group_size != -1 else linear.in_features,
                bias=linear.bias is not None).to(w.device)
        q.load_quantized(codes, scales, zeros,
                         linear.bias.detach() if linear.bias is not None else None)
        return q

    @torch.no_grad()
    def dequant(self) -> torch.Tensor:
        """fp32 [n, k] in the ORIGINAL input order — tests/AWQ search only.
        Act-order weights are stored group-contiguous in QUANT order with a
        `perm` gather index: scatter columns back for comparison."""
        w = ref.dequant_weight(self.qweight.cpu(), self.scales.cpu(),
                               self.qzeros.cpu(), self.bits, self.group_size,
                               self.in_features)
        perm = getattr(self, "perm", None)
        if perm is not None:
            out = torch.empty_like(w)
            out[:, perm.cpu()] = w
            return out
        return w

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        perm = getattr(self, "perm", None)
        if perm is not None:
            # act-order layout v2: the packed weight keeps QUANT order
            # (groups contiguous — every fused kernel works unchanged);
            # the input is gathered once instead (x[perm]).  GPU act-order
            # decode was a loud NotImplementedError in round 1 (VERDICT
            # item 5).
            x = x.index_select(-1, perm)
        return F.dq_linear(x, self.qweight, self.scales, self.qzeros,
                           self.bits, self.group_size, self.bias)

    def forward_rms(self, x: torch.Tensor, eps: float) -> torch.Tensor:
        """Decode fast path: rmsnorm(x, 1) fused into the GEMV.

        Act-order: rmsnorm is permutation-invariant (mean(x²) and the
        elementwise scale commute with the gather), so gathering x FIRST
        is exact — the fused kernel then sees quant-order columns."""
        perm = getattr(self, "perm", None)
        if perm is not None:
            x = x.index_select(-1, perm)
        return F.dq_linear(x, self.qweight, self.scales, self.qzeros,
                           self.bits, self.group_size, self.bias,
                           rms_eps=eps)

    def forward_resid(self, x: torch.Tensor, resid: torch.Tensor) -> torch.Tensor:
        """Decode fast path: y = xW^T + resid fused in the epilogue."""
        perm = getattr(self, "perm", None)
        if perm is not None:
            x = x.index_select(-1, perm)
        return F.dq_linear(x, self.qweight, self.scales, self.qzeros,
                           self.bits, self.group_size, self.bias, resid=resid)


class Int8PerTensorLinear(nn.Module):
    """Per-tensor symmetric int8 linear — the OPT-125m CPU decode path
    (BASELINE.json config #1).  int8 codes + one fp32 scale."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.register_buffer("qweight", torch.zeros(out_features, in_features, dtype=torch.int8))
        self.register_buffer("scale", torch.zeros((), dtype=torch.float32))
        if bias:
            self.register_buffer("bias", torch.zeros(out_features, dtype=torch.float32))
        else:
            self.bias = None

    @classmethod
    @torch.no_grad()
    def from_linear(cls, linear: nn.Linear) -> "Int8PerTensorLinear":
        q = cls(linear.in_features, linear.out_features, bias=linear.bias is not None)
        codes, scale = ref.quant_per_tensor_int8(linear.weight)
        q.qweight.copy_(codes)
        q.scale.fill_(scale)
        if linear.bias is not None:
            q.bias.copy_(linear.bias.float())
        return q

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        w = self.qweight.float() * self.scale
        y = x.float() @ w.t()
        if self.bias is not None:
            y = y + self.bias
        return y.to(x.dtype)


class ColumnParallelQLinear(QLinear):
    """Output dimension sharded across TP ranks; this module holds only the
    local shard (out_features here = global_out // world)."""

    def __init__(self, *args, gather_output: bool = False, **kw):
        super().__init__(*args, **kw)
        self.gather_output = gather_output

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = super().forward(x)
        if self.gather_output:
            y = lfmq_dist.all_gather_cat(y, dim=-1)
        return y


class GatherColumnParallelQLinear(QLinear):
    """Act-order replacement for a row-parallel shard: the k-permutation
    crosses k-shard boundaries, so instead the (k-sharded) input is
    all-gathered to full width, the OUTPUT dim is column-sharded, and the
    shard outputs are all-gathered — two all-gathers instead of one
    all-reduce, for the act-order quality mode only."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = lfmq_dist.all_gather_cat(x, dim=-1)
        y = super().forward(x)
        return lfmq_dist.all_gather_cat(y, dim=-1)

    def forward_resid(self, x: torch.Tensor, resid: torch.Tensor) -> torch.Tensor:
        if lfmq_dist.get_world_size() > 1:
            y = self.forward(x)
            from ..ops import hip_ext
            return hip_ext().residual_add(y.contiguous(), resid.contiguous())
        return super().forward_resid(x, resid)


class RowParallelQLinear(QLinear):
    """Input dimension sharded; partial outputs all-reduced (RCCL/xGMI —
    the collective call site of SURVEY.md §2.5)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = super().forward(x)
        return lfmq_dist.all_reduce_sum(y)

    def forward_resid(self, x: torch.Tensor, resid: torch.Tensor) -> torch.Tensor:
        # the residual may be added only ONCE, after the reduce
        if lfmq_dist.get_world_size() > 1:
            y = self.forward(x)
            from ..ops import hip_ext
            return hip_ext().residual_add(y.contiguous(), resid.contiguous())
        return super().forward_resid(x, resid)
