"""Round-to-nearest quantizer: per-tensor int8 (config #1) and per-group
int4/int8 (the RTN baseline for GPTQ/AWQ comparisons)."""
from __future__ import annotations

import torch
import torch.nn as nn

from ..config import QuantConfig
from ..models.qlinear import Int8PerTensorLinear, QLinear
from .common import iter_target_linears, replace_module


class RTNQuantizer:
    def __init__(self, cfg: QuantConfig):
        self.cfg = cfg

    @torch.no_grad()
    def run(self, model: nn.Module) -> nn.Module:
        for name, lin in iter_target_linears(model, self.cfg):
            if self.cfg.per_tensor:
                q = Int8PerTensorLinear.from_linear(lin)
            else:
                q = QLinear.from_rtn(lin, self.cfg)
            replace_module(model, name, q)
        return model
