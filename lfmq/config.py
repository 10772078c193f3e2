"""Quantization and run configuration.

The reference tree is empty; the binding spec is the north
star in BASELINE.json:5 — scheme ∈ {rtn, gptq, awq}, bits ∈ {4, 8},
group-wise scale/zero-point, int4 packing. QuantConfig is the single
source of truth for those knobs and round-trips to/from the packed
checkpoint's quant_config.json (lfmq/checkpoint.py).
"""
from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import Optional

# Packed-checkpoint layout version.  Bump whenever the qweight/scales/zeros
# tensor layout changes (lfmq/checkpoint.py records it; load() refuses a
# mismatch rather than silently mis-decoding weights).
FORMAT_VERSION = 1

SCHEMES = ("rtn", "gptq", "awq")


@dataclass
class QuantConfig:
    """Configuration of one quantization run.

    scheme:      "rtn" | "gptq" | "awq"
    bits:        4 or 8 (weight bits; activations stay bf16 — W4A16 / W8A16)
    group_size:  per-group quantization along the input (k) dimension;
                 -1 means per-channel (one group = whole row).
                 0 plus per_tensor=True means a single scale for the tensor
                 (config #1: OPT-125m per-tensor int8 RTN).
    sym:         symmetric (no zero-point) vs asymmetric quantization.
    per_tensor:  per-tensor scale (int8 RTN CPU path only).
    """

    scheme: str = "gptq"
    bits: int = 4
    group_size: int = 128
    sym: bool = False
    per_tensor: bool = False

    # GPTQ knobs
    damp_percent: float = 0.01      # λ = damp_percent · mean(diag(H))
    block_size: int = 128           # column block for the error-propagation loop
    act_order: bool = False         # permute columns by diag(H) descending

    # AWQ knobs
    awq_grid: int = 20              # α grid resolution over [0, 1]
    awq_clip: bool = False          # optional weight-clip search

    # fold RMSNorm weights into the next projection pre-quantization
    # (enables the fused-norm decode GEMV; lfmq/quant/fold.py)
    fold_rms: bool = True

    # calibration
    calib_seq_len: int = 512
    calib_n_seqs: int = 16
    calib_seed: int = 1234

    # which modules to skip (by suffix match on the module path)
    skip_modules: tuple = ("lm_head",)

    def __post_init__(self):
        if self.scheme not in SCHEMES:
            raise ValueError(f"scheme must be one of {SCHEMES}, got {self.scheme!r}")
        if self.bits not in (4, 8):
            raise ValueError(f"bits must be 4 or 8, got {self.bits}")
        if self.per_tensor:
            if self.bits != 8:
                raise ValueError("per-tensor quantization is int8-only")
        elif self.group_size not in (-1,) and self.group_size <= 0:
            raise ValueError(f"group_size must be positive or -1, got {self.group_size}")

    def to_dict(self) -> dict:
        d = dataclasses.asdict(self)
        d["skip_modules"] = list(self.skip_modules)
        d["format_version"] = FORMAT_VERSION
        return d

    @classmethod
    def from_dict(cls, d: dict) -> "QuantConfig":
        d = dict(d)
        ver = d.pop("format_version", FORMAT_VERSION)
        if ver != FORMAT_VERSION:
            raise ValueError(
                f"packed checkpoint layout version {ver} != supported {FORMAT_VERSION}"
            )
        d["skip_modules"] = tuple(d.get("skip_modules", ()))
        known = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in d.items() if k in known})

    def json(self) -> str:
        return json.dumps(self.to_dict(), indent=2, sort_keys=True)


@dataclass
class GenerationConfig:
    max_new_tokens: int = 64
    greedy: bool = True
    temperature: float = 1.0
    eos_token_id: Optional[int] = None
