"""Op bindings: the single compute backend is PyTorch-ROCm + the in-tree
HIP extension (csrc/ -> lfmq_hip.so, gfx950 only).

There is deliberately NO multi-backend dispatch (BASELINE.json north star):
on a GPU device the HIP extension is the only path and a missing extension
raises immediately instead of silently falling back to eager PyTorch.  The
CPU twins in lfmq.ops.ref exist for unit tests and for the OPT-125m int8
CPU decode config, which is spec'd as a CPU path.
"""
from __future__ import annotations

from typing import Optional

import torch

from . import ref  # noqa: F401

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import importlib
        _EXT = importlib.import_module("lfmq_hip")
    except Exception as e:  # pragma: no cover - exercised only without the .so
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def hip_ext():
    """The loaded HIP extension module, or raise.  GPU code MUST go through
    this — never fall back to eager torch on a GPU device."""
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "lfmq_hip extension not built (run `python setup.py build_ext "
            f"--inplace`, PYTORCH_ROCM_ARCH=gfx950). Import error: {_EXT_ERR}"
        )
    return ext


def hip_ext_available() -> bool:
    return _load_ext() is not None
