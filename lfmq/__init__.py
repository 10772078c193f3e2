"""lfmq — MI355X-native post-training LLM weight-quantization engine.

Built from scratch for gfx950 (CDNA4): GPTQ/AWQ/RTN calibration with
hand-written HIP Hessian/pack kernels, W4A16/W8A16 dequant-fused GEMV/GEMM
on MFMA with bf16 activations through LDS, fused RMSNorm/RoPE, TP
all-reduce on RCCL over xGMI.  See SURVEY.md for the blueprint.
"""
from .api import build_model, load, quantize, save
from .config import FORMAT_VERSION, GenerationConfig, QuantConfig

__version__ = "0.2.0"

__all__ = ["quantize", "save", "load", "build_model", "QuantConfig",
           "GenerationConfig", "FORMAT_VERSION", "__version__"]
