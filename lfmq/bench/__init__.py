"""Benchmark harnesses: decode tok/s + delta-ppl (BASELINE.md protocol)."""
from .decode import bench_decode, build_quantized_llama
from .ppl import bench_ppl
