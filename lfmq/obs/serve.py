"""Minimal FastAPI generation server with prometheus counters.

Not on the benchmark critical path — a thin serving
front for the quantized decode engine:
    POST /generate  {"input_ids": [[...]], "max_new_tokens": 32}
    GET  /metrics   prometheus text (tokens generated, requests, tok/s)
    GET  /health
"""
from __future__ import annotations

import time
from typing import List

import torch
from fastapi import FastAPI
from pydantic import BaseModel
from prometheus_client import (CONTENT_TYPE_LATEST, Counter, Gauge,
                               generate_latest)
from starlette.responses import Response

REQS = Counter("lfmq_requests_total", "generation requests")
TOKS = Counter("lfmq_tokens_total", "tokens generated")
TOKS_PER_S = Gauge("lfmq_decode_tok_per_s", "last request decode tok/s")


class GenRequest(BaseModel):
    input_ids: List[List[int]]
    max_new_tokens: int = 32
    greedy: bool = True
    temperature: float = 1.0
    eos_token_id: int | None = None
    seed: int | None = None


def make_app(model) -> FastAPI:
    app = FastAPI(title="lfmq")
    device = next(model.parameters()).device

    @app.get("/health")
    def health():
        return {"ok": True, "device": str(device),
                "arch": model.cfg.arch}

    @app.post("/generate")
    def generate(req: GenRequest):
        REQS.inc()
        ids = torch.tensor(req.input_ids, dtype=torch.int64, device=device)
        from ..config import GenerationConfig
        gc = GenerationConfig(max_new_tokens=req.max_new_tokens,
                              greedy=req.greedy, temperature=req.temperature,
                              eos_token_id=req.eos_token_id)
        gen = None
        if req.seed is not None:
            # generator device must match the logits device (multinomial)
            gen = torch.Generator(device=str(device)).manual_seed(req.seed)
        t0 = time.perf_counter()
        with torch.no_grad():
            out = model.generate(ids, max_new_tokens=req.max_new_tokens,
                                 gen_cfg=gc, generator=gen)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        new = out.shape[1] - ids.shape[1]
        TOKS.inc(new * ids.shape[0])
        TOKS_PER_S.set(new * ids.shape[0] / max(dt, 1e-9))
        return {"output_ids": out.tolist(), "new_tokens": new,
                "seconds": dt}

    @app.get("/metrics")
    def metrics():
        return Response(generate_latest(), media_type=CONTENT_TYPE_LATEST)

    return app
