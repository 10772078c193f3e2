"""Minimal FastAPI generation server with prometheus counters.

Not on the benchmark critical path — a thin serving
front for the quantized decode engine:
    POST /generate  {"input_ids": [[...]], "max_new_tokens": 32}
    GET  /metrics   prometheus text (tokens generated, requests, tok/s)
    GET  /health

Greedy GPU requests decode through the hipGraph path (GraphDecoder, one
graph replay per token — the same machinery as the benchmark's 522
tok/s); sampling requests and CPU models use the eager loop.  One
GraphDecoder is cached per batch size and guarded by a lock (FastAPI
sync endpoints run in a thread pool).
"""
from __future__ import annotations

import threading
import time
from typing import Dict, List

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
    graph_decoders: Dict[int, object] = {}   # batch size -> GraphDecoder
    graph_lock = threading.Lock()

    def _graph_generate(ids: torch.Tensor, max_new: int) -> torch.Tensor:
        from ..models.graph import GraphDecoder
        b = ids.shape[0]
        need = ids.shape[1] + max_new + 8
        with graph_lock:
            gd = graph_decoders.get(b)
            if gd is None or gd.max_seq < need or \
                    gd.ring.shape[1] < max_new:
                gd = GraphDecoder(model, b, max_seq=max(1024, need),
                                  max_new_tokens=max(4096, max_new))
                graph_decoders[b] = gd
            gd.prefill(ids)
            toks = gd.decode(max_new)
        return torch.cat([ids, toks], dim=1)

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
        use_graph = (device.type == "cuda" and req.greedy
                     and req.max_new_tokens > 0)
        with torch.no_grad():
            if use_graph:
                out = _graph_generate(ids, req.max_new_tokens)
            else:
                out = model.generate(ids, max_new_tokens=req.max_new_tokens,
                                     gen_cfg=gc, generator=gen)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        if req.eos_token_id is not None:
            # the graph ring decodes the full budget; truncate at the
            # step where every row has emitted eos (eager semantics)
            new_part = out[:, ids.shape[1]:]
            hit = (new_part == req.eos_token_id).cumsum(dim=1) > 0
            if bool(hit.all(dim=0).any()):
                stop = int(hit.all(dim=0).float().argmax()) + 1
                out = out[:, : ids.shape[1] + stop]
        new = out.shape[1] - ids.shape[1]
        TOKS.inc(new * ids.shape[0])
        TOKS_PER_S.set(new * ids.shape[0] / max(dt, 1e-9))
        return {"output_ids": out.tolist(), "new_tokens": new,
                "seconds": dt}

    @app.get("/metrics")
    def metrics():
        return Response(generate_latest(), media_type=CONTENT_TYPE_LATEST)

    return app
