"""Serve API (obs/serve.py) via the starlette test client (CPU)."""
import torch
from starlette.testclient import TestClient

import lfmq
from lfmq.models.llama import LlamaConfig
from lfmq.obs.serve import make_app


def test_serve_generate_and_metrics():
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=3), qcfg)
    client = TestClient(make_app(qm))
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["ok"]
    r = client.post("/generate", json={"input_ids": [[1, 2, 3, 4]],
                                       "max_new_tokens": 4})
    assert r.status_code == 200
    body = r.json()
    assert len(body["output_ids"][0]) == 8
    r = client.get("/metrics")
    assert r.status_code == 200 and b"lfmq_tokens_total" in r.content
