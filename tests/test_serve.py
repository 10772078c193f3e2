"""Serve API (obs/serve.py) via the starlette test client (CPU; one
GPU-marked test covers the hipGraph serving path)."""
import pytest
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


def test_serve_sampling_options():
    import torch
    from fastapi.testclient import TestClient

    import lfmq
    from lfmq.models.llama import LlamaConfig
    from lfmq.obs.serve import make_app

    cfg = LlamaConfig.tiny()
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=2),
                       lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32))
    client = TestClient(make_app(qm))
    body = {"input_ids": [[1, 2, 3, 4]], "max_new_tokens": 6,
            "greedy": False, "temperature": 0.7, "seed": 9}
    a = client.post("/generate", json=body).json()
    b = client.post("/generate", json=body).json()
    assert a["output_ids"] == b["output_ids"]      # seeded => reproducible
    assert a["new_tokens"] == 6
    toks = a["output_ids"][0][4:]
    assert all(0 <= t < cfg.vocab_size for t in toks)


@pytest.mark.gpu
def test_serve_graph_path_gpu():
    """Greedy GPU requests go through GraphDecoder (hipGraph replay per
    token) and must match the eager generate output exactly; repeated
    requests reuse the cached decoder."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    cfg = LlamaConfig(vocab_size=512, hidden_size=512, intermediate_size=1024,
                      n_layers=2, n_heads=8, n_kv_heads=4, head_dim=64,
                      max_seq=256)
    m = lfmq.build_model(cfg, seed=5).to("cuda")
    qm = lfmq.quantize(m, lfmq.QuantConfig(scheme="rtn", bits=4,
                                           group_size=64))
    ids = [[int(x) for x in torch.randint(0, 512, (12,))]]
    ref = qm.generate(torch.tensor(ids, device="cuda"),
                      max_new_tokens=6).cpu().tolist()
    client = TestClient(make_app(qm))
    for _ in range(2):  # second hit reuses the cached GraphDecoder
        r = client.post("/generate", json={"input_ids": ids,
                                           "max_new_tokens": 6})
        assert r.status_code == 200
        assert r.json()["output_ids"] == ref
