"""Driver-contract guard: `python bench.py` must emit ONE JSON line with
every field the driver parses (BENCH_rNN.json / SCALE_rNN.json inputs),
runnable on CPU with the tiny debug model.  Also covers the report-row
emitter and the scaling runner's JSON-line extraction."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model",
         "tiny-debug", "--steps", "2", "--warmup", "1", "--prompt", "8"],
        capture_output=True, text=True, cwd=REPO, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.strip().startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, out.stdout
    r = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in r, key
    assert r["metric"] == "decode tok/s" and r["unit"] == "tok/s"
    assert r["value"] > 0 and r["ms_per_step"] > 0
    assert r["n_gpus"] == 1 and r["steps"] == 2 and r["warmup"] == 1
    assert r["higher_is_better"] is True and r["data"] == "synthetic"
    assert r["dtype"] == "bf16" and r["scaling"] == "strong"
    cfgk = r["config"]
    for key in ("model", "quant", "global_batch", "seq_len", "parallelism"):
        assert key in cfgk, key


def test_report_rows_to_markdown():
    from lfmq.bench.report import rows_to_markdown
    rows = [{"scheme": "gptq", "tok_per_s": 506.8, "delta_ppl": 22.7,
             "logit_kl": 8.5e-5, "quant_seconds": 159.0}]
    md = rows_to_markdown(rows)
    assert "GPTQ" in md and "506.8" in md and "8.5e-05" in md
    assert md.count("\n") == 2  # header + separator + one row


def test_scaling_runner_single_point():
    from lfmq.bench.scaling import run_point
    r = run_point(1, steps=2, warmup=1,
                  extra=["--model", "tiny-debug", "--prompt", "8"])
    assert r["metric"] == "decode tok/s" and r["n_gpus"] == 1
