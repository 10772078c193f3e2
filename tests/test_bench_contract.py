"""Driver-contract guard: `python bench.py` must emit ONE JSON line with
every field the driver parses (BENCH_rNN.json / SCALE_rNN.json inputs),
runnable on CPU with the tiny debug model.  Also covers the report-row
emitter and the scaling runner's JSON-line extraction."""
import json
import os
import subprocess
import sys

import pytest

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


def _run_bench_torchrun(world: int, model: str, extra=()):
    """Launch bench.py exactly as the driver's SCALE_rNN runner does
    (torch.distributed.run, one rank per GPU, rendezvous on 127.0.0.1)
    and return rank 0's JSON line."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", str(world), "--steps", "2", "--warmup", "1",
         "--prompt", "8", "--model", model, "--no-fidelity", *extra],
        capture_output=True, text=True, cwd=REPO, timeout=600)
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-2000:])
    lines = [ln for ln in out.stdout.splitlines()
             if ln.strip().startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, out.stdout  # rank 0 only
    return json.loads(lines[0])


def test_bench_torchrun_world2_gloo():
    """TP-sharded tiny model at world 2 on CPU (gloo): all-reduce /
    all-gather in the decode loop, one JSON line with n_gpus == 2."""
    r = _run_bench_torchrun(2, "tiny-debug")
    assert r["n_gpus"] == 2
    assert r["config"]["parallelism"] == "tp2"
    assert r["value"] > 0


@pytest.mark.parametrize("world", [4, 8])
def test_bench_torchrun_world48_gloo(world):
    """The driver's 8-GPU scale run is the FIRST time TP 4/8 executes on
    real hardware — de-risk the rank math (head/kv/group splits, lm_head
    shard gather) at those worlds here on CPU."""
    r = _run_bench_torchrun(world, "tiny-debug-tp8",
                            extra=("--group-size", "32"))
    assert r["n_gpus"] == world
    assert r["config"]["parallelism"] == f"tp{world}"
    assert r["value"] > 0


def test_rocprof_wrapper_paths(tmp_path, monkeypatch):
    """profile_command must absolutize -d: rocprofv3 resolves a relative
    output dir against its own cwd (/tmp), not the caller's — the DB
    would land in /tmp and the glob would miss it.  Stub the binary and
    check the constructed command + returned DB path."""
    import lfmq.bench.rocprof as rp

    seen = {}

    def fake_run(cmd, **kw):
        seen["cmd"] = cmd
        d = cmd[cmd.index("-d") + 1]
        assert os.path.isabs(d), d
        name = cmd[cmd.index("-o") + 1]
        with open(os.path.join(d, f"{name}_results.db"), "w") as f:
            f.write("x")
        class R: pass
        return R()

    monkeypatch.setattr(rp.subprocess, "run", fake_run)
    monkeypatch.chdir(tmp_path)
    db = rp.profile_command("python bench.py --steps 2", out_dir="prof")
    assert db is not None and os.path.exists(db)
    assert seen["cmd"][0] == "rocprofv3" and "--kernel-trace" in seen["cmd"]
    db = rp.pmc_command("python bench.py", ["SQ_WAVES"], out_dir="pmc")
    assert db is not None and "--pmc" in seen["cmd"]
    assert "--kernel-trace" not in seen["cmd"]  # counters ride alone


def test_rocprof_summarize_db(tmp_path):
    """summarize_db against a synthetic rocpd-schema DB: steady-state
    tail filter, per-kernel aggregation, heaviest-first order."""
    import sqlite3

    from lfmq.bench.rocprof import summarize_db

    db_path = str(tmp_path / "fake_results.db")
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    cur.execute("CREATE TABLE rocpd_kernel_dispatch_abc "
                "(kernel_id INT, start INT, end INT)")
    cur.execute("CREATE TABLE rocpd_info_kernel_symbol_abc "
                "(id INT, display_name TEXT)")
    cur.execute("INSERT INTO rocpd_info_kernel_symbol_abc VALUES (1, 'gemv')")
    cur.execute("INSERT INTO rocpd_info_kernel_symbol_abc VALUES (2, 'attn')")
    # warmup dispatch in the first half (must be cut by tail_frac=0.5)
    cur.execute("INSERT INTO rocpd_kernel_dispatch_abc VALUES (1, 0, 100000)")
    # steady state: gemv heavier than attn
    for t in range(600000, 1000000, 100000):
        cur.execute("INSERT INTO rocpd_kernel_dispatch_abc VALUES "
                    f"(1, {t}, {t + 50000})")
        cur.execute("INSERT INTO rocpd_kernel_dispatch_abc VALUES "
                    f"(2, {t}, {t + 10000})")
    db.commit()
    db.close()
    rows = summarize_db(db_path)
    assert [r["kernel"] for r in rows] == ["gemv", "attn"]
    assert rows[0]["calls"] == 4  # the warmup dispatch was excluded
    assert rows[0]["total_ms"] > rows[1]["total_ms"]
