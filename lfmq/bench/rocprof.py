"""rocprofv3 wrapper: capture kernel traces/stats for a
bench command and summarize per-kernel totals from the result DB.

    from lfmq.bench.rocprof import profile_command, summarize_db
    db = profile_command("python bench.py --steps 16", out_dir="gpurun_out/prof")
    rows = summarize_db(db)

Counter captures (--pmc) must run in their own pass — never combined
with the trace domains (MI355X pool constraint); use pmc_command() for
those.  gfx950 FETCH_SIZE calibration caveat: MI355X_MICROARCH.md §HBM —
this wrapper reports wall times only.
"""
from __future__ import annotations

import glob
import os
import sqlite3
import subprocess
from typing import List, Optional


def profile_command(cmd: str, out_dir: str = "gpurun_out/prof",
                    name: str = "lfmq", timeout: int = 600) -> Optional[str]:
    """Run `rocprofv3 --kernel-trace --stats` around cmd; returns the
    results DB path (or None on failure)."""
    # rocprofv3 resolves a relative -d against ITS cwd (/tmp below, the
    # TMPDIR workaround) — absolutize so the DB lands where we glob
    out_dir = os.path.abspath(out_dir)
    os.makedirs(out_dir, exist_ok=True)
    env = dict(os.environ, TMPDIR="/tmp")
    full = ["rocprofv3", "--kernel-trace", "--stats", "-d", out_dir,
            "-o", name, "--"] + cmd.split()
    try:
        subprocess.run(full, env=env, cwd="/tmp", timeout=timeout, check=False,
                       stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    except (subprocess.TimeoutExpired, FileNotFoundError):
        return None
    dbs = sorted(glob.glob(os.path.join(out_dir, f"{name}*.db")))
    return dbs[-1] if dbs else None


def pmc_command(cmd: str, counters: List[str],
                out_dir: str = "gpurun_out/pmc", name: str = "lfmq_pmc",
                timeout: int = 600) -> Optional[str]:
    """Counter collection in its OWN pass (no trace domains)."""
    out_dir = os.path.abspath(out_dir)
    os.makedirs(out_dir, exist_ok=True)
    env = dict(os.environ, TMPDIR="/tmp")
    full = ["rocprofv3", "--pmc", ",".join(counters), "-d", out_dir,
            "-o", name, "--"] + cmd.split()
    try:
        subprocess.run(full, env=env, cwd="/tmp", timeout=timeout, check=False,
                       stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    except (subprocess.TimeoutExpired, FileNotFoundError):
        return None
    dbs = sorted(glob.glob(os.path.join(out_dir, f"{name}*.db")))
    return dbs[-1] if dbs else None


def summarize_db(db_path: str, top: int = 20, tail_frac: float = 0.5) -> list:
    """Per-kernel (count, total ms, us/call) over the steady-state tail
    of the capture, heaviest first."""
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'")]
    if not tables:
        return []
    sfx = tables[0].split("dispatch_")[1]
    lo, hi = list(cur.execute(
        f"SELECT MIN(start), MAX(end) FROM rocpd_kernel_dispatch_{sfx}"))[0]
    cut = lo + (hi - lo) * tail_frac
    q = (f"SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6, "
         f"AVG(d.end-d.start)/1e3 "
         f"FROM rocpd_kernel_dispatch_{sfx} d "
         f"JOIN rocpd_info_kernel_symbol_{sfx} s ON d.kernel_id=s.id "
         f"WHERE d.start > {cut} GROUP BY s.display_name "
         f"ORDER BY 3 DESC LIMIT {top}")
    return [{"kernel": n, "calls": c, "total_ms": ms, "us_per_call": us}
            for n, c, ms, us in cur.execute(q)]


def torch_profile(fn, steps: int = 8, out_trace: str = None):
    """torch.profiler wrapper for launch-gap analysis:
    runs fn() `steps` times under the ROCm-aware profiler and returns the
    key-averages table; optionally exports a chrome trace."""
    import torch
    from torch.profiler import ProfilerActivity, profile
    acts = [ProfilerActivity.CPU]
    if torch.cuda.is_available():
        acts.append(ProfilerActivity.CUDA)
    with profile(activities=acts) as prof:
        for _ in range(steps):
            fn()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
    if out_trace:
        prof.export_chrome_trace(out_trace)
    return prof.key_averages().table(sort_by="self_cuda_time_total"
                                     if torch.cuda.is_available()
                                     else "self_cpu_time_total", row_limit=25)
