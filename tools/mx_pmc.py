#!/usr/bin/env python3
"""PMC pass over the Mixtral bench; per-kernel counter summary
(VERDICT r1 item 4 evidence)."""
import glob
import os
import sqlite3
import subprocess
import sys

COUNTERS = ["SQ_WAVE_CYCLES", "SQ_WAIT_ANY", "SQ_ACTIVE_INST_ANY",
            "SQ_INSTS_VALU", "SQ_INSTS_VMEM", "SQ_LDS_BANK_CONFLICT"]
OUT = "/tmp/pmc_mx"


def main():
    os.makedirs(OUT, exist_ok=True)
    env = dict(os.environ, TMPDIR="/tmp")
    cmd = ["rocprofv3", "--pmc", ",".join(COUNTERS), "-d", OUT, "-o", "mx",
           "--", "python", "/root/repo/bench.py", "--model", "mixtral-8x7b",
           "--steps", "16", "--warmup", "4", "--no-fidelity"]
    subprocess.run(cmd, env=env, cwd="/tmp", check=False, timeout=400,
                   stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    dbs = sorted(glob.glob(os.path.join(OUT, "mx*.db")))
    if not dbs:
        print("no db")
        return 1
    db = sqlite3.connect(dbs[-1])
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'").fetchall()]

    def tbl(p):
        return next((t for t in tables if t.startswith(p)), None)

    t_pmc, t_disp, t_info = (tbl("rocpd_pmc_event"),
                             tbl("rocpd_kernel_dispatch"), tbl("rocpd_info_pmc"))
    t_ksym = next((t for t in tables if "kernel_symbol" in t), None)
    names = dict(cur.execute(
        f"SELECT id, name FROM {t_info}").fetchall())
    kname = {}
    if t_ksym:
        cols = [d[1] for d in cur.execute(f"PRAGMA table_info({t_ksym})")]
        nc = ("truncated_kernel_name" if "truncated_kernel_name" in cols
              else "display_name" if "display_name" in cols else cols[-1])
        kname = dict(cur.execute(f"SELECT id, {nc} FROM {t_ksym}").fetchall())
    rows = cur.execute(f"""
        SELECT d.kernel_id, e.pmc_id, SUM(e.value), COUNT(DISTINCT d.id)
        FROM {t_pmc} e JOIN {t_disp} d ON e.event_id = d.event_id
        GROUP BY d.kernel_id, e.pmc_id""").fetchall()
    agg = {}
    for kid, pid, tot, nd in rows:
        agg.setdefault(kid, {"n": nd})[names.get(pid, pid)] = tot
    top = sorted(agg.items(), key=lambda x: -x[1].get("SQ_WAVE_CYCLES", 0))[:8]
    for kid, v in top:
        nm = str(kname.get(kid, kid))[:56]
        wc = v.get("SQ_WAVE_CYCLES", 1) or 1
        print(f"{nm} calls={v['n']}")
        for c in COUNTERS:
            val = v.get(c, 0)
            print(f"   {c}: {val:.3e} ({100*val/wc:.1f}%)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
