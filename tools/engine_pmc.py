#!/usr/bin/env python3
"""PMC pass over the engine bench: where do engine waves spend cycles?
Aggregates on-box (the rocpd DB is huge); prints a compact summary."""
import glob
import os
import sqlite3
import subprocess
import sys

COUNTERS = ["SQ_WAVE_CYCLES", "SQ_WAIT_ANY", "SQ_ACTIVE_INST_ANY",
            "SQ_INSTS_LDS", "SQ_INSTS_VALU", "SQ_LDS_BANK_CONFLICT"]
OUT = "/tmp/pmc_eng"


def main():
    os.makedirs(OUT, exist_ok=True)
    env = dict(os.environ, TMPDIR="/tmp")
    cmd = ["rocprofv3", "--pmc", ",".join(COUNTERS), "-d", OUT,
           "-o", "eng", "--", "python", "/root/repo/bench.py", "--steps",
           "24", "--warmup", "4", "--no-fidelity"]
    subprocess.run(cmd, env=env, cwd="/tmp", check=False, timeout=420,
                   stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    dbs = sorted(glob.glob(os.path.join(OUT, "eng*.db")))
    if not dbs:
        print("no db")
        return 1
    db = sqlite3.connect(dbs[-1])
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'").fetchall()]

    def tbl(prefix):
        for t in tables:
            if t.startswith(prefix):
                return t
        return None

    t_pmc = tbl("rocpd_pmc_event")
    t_disp = tbl("rocpd_kernel_dispatch")
    t_info = tbl("rocpd_info_pmc")
    # counter names
    names = {}
    cols = [d[1] for d in cur.execute(f"PRAGMA table_info({t_info})")]
    print("info_pmc cols:", cols)
    idc = "id"
    nmc = "name" if "name" in cols else cols[1]
    for r in cur.execute(f"SELECT {idc}, {nmc} FROM {t_info}"):
        names[r[0]] = r[1]
    # dispatches: engine kernel = group_segment_size > 100000
    rows = cur.execute(f"""
        SELECT d.group_segment_size,
               (d.end - d.start) * 1e-6 AS ms,
               e.pmc_id, SUM(e.value), COUNT(*)
        FROM {t_pmc} e JOIN {t_disp} d ON e.event_id = d.event_id
        GROUP BY d.kernel_id, e.pmc_id
        HAVING d.group_segment_size > 100000""").fetchall()
    if not rows:
        print("no big-LDS kernel found; dumping top kernels by cycles")
    # aggregate engine dispatch stats
    agg = {}
    n_disp = cur.execute(f"""
        SELECT COUNT(DISTINCT d.id), AVG(d.end - d.start) * 1e-6
        FROM {t_disp} d WHERE d.group_segment_size > 100000""").fetchone()
    print(f"engine dispatches: {n_disp[0]}, avg wall {n_disp[1]:.3f} ms")
    rows = cur.execute(f"""
        SELECT e.pmc_id, SUM(e.value), COUNT(*)
        FROM {t_pmc} e JOIN {t_disp} d ON e.event_id = d.event_id
        WHERE d.group_segment_size > 100000
        GROUP BY e.pmc_id""").fetchall()
    base = None
    vals = {}
    for pid, tot, cnt in rows:
        nm = names.get(pid, str(pid))
        vals[nm] = tot / max(n_disp[0], 1)
    wc = vals.get("SQ_WAVE_CYCLES", 0) or 1
    for nm, v in sorted(vals.items()):
        print(f"  {nm}: {v:.3e} per dispatch ({100*v/wc:.1f}% of WAVE_CYCLES)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
