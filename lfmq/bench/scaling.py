"""TP scaling-curve runner (SURVEY.md §1 bench inventory).

Launches `bench.py` via torch.distributed.run for each N in gpus and
collects the JSON lines — the same protocol the driver uses for
SCALE_rNN.json (weak/strong efficiency is computed by the consumer from
the per-N values, never reported by us).

    python -m lfmq.bench.scaling --gpus 1,2,4,8 --steps 128
"""
from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys


def run_point(n: int, steps: int, warmup: int, extra: list) -> dict:
    repo = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    bench = os.path.join(repo, "bench.py")
    if n == 1:
        cmd = [sys.executable, bench]
    else:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
               "--master-port", "29631", bench]
    cmd += ["--gpus", str(n), "--steps", str(steps), "--warmup", str(warmup)]
    cmd += extra
    out = subprocess.run(cmd, capture_output=True, text=True, cwd=repo)
    for line in out.stdout.splitlines():
        line = line.strip()
        if line.startswith("{") and '"metric"' in line:
            return json.loads(line)
    raise RuntimeError(f"N={n}: no bench JSON in output:\n{out.stdout[-2000:]}"
                       f"\n{out.stderr[-2000:]}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", default="1,2,4,8")
    ap.add_argument("--steps", type=int, default=128)
    ap.add_argument("--warmup", type=int, default=16)
    ap.add_argument("--out", default="gpurun_out/scaling.json")
    args, extra = ap.parse_known_args()
    rows = []
    for n in [int(x) for x in args.gpus.split(",")]:
        r = run_point(n, args.steps, args.warmup, extra)
        print(json.dumps(r))
        rows.append(r)
    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    with open(args.out, "w") as f:
        json.dump(rows, f, indent=2)


if __name__ == "__main__":
    main()
