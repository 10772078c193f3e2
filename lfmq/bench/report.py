"""BASELINE.md results-row emitter (SURVEY.md §2 bench inventory)."""
from __future__ import annotations

import json
from typing import List


def rows_to_markdown(rows: List[dict]) -> str:
    """baseline_rows.json entries -> BASELINE.md table rows."""
    out = ["| config | TP | decode tok/s (b=1) | delta-ppl | logit KL | notes |",
           "|---|---|---|---|---|---|"]
    for r in rows:
        out.append(
            f"| Llama-3-8B {r['scheme'].upper()} W4A16 g128 | 1 "
            f"| {r['tok_per_s']:.1f} | {r.get('delta_ppl', float('nan')):+.1f} "
            f"| {r.get('logit_kl', float('nan')):.1e} "
            f"| quantize {r.get('quant_seconds', 0):.0f} s on-GPU |")
    return "\n".join(out)


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", default="gpurun_out/baseline_rows.json")
    args = ap.parse_args()
    with open(args.rows) as f:
        print(rows_to_markdown(json.load(f)))


if __name__ == "__main__":
    main()
