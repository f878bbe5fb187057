#!/usr/bin/env python3
"""Overlay tokens-vs-time CSV curves of 1/2/../8-node runs for one model.

Capability parity with /root/reference/src/plot_tok_time.py (the published
benchmark figure generator); joins runs by the CSV file-name convention.
"""

import argparse
import sys
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", type=str, required=True,
                   help="model name as it appears in the CSV file names")
    p.add_argument("--logs", type=Path, default=SCRIPT_DIR / "logs")
    p.add_argument("--out", type=Path, default=None)
    args = p.parse_args()

    from mdi_llm_amd.utils.plots import collect_csv_runs, plot_tokens_per_time

    runs = collect_csv_runs(args.logs, args.model)
    if not runs:
        sys.exit(f"no tokens_time CSVs for {args.model!r} in {args.logs}")
    out = args.out or args.logs / f"tokens_time_{args.model}.png"
    plot_tokens_per_time(runs, out, args.model)
    print(f"[plot] {len(runs)} runs -> {out}")
