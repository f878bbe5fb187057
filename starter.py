#!/usr/bin/env python3
"""Starter node CLI — launches model-distributed generation.

Flag-compatible with the reference starter (/root/reference/src/starter.py):
--nodes-config, --ckpt, --chunk, --n-samples, --n-tokens,
--sequence-length/--block-size, --dtype, --device, --prompt (incl. FILE:),
--time-run, --plots, --seed, -v/--verb, -d/--debug.
"""

import argparse
import cProfile
import csv
import sys
import time
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))


def main(args):
    import os as _os

    if getattr(args, "weights", "bf16") == "fp8":
        _os.environ["MDI_WEIGHT_DTYPE"] = "fp8"
    if getattr(args, "kv", "bf16") == "fp8":
        _os.environ["MDI_KV_DTYPE"] = "fp8"
    import torch

    from mdi_llm_amd.parallel.orchestrator import MDIRuntime
    from mdi_llm_amd.utils.plots import (
        plot_tokens_per_time,
        tok_time_csv_name,
        write_tok_time_csv,
    )

    torch.manual_seed(args.seed)
    rt = MDIRuntime(
        "starter",
        config_file=args.nodes_config,
        ckpt_dir=args.ckpt,
        chunk_path=args.chunk,
        device=args.device,
        dtype=args.dtype,
        model_seq_length=args.sequence_length,
        verb=args.verb,
    )
    res = rt.start(
        n_samples=args.n_samples,
        tokens_per_sample=args.n_tokens,
        prompt=args.prompt,
        seed=args.seed,
    )

    for i, seq in enumerate(res.sequences):
        text = rt.tokenizer.decode(seq)
        print(f"\n========== sample {i} ==========\n{text}")
    print(
        f"\n[starter] {res.total_new_tokens} tokens in {res.gen_time:.2f}s "
        f"({res.tokens_per_second:.2f} tok/s, {rt.world} node(s))"
    )

    model_name = rt.stage.config.name
    logs = SCRIPT_DIR / "logs"
    csv_path = logs / tok_time_csv_name(rt.world, model_name, args.n_samples)
    write_tok_time_csv(csv_path, res.tok_time)
    print(f"[starter] tok/time CSV -> {csv_path}")

    if args.time_run is not None:
        # run-level stats CSV (reference starter.py:89-105)
        args.time_run.parent.mkdir(parents=True, exist_ok=True)
        new = not args.time_run.exists()
        with open(args.time_run, "a", newline="") as fp:
            w = csv.writer(fp)
            if new:
                w.writerow(["timestamp", "n_samples", "n_layers",
                            "context_size", "gen_time"])
            w.writerow([
                time.strftime("%Y-%m-%d %H:%M:%S"),
                args.n_samples,
                rt.stage.config.n_layer,
                rt.stage.max_seq_length,
                f"{res.gen_time:.3f}",
            ])

    if args.plots:
        png = logs / f"tokens_time_{model_name}.png"
        from mdi_llm_amd.utils.plots import collect_csv_runs

        plot_tokens_per_time(collect_csv_runs(logs, model_name) or [csv_path],
                             png, model_name)
        print(f"[starter] plot -> {png}")


def build_parser():
    p = argparse.ArgumentParser(description="Starter node - MDI (MI355X)")
    p.add_argument("-d", "--debug", action="store_true",
                   help="enable debug mode (profiler)")
    p.add_argument("-v", "--verb", action="store_true")
    p.add_argument("-p", "--plots", action="store_true")
    p.add_argument("--ckpt", type=Path,
                   default=SCRIPT_DIR / "checkpoints" / "custom" / "NanoLlama")
    p.add_argument("--chunk", type=Path, default=None)
    p.add_argument("--nodes-config", type=Path,
                   default=SCRIPT_DIR / "settings_distr" / "configuration.json")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--prompt", type=str, default="Who are you?",
                   help="prompt text, or FILE:<path> for per-sample paragraphs")
    p.add_argument("--n-samples", type=int, default=3)
    p.add_argument("--n-tokens", type=int, default=300)
    p.add_argument("--sequence-length", "--context-length", "--block-size",
                   type=int, default=None, dest="sequence_length")
    p.add_argument("--dtype", type=str, default=None)
    p.add_argument("--weights", choices=["bf16", "fp8"], default="bf16",
                   help="decode weight dtype on the HIP engine (fp8 = "
                        "e4m3 per-row-scaled)")
    p.add_argument("--kv", choices=["bf16", "fp8"], default="bf16",
                   help="KV-cache dtype on the HIP engine (fp8 halves "
                        "cache memory per sample)")
    p.add_argument("--time-run", type=Path, default=None)
    p.add_argument("--seed", type=int, default=10137)
    return p


if __name__ == "__main__":
    args = build_parser().parse_args()
    if args.debug:
        from mdi_llm_amd.utils.console import setup_debug_logging

        setup_debug_logging("starter", SCRIPT_DIR / "logs")
        prof = cProfile.Profile()
        prof.enable()
        main(args)
        prof.disable()
        out = SCRIPT_DIR / "logs" / "starter_profile.prof"
        out.parent.mkdir(exist_ok=True)
        prof.dump_stats(out)
    else:
        main(args)
