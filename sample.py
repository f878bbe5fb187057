#!/usr/bin/env python3
"""Single-device generation CLI (the reference's 1-node baseline curve).

Capability parity with /root/reference/src/sample.py: checkpoint
auto-convert, dtype inference, KV-cached generation of N samples with
tok/time CSV, cProfile under --debug.
"""

import argparse
import cProfile
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

    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.parallel.orchestrator import default_dtype
    from mdi_llm_amd.parallel.runner import make_runner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams
    from mdi_llm_amd.prompts import (
        get_user_prompt,
        has_prompt_style,
        load_prompt_style,
        model_name_to_prompt_style,
    )
    from mdi_llm_amd.tokenizer import Tokenizer
    from mdi_llm_amd.utils.checkpoint import get_checkpoint_files, load_from_pt
    from mdi_llm_amd.utils.plots import tok_time_csv_name, write_tok_time_csv

    torch.manual_seed(args.seed)
    ckpt = args.ckpt
    cfg_file, model_file = get_checkpoint_files(ckpt)
    if not model_file.is_file():
        # auto-convert an HF checkpoint dir (reference sample.py:66-76)
        from mdi_llm_amd.utils.convert_hf import convert_hf_checkpoint

        print(f"[sample] converting HF checkpoint in {ckpt}")
        convert_hf_checkpoint(ckpt)

    device = torch.device(
        args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    )
    dtype = default_dtype(args.dtype)
    config, sd = load_from_pt(ckpt)
    stage = StarterStage(config, config.n_layer)
    stage.load_state_dict(sd)
    stage = stage.to(device=device, dtype=dtype)
    if args.sequence_length:
        stage.max_seq_length = min(args.sequence_length, config.block_size)
    stage.eval()

    runner = make_runner(stage, args.n_samples, device)
    rt = PipelineRuntime(runner, device=device)
    print(f"[sample] backend={runner.backend} device={device} dtype={dtype}")

    tokenizer = Tokenizer(ckpt)
    style = (load_prompt_style(ckpt) if has_prompt_style(ckpt)
             else model_name_to_prompt_style(config.name))
    stop_tokens = style.stop_tokens(tokenizer)
    prompts = [
        tokenizer.encode(style.apply(p), device=device)
        for p in get_user_prompt(args.prompt, args.n_samples)
    ]

    t0 = time.time()
    res = rt.generate(
        prompts, args.n_tokens,
        SamplingParams(args.temperature, args.top_k, args.top_p, args.seed),
        stop_tokens=stop_tokens,
    )
    for i, seq in enumerate(res.sequences):
        print(f"\n========== sample {i} ==========\n{tokenizer.decode(seq)}")
    print(f"\n[sample] {res.total_new_tokens} tokens in {res.gen_time:.2f}s "
          f"({res.tokens_per_second:.2f} tok/s)")

    logs = SCRIPT_DIR / "logs"
    csv_path = logs / tok_time_csv_name(1, config.name, args.n_samples)
    write_tok_time_csv(csv_path, res.tok_time)
    print(f"[sample] tok/time CSV -> {csv_path}")


def build_parser():
    p = argparse.ArgumentParser(description="Single-device generation")
    p.add_argument("-d", "--debug", action="store_true")
    p.add_argument("-v", "--verb", action="store_true")
    p.add_argument("--ckpt", type=Path,
                   default=SCRIPT_DIR / "checkpoints" / "custom" / "NanoLlama")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--dtype", type=str, default=None)
    p.add_argument("--weights", choices=["bf16", "fp8"], default="bf16",
                   help="decode weight dtype on the HIP engine (fp8 = "
                        "e4m3 per-row-scaled)")
    p.add_argument("--kv", choices=["bf16", "fp8"], default="bf16",
                   help="KV-cache dtype on the HIP engine (fp8 halves "
                        "cache memory per sample)")
    p.add_argument("--prompt", type=str, default="Who are you?")
    p.add_argument("--n-samples", type=int, default=1)
    p.add_argument("--n-tokens", type=int, default=300)
    p.add_argument("--temperature", type=float, default=0.8)
    p.add_argument("--top-k", type=int, default=200)
    p.add_argument("--top-p", type=float, default=1.0)
    p.add_argument("--sequence-length", "--context-length", "--block-size",
                   type=int, default=None, dest="sequence_length")
    p.add_argument("--seed", type=int, default=10137)
    return p


if __name__ == "__main__":
    args = build_parser().parse_args()
    if args.debug:
        from mdi_llm_amd.utils.console import setup_debug_logging

        setup_debug_logging("sample", SCRIPT_DIR / "logs")
        prof = cProfile.Profile()
        prof.enable()
        main(args)
        prof.disable()
        out = SCRIPT_DIR / "logs" / "sample_profile.prof"
        out.parent.mkdir(exist_ok=True)
        prof.dump_stats(out)
    else:
        main(args)
