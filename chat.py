#!/usr/bin/env python3
"""Interactive streaming chat CLI (single device).

Capability parity with /root/reference/src/chat.py: prompt-styled turns,
incremental token streaming to stdout, stop-token truncation, KV cache
persisting across the conversation.
"""

import argparse
import sys
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))


def main(args):
    import os as _os

    if args.weights == "fp8":
        _os.environ["MDI_WEIGHT_DTYPE"] = "fp8"
    if args.kv == "fp8":
        _os.environ["MDI_KV_DTYPE"] = "fp8"
    import torch

    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.parallel.orchestrator import default_dtype
    from mdi_llm_amd.parallel.runner import make_runner
    from mdi_llm_amd.prompts import (
        has_prompt_style,
        load_prompt_style,
        model_name_to_prompt_style,
    )
    from mdi_llm_amd.models.sampling import sample as sample_token
    from mdi_llm_amd.tokenizer import Tokenizer
    from mdi_llm_amd.utils.checkpoint import load_from_pt

    torch.manual_seed(args.seed)
    device = torch.device(
        args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    )
    dtype = default_dtype(args.dtype)
    config, sd = load_from_pt(args.ckpt)
    stage = StarterStage(config, config.n_layer)
    stage.load_state_dict(sd)
    stage = stage.to(device=device, dtype=dtype)
    if args.sequence_length:
        stage.max_seq_length = min(args.sequence_length, config.block_size)
    stage.eval()
    runner = make_runner(stage, 1, device)

    tokenizer = Tokenizer(args.ckpt)
    style = (load_prompt_style(args.ckpt) if has_prompt_style(args.ckpt)
             else model_name_to_prompt_style(config.name))
    stop_tokens = style.stop_tokens(tokenizer)

    print(f"[chat] {config.name} on {device} ({runner.backend}); "
          "Ctrl-D or empty line to exit")
    pos = 0
    while True:
        try:
            user = input(">> ").strip()
        except (EOFError, KeyboardInterrupt):
            print()
            break
        if not user:
            break
        text = style.apply(user)
        toks = tokenizer.encode(text, device=device,
                                bos=(pos == 0))
        if pos + toks.numel() + args.max_new_tokens >= stage.max_seq_length:
            print("[chat] context full — resetting conversation")
            runner.reset()
            pos = 0
        # prefill this turn at the current position
        if pos == 0:
            x = runner.prefill_head(toks, 0)
        elif (hasattr(runner, "engine")
              and runner.engine.supports_hip_prefill):
            # continue the cache through the HIP prefill at the current
            # position (also the only writer the fp8 KV cache allows)
            x = runner.engine.prefill_prompt(toks, 0, pos)
            runner.pos[0] = pos + toks.numel()
            runner.engine.set_slot_pos(0, runner.pos[0])
        else:
            # continue the cache: feed tokens one batch at the current pos
            x = runner.stage.forward_head(toks.view(1, -1), slot=0,
                                          input_pos=pos)
            runner.pos[0] = pos + toks.numel()
            if hasattr(runner, "engine"):
                runner.engine.set_slot_pos(0, runner.pos[0])
            x = x[0]
        pos = runner.pos[0]
        logits = runner.tail(x.view(-1, x.size(-1))[-1])

        generated = []
        prev_decoded = ""
        for _ in range(args.max_new_tokens):
            tok = sample_token(logits, args.temperature, args.top_k)
            generated.append(int(tok))
            # incremental decode: print only the new suffix
            decoded = tokenizer.decode(torch.tensor(generated))
            sys.stdout.write(decoded[len(prev_decoded):])
            sys.stdout.flush()
            prev_decoded = decoded
            if any(len(st) and generated[-len(st):] == list(st)
                   for st in stop_tokens):
                break
            if pos + 1 >= stage.max_seq_length:
                break
            x = runner.decode_head(tok.view(1).to(device), 0)
            pos = runner.pos[0]
            logits = runner.tail(x)
        print()


if __name__ == "__main__":
    p = argparse.ArgumentParser(description="Interactive chat")
    p.add_argument("--ckpt", type=Path,
                   default=SCRIPT_DIR / "checkpoints" / "custom" / "NanoLlama")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--dtype", type=str, default=None)
    p.add_argument("--max-new-tokens", type=int, default=256)
    p.add_argument("--temperature", type=float, default=0.8)
    p.add_argument("--top-k", type=int, default=200)
    p.add_argument("--sequence-length", type=int, default=None)
    p.add_argument("--weights", choices=["bf16", "fp8"], default="bf16",
                   help="decode weight dtype on the HIP engine")
    p.add_argument("--kv", choices=["bf16", "fp8"], default="bf16",
                   help="KV-cache dtype on the HIP engine")
    p.add_argument("--seed", type=int, default=10137)
    p.add_argument("-d", "--debug", action="store_true",
                   help="write debug logs to logs/logs_chat.log")
    _args = p.parse_args()
    if _args.debug:
        from mdi_llm_amd.utils.console import setup_debug_logging

        setup_debug_logging("chat", SCRIPT_DIR / "logs")
    main(_args)
