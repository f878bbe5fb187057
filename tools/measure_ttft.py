#!/usr/bin/env python3
"""Time-to-first-token (prefill latency) for the serving record."""

import sys
import time

import torch

sys.path.insert(0, ".")
from mdi_llm_amd.config import ModelConfig  # noqa: E402
from mdi_llm_amd.models.stages import build_stage  # noqa: E402
from mdi_llm_amd.parallel.runner import make_runner  # noqa: E402


def main(model="Meta-Llama-3-8B-Instruct", lens=(128, 1024, 4096, 8000)):
    cfg = ModelConfig.from_name(model)
    torch.set_default_dtype(torch.bfloat16)
    with torch.device("cuda:0"):
        stage = build_stage(cfg, 0, cfg.n_layer)
    torch.set_default_dtype(torch.float32)
    with torch.no_grad():
        for p in stage.parameters():
            p.normal_(0.0, 0.02)
    stage.max_seq_length = min(8192, cfg.block_size)
    stage.eval()
    runner = make_runner(stage, 1, torch.device("cuda:0"))
    print("backend", runner.backend)
    for T in lens:
        if T >= cfg.block_size:
            continue
        prompt = torch.randint(0, cfg.vocab_size - 1, (T,), device="cuda:0")
        for rep in range(3):
            runner.reset()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            x = runner.prefill_head(prompt, 0)
            logits = runner.tail(x.view(-1, x.size(-1))[-1])
            _ = int(logits.float().argmax())
            torch.cuda.synchronize()
            dt = time.perf_counter() - t0
        print(f"T={T:6d} ttft {dt*1000:8.1f} ms "
              f"({T/dt:9.0f} prefill tok/s)")


if __name__ == "__main__":
    main(*sys.argv[1:2])
