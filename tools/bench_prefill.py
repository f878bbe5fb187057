#!/usr/bin/env python3
"""A/B: HIP prefill (rope-append + MFMA causal flash attention) vs the
torch stage prefill (matmul+softmax, S^2 scores) on Llama-3-8B."""

import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    from mdi_llm_amd.config import ModelConfig
    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.ops.engine import DecodeEngine

    dev = torch.device("cuda:0")
    cfg = ModelConfig.from_name("Meta-Llama-3-8B-Instruct")
    torch.manual_seed(0)
    torch.set_default_dtype(torch.bfloat16)
    with torch.device(dev):
        stage = StarterStage(cfg, cfg.n_layer)
    torch.set_default_dtype(torch.float32)
    with torch.no_grad():
        for p in stage.parameters():
            p.normal_(0, 0.02)
    stage.max_seq_length = 4096
    stage.eval()
    stage.set_kv_cache(1)
    eng = DecodeEngine(stage, stage.kv_pool, use_graphs=False)

    for T in (512, 2048, 4096):
        toks = torch.randint(0, 128000, (T,), device=dev)

        def hip():
            eng.prefill_prompt(toks, 0, 0)

        def ref():
            with torch.inference_mode():
                stage.forward_head(toks.view(1, -1), slot=0, input_pos=0)

        for name, fn in (("hip", hip), ("torch", ref)):
            fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            n = 3
            for _ in range(n):
                fn()
            torch.cuda.synchronize()
            ms = (time.perf_counter() - t0) / n * 1e3
            print(f"prefill T={T:5d} {name:>6}: {ms:8.2f} ms", flush=True)


if __name__ == "__main__":
    main()
