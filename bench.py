#!/usr/bin/env python3
"""Flagship benchmark: Llama-3-8B recurrent-pipeline decode on N MI355X GPUs.

Measures the BASELINE.json metric — generation throughput (tokens/sec,
whole job) of Llama-3-8B over N pipeline stages with N concurrent samples —
on synthetic prompts and random-init weights (no network for checkpoints).

Contract (driver):
  python bench.py --gpus N --steps K --warmup W
N>1 is launched via torch.distributed.run with one rank per GPU (RCCL).
One decode "step" = one full pipeline rotation = every sample advances one
token (N tokens whole-job per step).  Warmup = prefill + W untimed
rotations.  The timed region is exactly K rotations bracketed by
barrier + torch.cuda.synchronize() on both sides; the reported time is the
MAX over ranks.  Rank 0 prints ONE JSON line.
"""

import argparse
import json
import os
import sys
import time

import torch


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=24)
    ap.add_argument("--warmup", type=int, default=4)
    ap.add_argument("--model", default="Meta-Llama-3-8B-Instruct")
    ap.add_argument("--prompt-len", type=int, default=128)
    ap.add_argument("--samples", type=int, default=0,
                    help="in-flight samples (default = n stages * group)")
    ap.add_argument("--group-size", type=int, default=1,
                    help="batch B samples per stage pass (grouped decode; "
                         "weights stream once per B tokens)")
    ap.add_argument("--seq-len", type=int, default=2048,
                    help="max sequence length (KV budget)")
    ap.add_argument("--backend", choices=["hip", "torch"], default="hip")
    ap.add_argument("--weights", choices=["bf16", "fp8"], default="bf16",
                    help="decode weight dtype (fp8 = e4m3 per-row-scaled; "
                         "NOT the headline config — reported in dtype)")
    ap.add_argument("--kv", choices=["bf16", "fp8"], default="bf16",
                    help="KV-cache dtype (fp8 = e4m3 per-row-scaled cache; "
                         "NOT the headline config — reported in dtype)")
    ap.add_argument("--no-graphs", action="store_true")
    return ap.parse_args()


def log(msg):
    print(f"[bench] {msg}", file=sys.stderr, flush=True)


def main():
    args = parse_args()
    if args.weights == "fp8":
        os.environ["MDI_WEIGHT_DTYPE"] = "fp8"
    if args.kv == "fp8":
        os.environ["MDI_KV_DTYPE"] = "fp8"
    from mdi_llm_amd.config import ModelConfig
    from mdi_llm_amd.models.stages import build_stage
    from mdi_llm_amd.parallel.ring import RingComm
    from mdi_llm_amd.parallel.runner import make_runner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams

    # ---- distributed setup ---------------------------------------------
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    if world == 1 and args.gpus > 1:
        log("WARNING: --gpus>1 but WORLD_SIZE=1; launch via torchrun")
    n_stages = world
    on_gpu = torch.cuda.is_available()
    share_gpu = on_gpu and world > torch.cuda.device_count()
    if on_gpu:
        # more ranks than GPUs (single-box pipeline validation): ranks
        # share devices round-robin and the ring stages via gloo (RCCL
        # refuses two ranks on one device)
        local = int(os.environ.get("LOCAL_RANK", 0))
        device = torch.device(f"cuda:{local % torch.cuda.device_count()}")
        torch.cuda.set_device(device)
    else:
        # CPU fallback so the exact driver invocation is testable without a
        # GPU (gloo ring, torch backend, fp32)
        device = torch.device("cpu")
        log("no GPU: falling back to CPU/gloo/torch-backend (testing mode)")
        args.backend = "torch"

    import torch.distributed as dist

    cpu_group = None
    if world > 1:
        backend = "nccl" if (on_gpu and not share_gpu) else "gloo"
        backend = os.environ.get("MDI_PP_BACKEND", backend)
        dist.init_process_group(backend)
        cpu_group = dist.new_group(backend="gloo")

    B = max(args.group_size, 1)
    n_samples = args.samples or max(n_stages, 1) * B
    cfg = ModelConfig.from_name(args.model)
    need = args.prompt_len + args.warmup + args.steps + 2
    if need > min(args.seq_len, cfg.block_size):
        raise SystemExit(
            f"[bench] prompt+warmup+steps = {need} exceeds the sequence "
            f"budget {min(args.seq_len, cfg.block_size)}; raise --seq-len "
            "or lower --steps"
        )

    # ---- build this rank's stage (random init, bf16, on device) ---------
    from mdi_llm_amd.utils.partition import balanced_split

    split = balanced_split(cfg, n_stages)
    if rank == 0:
        log(f"layer split: {split}")
    t0 = time.time()
    torch.manual_seed(1234 + rank)
    model_dtype = torch.bfloat16 if on_gpu else torch.float32
    if on_gpu:
        # build directly on-device in bf16 (a 70B stage would not survive
        # an fp32 detour: 2x the bytes)
        torch.set_default_dtype(model_dtype)
        with torch.device(device):
            stage = build_stage(cfg, rank, split[rank])
        torch.set_default_dtype(torch.float32)
    else:
        stage = build_stage(cfg, rank, split[rank]).to(dtype=model_dtype)
    with torch.no_grad():
        for p in stage.parameters():
            p.normal_(0.0, 0.02)
    stage.max_seq_length = min(args.seq_len, cfg.block_size)
    stage.eval()
    log(f"rank {rank}: stage built ({split[rank]} layers) "
        f"in {time.time()-t0:.1f}s")

    runner = make_runner(
        stage, n_samples, device, use_graphs=not args.no_graphs,
        force_torch=args.backend == "torch",
        expected_s=args.prompt_len + args.warmup + args.steps + 2,
    )
    log(f"rank {rank}: runner backend={runner.backend}")
    if args.backend == "hip" and runner.backend != "hip":
        raise RuntimeError(
            f"HIP decode engine unavailable for {cfg.name} — refusing to "
            "silently bench the torch fallback"
        )

    comm = None
    if world > 1:
        comm = RingComm(cfg.n_embd, stage.max_seq_length, device, n_samples,
                        dtype=model_dtype)
    rt = PipelineRuntime(runner, rank=rank, world=world, comm=comm,
                         device=device)

    # grouped decode engine (B samples per stage pass)
    geng = None
    group_slots = None
    if B > 1:
        from mdi_llm_amd.ops.group_engine import (
            GroupDecodeEngine,
            group_engine_supported,
        )

        if not (on_gpu and runner.backend == "hip"
                and group_engine_supported(cfg)):
            raise RuntimeError("--group-size>1 requires the HIP engine and "
                               "a llama-family config on GPU")
        G = n_samples // B
        geng = GroupDecodeEngine(stage, stage.kv_pool, B)
        group_slots = [
            torch.arange(g * B, (g + 1) * B, device=device,
                         dtype=torch.int32)
            for g in range(G)
        ]
        if world > 1:
            comm.alloc_groups(G, B)

    def sync_barrier():
        if on_gpu:
            torch.cuda.synchronize()
        if world > 1:
            dist.barrier(group=cpu_group)
            if on_gpu:
                torch.cuda.synchronize()

    sampling = SamplingParams(temperature=0.8, top_k=200, seed=1234)
    gens = rt._generators(sampling, n_samples, device)

    # ---- warmup: prefill + W rotations -----------------------------------
    torch.manual_seed(99)
    prompts = [
        torch.randint(0, cfg.vocab_size - 1, (args.prompt_len,))
        for _ in range(n_samples)
    ]
    t0 = time.time()
    if geng is not None:
        n_groups = max(1, n_samples // B)
        lanes = 1
        if world == 1:
            import os as _os
            lanes = max(1, min(int(_os.environ.get("MDI_LANES", "0") or 0)
                               or 2, n_groups))
        geng.ensure_graphs(sampling.temperature, sampling.top_k,
                           sampling.seed or 0, n_lanes=lanes)
        runner.reset()
    if rank == 0:
        if geng is None:
            rt.prepare_bench(sampling, n_samples)
        toks = rt.bench_prefill(prompts)
        if geng is not None:
            for si in range(n_samples):
                geng.set_slot_pos(si, runner.pos[si])
                geng.token_table[si] = toks[si].view(())
            if args.warmup > 0:
                rt.bench_group_rounds(geng, group_slots, args.warmup)
        elif args.warmup > 0:
            toks = rt.bench_decode_rounds(toks, args.warmup, sampling, gens)
    else:
        rt.bench_serve_prefill(n_samples)
        if geng is not None:
            for si in range(n_samples):
                geng.set_slot_pos(si, runner.pos[si])
            if args.warmup > 0:
                rt.bench_group_serve(geng, group_slots, args.warmup)
        elif args.warmup > 0:
            rt.bench_serve_rounds(n_samples, args.warmup)
    sync_barrier()
    log(f"rank {rank}: warmup done in {time.time()-t0:.1f}s")

    # ---- timed region: exactly K rotations -------------------------------
    t_start = time.perf_counter()
    if rank == 0:
        if geng is not None:
            rt.bench_group_rounds(geng, group_slots, args.steps)
        else:
            toks = rt.bench_decode_rounds(toks, args.steps, sampling, gens)
    else:
        if geng is not None:
            rt.bench_group_serve(geng, group_slots, args.steps)
        else:
            rt.bench_serve_rounds(n_samples, args.steps)
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t_start
    sync_barrier()

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX, group=cpu_group)
        elapsed = float(t[0])

    if rank == 0:
        total_tokens = n_samples * args.steps
        tps = total_tokens / elapsed
        metric = ("llama3_8b_pipeline_decode_tok_per_s"
                  if "Llama-3-8B" in cfg.name
                  else f"{cfg.name}_pipeline_decode_tok_per_s")
        out = {
            "metric": metric,
            "value": round(tps, 2),
            "unit": "tokens/s",
            "n_gpus": n_stages,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": (("bf16_act_fp8_weights" if args.weights == "fp8"
                       else ("bf16" if on_gpu else "float32"))
                      + ("_fp8kv" if args.kv == "fp8" else "")),
            "data": "synthetic",
            "config": {
                "model": cfg.name,
                "global_batch": n_samples,
                "seq_len": args.prompt_len,
                "max_seq_len": stage.max_seq_length,
                "parallelism": f"pp{n_stages}",
                "samples_in_flight": n_samples,
                "group_size": B,
                "backend": runner.backend,
            },
        }
        print(json.dumps(out), flush=True)

    if world > 1:
        dist.barrier(group=cpu_group)
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
