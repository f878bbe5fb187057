#!/usr/bin/env python3
"""Training / finetuning CLI with DDP over RCCL.

Capability parity with /root/reference/src/train.py: memmapped token data,
init from scratch / resume / pretrained litGPT checkpoint, gradient
accumulation with DDP sync gating, AdamW + cosine LR with warmup, gradient
clipping, AMP (GradScaler for fp16; bf16 autocast natively), periodic eval
+ checkpointing (``train_ckpt.pkl`` with optimizer state + ``lit_model.pth``),
MFU logging.  Launch multi-GPU via:
  torchrun --nproc-per-node N --master-addr 127.0.0.1 train.py ...
(``nccl`` backend IS RCCL on ROCm.)
"""

import argparse
import math
import os
import pickle
import sys
import time
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))

import numpy as np
import torch


def get_lr(it, *, lr, warmup, decay_iters, min_lr):
    """Cosine schedule with linear warmup (reference utils/utils.py:110)."""
    if it < warmup:
        return lr * it / warmup
    if it > decay_iters:
        return min_lr
    ratio = (it - warmup) / (decay_iters - warmup)
    coeff = 0.5 * (1.0 + math.cos(math.pi * ratio))
    return min_lr + coeff * (lr - min_lr)


def main(args):
    import torch.distributed as dist
    from torch.nn.parallel import DistributedDataParallel as DDP

    from mdi_llm_amd.config import ModelConfig
    from mdi_llm_amd.models.model import GPT
    from mdi_llm_amd.utils.data import get_batch, load_bin

    # ---- DDP setup (reference train.py:88-103) --------------------------
    ddp = int(os.environ.get("RANK", -1)) != -1
    if ddp:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
        rank = dist.get_rank()
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
        world = dist.get_world_size()
        device = torch.device(
            f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"
        )
        if device.type == "cuda":
            torch.cuda.set_device(device)
        assert args.grad_accum % world == 0
        args.grad_accum //= world
        seed_offset = rank
    else:
        rank, world, seed_offset = 0, 1, 0
        device = torch.device(
            args.device or ("cuda" if torch.cuda.is_available() else "cpu")
        )
    master = rank == 0
    torch.manual_seed(args.seed + seed_offset)

    dtype = {"float32": torch.float32, "bfloat16": torch.bfloat16,
             "float16": torch.float16}[args.dtype]
    autocast = torch.autocast(device_type=device.type, dtype=dtype,
                              enabled=dtype != torch.float32)
    scaler = torch.amp.GradScaler(enabled=dtype == torch.float16)

    # ---- data -----------------------------------------------------------
    train_data = load_bin(Path(args.data_dir) / "train.bin")
    val_data = load_bin(Path(args.data_dir) / "val.bin")

    # ---- model init: scratch / resume / finetune ------------------------
    ckpt_dir = Path(args.ckpt)
    config = ModelConfig.from_checkpoint(ckpt_dir)
    if args.sequence_length:
        config.block_size = args.sequence_length
    iter_num, best_val_loss = 0, float("inf")
    model = GPT(config)
    train_ckpt = ckpt_dir / "train_ckpt.pkl"
    if args.init == "scratch":
        model.apply_init()
    elif args.init in ("resume", "finetune"):
        sd = torch.load(ckpt_dir / "lit_model.pth", map_location="cpu",
                        weights_only=True)
        model.load_state_dict(sd)
        if args.init == "resume" and train_ckpt.is_file():
            with open(train_ckpt, "rb") as fp:
                state = pickle.load(fp)
            iter_num = state["iter_num"]
            best_val_loss = state["best_val_loss"]
    model = model.to(device)
    model.max_seq_length = min(config.block_size, args.block_size)

    if args.tie_weights:
        model.transformer.wte.weight = model.lm_head.weight

    optimizer = torch.optim.AdamW(
        model.parameters(), lr=args.lr, betas=(0.9, 0.95),
        weight_decay=args.weight_decay,
    )
    if args.init == "resume" and train_ckpt.is_file():
        with open(train_ckpt, "rb") as fp:
            state = pickle.load(fp)
        if "optimizer" in state:
            optimizer.load_state_dict(state["optimizer"])

    raw_model = model
    if ddp:
        model = DDP(model, device_ids=[device.index] if
                    device.type == "cuda" else None)

    gen = torch.Generator().manual_seed(args.seed + 7 + seed_offset)

    def batch(split):
        data = train_data if split == "train" else val_data
        return get_batch(data, args.batch_size, model_block(), device, gen)

    def model_block():
        return raw_model.max_seq_length

    @torch.no_grad()
    def estimate_loss():
        raw_model.eval()
        out = {}
        for split in ("train", "val"):
            losses = torch.zeros(args.eval_iters)
            for i in range(args.eval_iters):
                X, Y = batch(split)
                with autocast:
                    logits = raw_model(X)
                    losses[i] = torch.nn.functional.cross_entropy(
                        logits.view(-1, logits.size(-1)), Y.view(-1)
                    ).item()
            out[split] = losses.mean().item()
        raw_model.train()
        return out

    # ---- training loop (reference train.py:272-370) ---------------------
    model.train()
    X, Y = batch("train")
    t0 = time.time()
    peak_flops = 2.5e15 if device.type == "cuda" else 1e12  # MI355X bf16 dense
    while iter_num <= args.max_iters:
        lr = get_lr(iter_num, lr=args.lr, warmup=args.warmup_iters,
                    decay_iters=args.max_iters, min_lr=args.lr / 10)
        for g in optimizer.param_groups:
            g["lr"] = lr

        if iter_num % args.eval_interval == 0 and master:
            losses = estimate_loss()
            print(f"iter {iter_num}: train {losses['train']:.4f} "
                  f"val {losses['val']:.4f} lr {lr:.2e}", flush=True)
            if losses["val"] < best_val_loss or args.always_save:
                best_val_loss = losses["val"]
                if iter_num > 0:
                    torch.save(raw_model.state_dict(),
                               ckpt_dir / "lit_model.pth")
                    with open(train_ckpt, "wb") as fp:
                        pickle.dump(
                            {
                                "optimizer": optimizer.state_dict(),
                                "iter_num": iter_num,
                                "best_val_loss": best_val_loss,
                                "train_settings": vars(args),
                                "config": config.to_dict(),
                            },
                            fp,
                        )
                    config.save(ckpt_dir / "model_config.yaml")

        for micro in range(args.grad_accum):
            if ddp:
                model.require_backward_grad_sync = (
                    micro == args.grad_accum - 1
                )
            with autocast:
                logits = model(X)
                loss = torch.nn.functional.cross_entropy(
                    logits.view(-1, logits.size(-1)), Y.view(-1)
                ) / args.grad_accum
            X, Y = batch("train")
            scaler.scale(loss).backward()
        if args.grad_clip > 0:
            scaler.unscale_(optimizer)
            torch.nn.utils.clip_grad_norm_(model.parameters(), args.grad_clip)
        scaler.step(optimizer)
        scaler.update()
        optimizer.zero_grad(set_to_none=True)

        if iter_num % args.log_interval == 0 and master:
            dt = time.time() - t0
            t0 = time.time()
            mfu = raw_model.estimate_mfu(
                args.batch_size * args.grad_accum,
                max(dt / max(args.log_interval, 1), 1e-9), peak_flops,
            )
            print(f"iter {iter_num}: loss {loss.item()*args.grad_accum:.4f} "
                  f"time {dt*1000/max(args.log_interval,1):.0f}ms/iter "
                  f"mfu {mfu*100:.2f}%", flush=True)
        iter_num += 1

    if ddp:
        dist.destroy_process_group()


if __name__ == "__main__":
    p = argparse.ArgumentParser(description="Train/finetune (DDP over RCCL)")
    p.add_argument("--ckpt", type=Path, required=True,
                   help="model dir (model_config.yaml [+ lit_model.pth])")
    p.add_argument("--data-dir", type=Path, required=True)
    p.add_argument("--init", choices=("scratch", "resume", "finetune"),
                   default="scratch")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--dtype", default="bfloat16",
                   choices=("float32", "bfloat16", "float16"))
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--block-size", type=int, default=512)
    p.add_argument("--sequence-length", type=int, default=None)
    p.add_argument("--grad-accum", type=int, default=8)
    p.add_argument("--max-iters", type=int, default=1000)
    p.add_argument("--lr", type=float, default=6e-4)
    p.add_argument("--weight-decay", type=float, default=0.1)
    p.add_argument("--grad-clip", type=float, default=1.0)
    p.add_argument("--warmup-iters", type=int, default=100)
    p.add_argument("--eval-interval", type=int, default=200)
    p.add_argument("--eval-iters", type=int, default=20)
    p.add_argument("--log-interval", type=int, default=10)
    p.add_argument("--tie-weights", action="store_true")
    p.add_argument("--always-save", action="store_true")
    p.add_argument("--seed", type=int, default=1337)
    main(p.parse_args())
