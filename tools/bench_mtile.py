#!/usr/bin/env python3
"""A/B the hand-written M-tile GEMM vs hipBLASLt on the grouped-decode
shapes (Llama-3-8B), plus GB/s of the weight stream."""

import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, ".")
from mdi_llm_amd.ops import require_hip_ops  # noqa: E402

DEV = "cuda:0"
SHAPES = [  # (name, K, M)
    ("qkv", 4096, 6144),
    ("proj", 4096, 4096),
    ("gate", 4096, 14336),
    ("down", 14336, 4096),
    ("head", 4096, 128256),
]


def bench(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ops = require_hip_ops()
    for B in (32, 64, 128):
        for name, K, M in SHAPES:
            X = torch.randn(B, K, device=DEV, dtype=torch.bfloat16)
            W = torch.randn(M, K, device=DEV, dtype=torch.bfloat16) * 0.02
            Y = torch.zeros(B, M, device=DEV, dtype=torch.bfloat16)
            gb = M * K * 2 / 1e9
            t_blas = bench(lambda: F.linear(X, W))
            try:
                t_mine = bench(lambda: ops.mtile_gemm(Y, W, X, None, None))
            except Exception as e:  # noqa: BLE001
                print(f"B={B:4d} {name:5s} mtile unsupported: {e}")
                continue
            print(f"B={B:4d} {name:5s} K={K:6d} M={M:6d} "
                  f"hipblaslt {t_blas*1e6:8.1f} us ({gb/t_blas:6.2f} GB/s)  "
                  f"mtile {t_mine*1e6:8.1f} us ({gb/t_mine:6.2f} GB/s)  "
                  f"ratio {t_blas/t_mine:5.2f}x")


if __name__ == "__main__":
    main()
