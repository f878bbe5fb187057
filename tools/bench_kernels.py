#!/usr/bin/env python3
"""Micro-benchmark of the decode kernels (per-shape GEMV variants, attn,
combine) on one MI355X.  Prints achieved TB/s per variant so per-shape
`rows` choices in the engine can be set from measurement, not guesses.

Usage (on a GPU box):  python tools/bench_kernels.py
"""

import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mdi_llm_amd.ops import require_hip_ops  # noqa: E402

DEV = "cuda:0"
ops = require_hip_ops()


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(iters):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / iters * 1e3  # us


def bench_gemv():
    shapes = [
        ("qkv", 6144, 4096, 1),     # norm fused
        ("proj", 4096, 4096, 0),
        ("down", 4096, 14336, 0),
        ("lm_head", 128256, 4096, 1),
        ("fc70b", 8192, 8192, 1),
        ("down70b", 8192, 28672, 0),
    ]
    print(f"{'shape':>10} {'M':>7} {'K':>6} norm | " +
          " | ".join(f"rows={r}: us / TB/s" for r in (1, 2, 4)))
    for name, M, K, nk in shapes:
        W = torch.randn(M, K, device=DEV).to(torch.bfloat16) * 0.02
        x = torch.randn(K, device=DEV).to(torch.bfloat16)
        g = torch.ones(K, device=DEV, dtype=torch.bfloat16)
        out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
        gb = M * K * 2 / 1e12
        row_res = []
        for rows in (1, 2, 4):
            us = timeit(lambda: ops.gemv(out, W, x, None, None, 0,
                                         g if nk else None, None, nk, 1e-5,
                                         rows))
            row_res.append(f"{us:7.1f} / {gb/(us*1e-6):5.2f}")
        print(f"{name:>10} {M:>7} {K:>6} {nk:>4} | " + " | ".join(row_res))


def bench_swiglu():
    for name, M, K in [("swiglu8b", 14336, 4096), ("swiglu70b", 28672, 8192)]:
        Wg = torch.randn(M, K, device=DEV).to(torch.bfloat16) * 0.02
        Wu = torch.randn(M, K, device=DEV).to(torch.bfloat16) * 0.02
        x = torch.randn(K, device=DEV).to(torch.bfloat16)
        g = torch.ones(K, device=DEV, dtype=torch.bfloat16)
        out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
        gb = 2 * M * K * 2 / 1e12
        us = timeit(lambda: ops.gemv_swiglu(out, Wg, Wu, x, False, g, None,
                                            1, 1e-5))
        print(f"{name}: {us:.1f} us  {gb/(us*1e-6):.2f} TB/s")


def bench_attn():
    import math
    n_kv, qpk, hs = 8, 4, 128
    n_head = n_kv * qpk
    max_seq = 8192
    n_chunks = 32
    from mdi_llm_amd.models.model import build_rope_cache

    kpool = torch.randn(1, 1, n_kv, max_seq, hs, device=DEV).to(torch.bfloat16)
    vpool = torch.randn_like(kpool)
    qkv = torch.randn(n_kv * (qpk + 2) * hs, device=DEV).to(torch.bfloat16)
    out = torch.empty(n_head * hs, device=DEV, dtype=torch.bfloat16)
    part_o = torch.empty(n_head * n_chunks * hs, device=DEV)
    part_ml = torch.empty(n_head * n_chunks * 2, device=DEV)
    slot = torch.zeros(1, device=DEV, dtype=torch.int32)
    cos, sin = build_rope_cache(max_seq, hs, device=DEV)
    cos, sin = cos.contiguous(), sin.contiguous()
    for S in (128, 512, 2048, 8192):
        pos = torch.tensor([S - 1], device=DEV, dtype=torch.int32)
        us = timeit(lambda: ops.attn_decode(out, part_o, part_ml, qkv, kpool,
                                            vpool, cos, sin, pos, slot, 0,
                                            n_chunks, 1 / math.sqrt(hs)))
        gb = 2 * n_kv * S * hs * 2 / 1e12
        print(f"attn S={S:5d}: {us:7.1f} us  KV-read {gb/(us*1e-6):5.2f} TB/s"
              f"  (incl. combine)")


def bench_sampling():
    from mdi_llm_amd.models.sampling import sample
    logits = torch.randn(128256, device=DEV, dtype=torch.bfloat16)
    gen = torch.Generator(device=DEV)
    gen.manual_seed(0)
    us = timeit(lambda: sample(logits, 0.8, 200, 1.0, gen), iters=20)
    print(f"torch sampling (topk200): {us:.1f} us")
    us = timeit(lambda: logits.float().argmax(), iters=20)
    print(f"argmax: {us:.1f} us")
    scratch = torch.zeros(520, device=DEV, dtype=torch.int32)
    out = torch.zeros(1, device=DEV, dtype=torch.int32)
    pos = torch.zeros(1, device=DEV, dtype=torch.int32)

    def fused():
        scratch.zero_()
        ops.sample(out, logits, scratch, 0.8, 200, True, 7, pos=pos)

    us = timeit(fused, iters=50)
    print(f"fused HIP sampling (topk200): {us:.1f} us")


if __name__ == "__main__":
    torch.manual_seed(0)
    bench_gemv()
    bench_swiglu()
    bench_attn()
    bench_sampling()
