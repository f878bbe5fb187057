import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch, torch.nn.functional as F
from mdi_llm_amd.ops import require_hip_ops
ops = require_hip_ops()
DEV = "cuda:0"
torch.manual_seed(0)

def timeit(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    import time; t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

K, I = 4096, 14336
Wg = (torch.randn(I, K, device=DEV) * 0.02).to(torch.bfloat16).contiguous()
Wu = (torch.randn(I, K, device=DEV) * 0.02).to(torch.bfloat16).contiguous()
for B in (64, 128):
    X = torch.randn(B, K, device=DEV).to(torch.bfloat16).contiguous()
    out = torch.empty(B, I, device=DEV, dtype=torch.bfloat16)
    ops.gemm_swiglu(out, Wg, Wu, X, False)
    ref = F.silu(X.float() @ Wg.float().t()) * (X.float() @ Wu.float().t())
    d = (out.float() - ref).abs().max().item()
    rel = d / ref.abs().max().item()
    print(f"B={B} maxdiff {d:.4f} (rel {rel:.4f})")
    us_mine = timeit(lambda: ops.gemm_swiglu(out, Wg, Wu, X, False))
    def blas():
        g = F.linear(X, Wg); u = F.linear(X, Wu)
        ops.swiglu_mul(u, g, u, False)
    us_blas = timeit(blas)
    gb = 2 * I * K * 2 / 1e9
    print(f"B={B}: fused {us_mine:7.1f} us ({gb/us_mine*1e3:5.2f} TB/s)  "
          f"blas-pair {us_blas:7.1f} us ({gb/us_blas*1e3:5.2f} TB/s)")
