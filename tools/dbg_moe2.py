import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch
from mdi_llm_amd.ops import require_hip_ops
ops = require_hip_ops()
DEV = "cuda:0"
torch.manual_seed(0)
nE, I, E = 4, 224, 256
Wg = torch.randn(nE, I, E, device=DEV).to(torch.bfloat16).contiguous() * 0.1
Wu = torch.randn(nE, I, E, device=DEV).to(torch.bfloat16).contiguous() * 0.1
x = torch.randn(E, device=DEV).to(torch.bfloat16)
act = torch.zeros(I, device=DEV, dtype=torch.bfloat16)
for e in (0, 1, 3):
    eidx = torch.tensor([e], device=DEV, dtype=torch.int32)
    esc = torch.tensor([0.625], device=DEV, dtype=torch.float32)
    ops.gemv_swiglu(act, Wg, Wu, x, False, None, None, 0, 1e-5,
                    eidx=eidx, estride=I * E, escale=esc)
    g = (Wg[e].float() @ x.float())
    u = (Wu[e].float() @ x.float())
    ref = torch.nn.functional.silu(g) * u * 0.625
    print(e, "maxdiff:", (act.float() - ref).abs().max().item())
# down with indirection
Wd = torch.randn(nE, E, I, device=DEV).to(torch.bfloat16).contiguous() * 0.1
res = torch.randn(E, device=DEV).to(torch.bfloat16)
out = torch.zeros(E, device=DEV, dtype=torch.bfloat16)
for e in (2, 3):
    eidx = torch.tensor([e], device=DEV, dtype=torch.int32)
    ops.gemv(out, Wd, act, None, res, 1, None, None, 0, 1e-5, 1,
             eidx=eidx, estride=E * I)
    ref = Wd[e].float() @ act.float() + res.float()
    print("down", e, "maxdiff:", (out.float() - ref).abs().max().item())
# topk
logits = torch.tensor([0.3, -0.2, 0.8, 0.25], device=DEV).to(torch.bfloat16)
ei = torch.zeros(2, device=DEV, dtype=torch.int32)
es = torch.zeros(2, device=DEV, dtype=torch.float32)
ops.moe_gate_topk(ei, es, logits, 2)
pr, idx = torch.topk(logits, 2)
print("topk:", ei.tolist(), idx.tolist(), es.tolist(),
      pr.softmax(0, dtype=torch.float).tolist())
