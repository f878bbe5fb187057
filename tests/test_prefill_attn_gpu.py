"""Direct numerics for prefill_attn (the v2 GQA/MFMA causal flash kernel)
across the geometry space its block mapping special-cases: qpk>=4 (4 query
heads per workgroup), qpk 2 / 1 (multiple q tiles per workgroup with
divergent causal bounds), qpk 3 (an idle wave), qpk not a multiple of 4
(remainder head group), head sizes 64/128/256, ragged T, and pos0>0
continuation.  Reference: fp32 SDPA with the causal mask of
/root/reference/src/sub/model.py:738-751."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _run_case(n_kv, qpk, hs, T, pos0, kv8=False, max_seq=256, seed=0):
    from mdi_llm_amd.ops import require_hip_ops

    ops = require_hip_ops()
    torch.manual_seed(seed)
    n_head = n_kv * qpk
    L, slot, layer = 2, 1, 1
    S = pos0 + T
    assert S <= max_seq
    kpool = torch.randn(2, L, n_kv, max_seq, hs, device=DEV) \
        .to(torch.bfloat16).contiguous()
    vpool = torch.randn(2, L, n_kv, max_seq, hs, device=DEV) \
        .to(torch.bfloat16).contiguous()
    qkv = torch.randn(T, n_kv * (qpk + 2) * hs, device=DEV) \
        .to(torch.bfloat16).contiguous()
    out = torch.empty(T, n_head * hs, device=DEV, dtype=torch.bfloat16)
    scale = 1.0 / (hs ** 0.5)

    if kv8:
        # fill an fp8 cache through the real append path (rope disabled via
        # a 1-D cos/sin => rope_ne 0), then dequantize e4m3 for the ref
        kq = torch.zeros_like(kpool, dtype=torch.uint8)
        vq = torch.zeros_like(vpool, dtype=torch.uint8)
        kscale = torch.zeros(2, L, n_kv, max_seq, device=DEV)
        vscale = torch.zeros_like(kscale)
        qkv_full = torch.randn(S, n_kv * (qpk + 2) * hs, device=DEV) \
            .to(torch.bfloat16).contiguous()
        no_rope = torch.zeros(1, device=DEV, dtype=torch.float32)
        ops.rope_prefill_append(qkv_full, kq, vq, no_rope, no_rope, 0,
                                slot, layer, kscale=kscale, vscale=vscale)
        qkv = qkv_full[pos0:].contiguous()
        ops.prefill_attn(out, qkv, kq, vq, pos0, slot, layer, scale,
                         kscale=kscale, vscale=vscale)
        torch.cuda.synchronize()
        k_all = _e4m3(kq[slot, layer]) * kscale[slot, layer, :, :, None]
        v_all = _e4m3(vq[slot, layer]) * vscale[slot, layer, :, :, None]
    else:
        ops.prefill_attn(out, qkv, kpool, vpool, pos0, slot, layer, scale)
        torch.cuda.synchronize()
        k_all = kpool[slot, layer].float()   # (n_kv, max_seq, hs)
        v_all = vpool[slot, layer].float()
    qv = qkv.float().view(T, n_kv, qpk + 2, hs)
    ref = torch.empty(T, n_head, hs)
    for g in range(n_kv):
        for hj in range(qpk):
            q = qv[:, g, hj]                       # (T, hs)
            k = k_all[g, :S]                       # (S, hs)
            v = v_all[g, :S]
            sc = (q @ k.t()) * scale               # (T, S)
            keys = torch.arange(S).view(1, -1)
            mask = keys > (pos0 + torch.arange(T).view(-1, 1))
            sc = sc.masked_fill(mask.to(DEV), float("-inf"))
            p = torch.softmax(sc, dim=-1)
            ref[:, g * qpk + hj] = (p @ v).cpu()
    got = out.float().view(T, n_head, hs).cpu()
    diff = (got - ref).abs().max()
    assert diff < 0.12, (n_kv, qpk, hs, T, pos0, kv8, float(diff))


def _e4m3(b):
    """OCP e4m3 byte -> float (matches the kernel's fp8_to_f32; the
    NaN pattern 0x7f/0xff never occurs for quantized finite inputs)."""
    b = b.to(torch.int32)
    sign = torch.where((b & 0x80) != 0, -1.0, 1.0).to(b.device)
    e = ((b >> 3) & 0xF).float()
    m = (b & 7).float()
    mag = torch.where(e == 0, m / 8.0 * 2.0 ** -6,
                      (1.0 + m / 8.0) * torch.pow(2.0, e - 7.0))
    return sign * mag


CASES = [
    # (n_kv, qpk, hs, T, pos0) — each exercises a distinct block mapping
    (8, 4, 128, 100, 0),    # llama-8b geometry, ragged T
    (8, 4, 128, 33, 91),    # chat continuation (pos0 > 0)
    (8, 8, 128, 50, 0),     # llama-70b geometry: two head-groups per kv
    (1, 7, 64, 67, 5),      # falcon-style: remainder head group (7 = 4+3)
    (4, 1, 64, 130, 0),     # MHA: 4 q tiles per block, divergent bounds
    (2, 2, 128, 49, 3),     # hpb=2, qtpb=2
    (8, 3, 64, 33, 0),      # hpb=3: one idle wave per block
    (2, 16, 256, 40, 0),    # hs=256, four head-groups per kv head
    (8, 4, 128, 1, 10),     # T=1 incremental prefill
]


@torch.inference_mode()
@pytest.mark.parametrize("n_kv,qpk,hs,T,pos0", CASES)
def test_prefill_attn_geometries(n_kv, qpk, hs, T, pos0):
    _run_case(n_kv, qpk, hs, T, pos0)


@torch.inference_mode()
def test_prefill_attn_fp8_kv():
    _run_case(8, 4, 128, 100, 17, kv8=True)
