"""GPU numerics: each hand-written CDNA4 kernel vs a plain PyTorch fp32
reference of the same op (inputs bf16-rounded, reference computed in fp32).
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.fixture(scope="module")
def ops():
    from mdi_llm_amd.ops import require_hip_ops

    return require_hip_ops()


def bf(x):
    return x.to(torch.bfloat16)


def mk(*shape, scale=1.0, seed=None):
    if seed is not None:
        torch.manual_seed(seed)
    return bf(torch.randn(*shape, device=DEV) * scale).contiguous()


# ---------------------------------------------------------------------------
def test_rmsnorm(ops):
    n = 4096
    x = mk(n, seed=0)
    w = mk(n, seed=1)
    out = torch.empty_like(x)
    ops.rmsnorm(out, x, w, 1e-5)
    xf = x.float()
    ref = xf * torch.rsqrt(xf.pow(2).mean() + 1e-5) * w.float()
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_layernorm(ops):
    n = 2048
    x = mk(n, seed=2)
    w = mk(n, seed=3)
    b = mk(n, seed=4)
    out = torch.empty_like(x)
    ops.layernorm(out, x, w, b, 1e-5)
    ref = torch.nn.functional.layer_norm(
        x.float(), (n,), w.float(), b.float(), 1e-5
    )
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2)
    # no-bias path
    ops.layernorm(out, x, w, None, 1e-5)
    ref = torch.nn.functional.layer_norm(x.float(), (n,), w.float(), None, 1e-5)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("M,K", [(4096, 4096), (6144, 4096), (128256, 512),
                                 (176, 64), (64, 176)])
def test_gemv_plain(ops, M, K):
    W = mk(M, K, scale=1.0 / math.sqrt(K), seed=5)
    x = mk(K, seed=6)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv(out, W, x, None, None, 0)
    ref = W.float() @ x.float()
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2), \
        (out.float() - ref).abs().max()


def test_gemv_bias_res(ops):
    M, K = 1024, 2048
    W = mk(M, K, scale=1.0 / math.sqrt(K), seed=7)
    x = mk(K, seed=8)
    bias = mk(M, seed=9)
    res = mk(M, seed=10)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv(out, W, x, bias, res, 1)
    ref = W.float() @ x.float() + bias.float() + res.float()
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_gemv_gelu(ops):
    M, K = 512, 1024
    W = mk(M, K, scale=1.0 / math.sqrt(K), seed=11)
    x = mk(K, seed=12)
    bias = mk(M, seed=13)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv(out, W, x, bias, None, 2)
    ref = torch.nn.functional.gelu(
        W.float() @ x.float() + bias.float(), approximate="tanh"
    )
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_gemv_swiglu(ops):
    M, K = 14336, 4096
    Wg = mk(M, K, scale=1.0 / math.sqrt(K), seed=14)
    Wu = mk(M, K, scale=1.0 / math.sqrt(K), seed=15)
    x = mk(K, seed=16)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv_swiglu(out, Wg, Wu, x, False)
    g = Wg.float() @ x.float()
    u = Wu.float() @ x.float()
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_gemv_fused_rmsnorm(ops):
    M, K = 1024, 4096
    W = mk(M, K, scale=1.0 / math.sqrt(K), seed=30)
    x = mk(K, seed=31)
    g = mk(K, seed=32)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv(out, W, x, None, None, 0, g, None, 1, 1e-5)
    xf = x.float()
    xn = xf * torch.rsqrt(xf.pow(2).mean() + 1e-5) * g.float()
    ref = W.float() @ xn.to(torch.bfloat16).float()
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_gemv_fused_layernorm(ops):
    M, K = 512, 2048
    W = mk(M, K, scale=1.0 / math.sqrt(K), seed=33)
    x = mk(K, seed=34)
    g = mk(K, seed=35)
    b = mk(K, seed=36)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv(out, W, x, None, None, 0, g, b, 2, 1e-5)
    xn = torch.nn.functional.layer_norm(x.float(), (K,), g.float(),
                                        b.float(), 1e-5)
    ref = W.float() @ xn.to(torch.bfloat16).float()
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_swiglu_fused_rmsnorm(ops):
    M, K = 688, 256
    Wg = mk(M, K, scale=1.0 / math.sqrt(K), seed=37)
    Wu = mk(M, K, scale=1.0 / math.sqrt(K), seed=38)
    x = mk(K, seed=39)
    g = mk(K, seed=40)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv_swiglu(out, Wg, Wu, x, False, g, None, 1, 1e-5)
    xf = x.float()
    xn = (xf * torch.rsqrt(xf.pow(2).mean() + 1e-5) * g.float()).to(
        torch.bfloat16).float()
    gg = Wg.float() @ xn
    uu = Wu.float() @ xn
    ref = torch.nn.functional.silu(gg) * uu
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_gemv_fp8(ops):
    M, K = 4096, 4096
    torch.manual_seed(50)
    W = mk(M, K, scale=1.0 / math.sqrt(K), seed=50)
    x = mk(K, seed=51)
    from mdi_llm_amd.ops.engine import quantize_fp8_rowwise

    Wq, s = quantize_fp8_rowwise(W)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv_fp8(out, Wq, s, x, None, None, 0)
    # exact reference of the dequantized weights
    ref = (Wq.float() * s[:, None]) @ x.float()
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2), \
        (out.float() - ref).abs().max()
    # and close to the bf16 weights' result (quantization error bounded)
    full = W.float() @ x.float()
    rel = (out.float() - full).abs().mean() / full.abs().mean()
    assert rel < 0.05, float(rel)


def test_gemv_swiglu_fp8(ops):
    M, K = 2048, 1024
    Wg = mk(M, K, scale=1.0 / math.sqrt(K), seed=52)
    Wu = mk(M, K, scale=1.0 / math.sqrt(K), seed=53)
    x = mk(K, seed=54)
    from mdi_llm_amd.ops.engine import quantize_fp8_rowwise

    Wgq, gs = quantize_fp8_rowwise(Wg)
    Wuq, us = quantize_fp8_rowwise(Wu)
    out = torch.empty(M, device=DEV, dtype=torch.bfloat16)
    ops.gemv_swiglu_fp8(out, Wgq, gs, Wuq, us, x, False)
    g = (Wgq.float() * gs[:, None]) @ x.float()
    u = (Wuq.float() * us[:, None]) @ x.float()
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_engine_fp8_mode():
    """fp8-weight engine decode stays close to the bf16 engine."""
    import os

    from mdi_llm_amd import GPT, ModelConfig
    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.ops.engine import DecodeEngine

    torch.manual_seed(55)
    cfg = ModelConfig.from_name("nano-gpu")
    m = GPT(cfg)
    m.apply_init()
    m = m.to(device=DEV, dtype=torch.bfloat16)
    m.eval()
    stage = StarterStage(cfg, cfg.n_layer).to(device=DEV, dtype=torch.bfloat16)
    stage.load_state_dict(m.state_dict())
    stage.eval()
    stage.set_kv_cache(1)

    prompt = torch.randint(0, 511, (8,), device=DEV)
    stage.forward_head(prompt.view(1, -1), slot=0, input_pos=0)
    os.environ["MDI_WEIGHT_DTYPE"] = "fp8"
    try:
        eng = DecodeEngine(stage, stage.kv_pool, n_chunks=8, use_graphs=False)
    finally:
        del os.environ["MDI_WEIGHT_DTYPE"]
    assert eng.fp8
    eng.set_slot_pos(0, 8)
    tok = torch.tensor([5], device=DEV, dtype=torch.int32)
    x = eng.decode_step_head(tok, 0)
    logits8 = eng.tail(x).float().clone()

    m.set_kv_cache(1)
    m(prompt.view(1, -1), input_pos=0, slot=0)
    ref = m(torch.tensor([[5]], device=DEV), input_pos=8, slot=0)
    ref = ref[0, -1].float()
    rel = (logits8 - ref).abs().mean() / ref.abs().std()
    assert rel < 0.2, float(rel)  # fp8 quantization-level agreement


def test_embed(ops):
    V, E = 1000, 512
    wte = mk(V, E, seed=17)
    tok = torch.tensor([123], device=DEV, dtype=torch.int32)
    out = torch.empty(E, device=DEV, dtype=torch.bfloat16)
    ops.embed(out, wte, tok, 1.0)
    assert torch.equal(out, wte[123])
    ops.embed(out, wte, tok, 2.0)
    assert torch.allclose(out.float(), wte[123].float() * 2.0, atol=2e-2,
                          rtol=2e-2)


def test_add(ops):
    a = mk(4096, seed=18)
    b = mk(4096, seed=19)
    out = torch.empty_like(a)
    ops.add(out, a, b)
    assert torch.allclose(out.float(), a.float() + b.float(), atol=2e-2)


# ---------------------------------------------------------------------------
def _rope_ref(x, cos, sin):
    # rotate-half on the first rope_n_elem dims (model.py apply_rope)
    ne = cos.shape[-1]
    xr = x[..., :ne].float()
    half = ne // 2
    x1, x2 = xr[..., :half], xr[..., half:]
    rot = torch.cat([-x2, x1], dim=-1)
    out = x.clone().float()
    out[..., :ne] = xr * cos + rot * sin
    return out


@pytest.mark.parametrize("qpk,n_kv,hs,ne", [(4, 8, 128, 128), (1, 4, 64, 16),
                                            (16, 1, 64, 64)])
def test_rope_kv_append(ops, qpk, n_kv, hs, ne):
    from mdi_llm_amd.models.model import build_rope_cache

    torch.manual_seed(20)
    max_seq, n_layers, n_slots = 64, 3, 2
    qkv = mk(n_kv * (qpk + 2) * hs, seed=21)
    qkv_orig = qkv.clone()
    kpool = torch.zeros(n_slots, n_layers, n_kv, max_seq, hs, device=DEV,
                        dtype=torch.bfloat16)
    vpool = torch.zeros_like(kpool)
    cos, sin = build_rope_cache(max_seq, ne, device=DEV)
    pos = torch.tensor([7], device=DEV, dtype=torch.int32)
    slot = torch.tensor([1], device=DEV, dtype=torch.int32)
    ops.rope_kv_append(qkv, kpool, vpool, cos.contiguous(), sin.contiguous(),
                       pos, slot, 2)

    grp = qkv_orig.view(n_kv, qpk + 2, hs)
    c, s = cos[7:8], sin[7:8]
    for g in range(n_kv):
        for r in range(qpk + 1):  # q rows + k row get rope
            ref = _rope_ref(grp[g, r].view(1, -1), c, s)[0]
            got = qkv.view(n_kv, qpk + 2, hs)[g, r].float()
            assert torch.allclose(got, ref, atol=2e-2, rtol=2e-2), (g, r)
        # k/v appended at [slot=1, layer=2, g, pos=7]
        kref = _rope_ref(grp[g, qpk].view(1, -1), c, s)[0]
        assert torch.allclose(kpool[1, 2, g, 7].float(), kref, atol=2e-2,
                              rtol=2e-2)
        assert torch.equal(vpool[1, 2, g, 7], grp[g, qpk + 1])
        # nothing else touched
        assert kpool[0].abs().sum() == 0
        assert kpool[1, 2, g, 8:].abs().sum() == 0


# ---------------------------------------------------------------------------
# max_seq <= 4096 exercises the one-launch block-local variant;
# max_seq = 8192 exercises the global split-S + combine path
@pytest.mark.parametrize("qpk,n_kv,hs,S,ne,max_seq,n_chunks", [
    (4, 8, 128, 1, 128, 2048, 32),
    (4, 8, 128, 500, 128, 2048, 32),
    (4, 8, 128, 2048, 128, 2048, 32),
    (8, 8, 128, 333, 128, 2048, 32),
    (1, 8, 64, 100, 64, 2048, 32),
    (2, 4, 128, 77, 0, 2048, 32),      # no rope (learned-pos models)
    (16, 2, 64, 129, 16, 2048, 32),    # partial rotary
    (4, 8, 128, 500, 128, 8192, 32),   # split-S path, short S
    (4, 8, 128, 2048, 128, 8192, 32),  # split-S path
    (4, 8, 128, 2048, 128, 8192, 128), # >64 active chunks in the combine
])
def test_attn_decode(ops, qpk, n_kv, hs, S, ne, max_seq, n_chunks):
    """Fused kernel: ropes q and the current k from the RAW qkv buffer,
    attends over pool[0..S-2] + current, and appends k/v at pos."""
    from mdi_llm_amd.models.model import build_rope_cache

    torch.manual_seed(22)
    n_head = n_kv * qpk
    n_layers, n_slots, layer, slot_i = 2, 2, 1, 1
    pos_i = S - 1

    kpool = mk(n_slots, n_layers, n_kv, max_seq, hs, seed=23)
    vpool = mk(n_slots, n_layers, n_kv, max_seq, hs, seed=24)
    qkv = mk(n_kv * (qpk + 2) * hs, seed=25)
    cos, sin = build_rope_cache(max_seq, ne, device=DEV)
    cos, sin = cos.contiguous(), sin.contiguous()
    pos = torch.tensor([pos_i], device=DEV, dtype=torch.int32)
    slot = torch.tensor([slot_i], device=DEV, dtype=torch.int32)
    out = torch.empty(n_head * hs, device=DEV, dtype=torch.bfloat16)
    part_o = torch.empty(n_head * n_chunks * hs, device=DEV,
                         dtype=torch.float32)
    part_ml = torch.empty(n_head * n_chunks * 2, device=DEV,
                          dtype=torch.float32)
    scale = 1.0 / math.sqrt(hs)
    ops.attn_decode(out, part_o, part_ml, qkv, kpool, vpool, cos, sin, pos,
                    slot, layer, n_chunks, scale)

    # fp32 reference
    c, s = cos[pos_i:pos_i + 1], sin[pos_i:pos_i + 1]
    grp = qkv.view(n_kv, qpk + 2, hs)
    ref = torch.empty(n_head, hs, device=DEV)
    for g in range(n_kv):
        k_cur = _rope_ref(grp[g, qpk].view(1, -1), c, s)[0] if ne else \
            grp[g, qpk].float()
        K = torch.cat([kpool[slot_i, layer, g, :pos_i].float(),
                       k_cur.view(1, -1)])
        V = torch.cat([vpool[slot_i, layer, g, :pos_i].float(),
                       grp[g, qpk + 1].float().view(1, -1)])
        for j in range(qpk):
            q = _rope_ref(grp[g, j].view(1, -1), c, s)[0] if ne else \
                grp[g, j].float()
            att = (K @ q) * scale
            p = torch.softmax(att, dim=0)
            ref[g * qpk + j] = p @ V
        # append happened
        assert torch.allclose(kpool[slot_i, layer, g, pos_i].float(), k_cur,
                              atol=2e-2, rtol=2e-2)
        assert torch.equal(vpool[slot_i, layer, g, pos_i], grp[g, qpk + 1])
    got = out.view(n_head, hs).float()
    assert torch.allclose(got, ref, atol=3e-2, rtol=3e-2), \
        (got - ref).abs().max()


@pytest.mark.parametrize("gelu", [False, True])
def test_swiglu_mul(ops, gelu):
    """Batched act(gate)*up elementwise (grouped-engine glue) vs torch."""
    torch.manual_seed(33)
    g = torch.randn(64, 224, device=DEV).to(torch.bfloat16)
    u = torch.randn(64, 224, device=DEV).to(torch.bfloat16)
    out = torch.empty_like(u)
    ops.swiglu_mul(out, g, u, gelu)
    act = (torch.nn.functional.gelu(g.float(), approximate="tanh") if gelu
           else torch.nn.functional.silu(g.float()))
    ref = act * u.float()
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2)
    # aliasing: out may be u
    u2 = u.clone()
    ops.swiglu_mul(u2, g, u2, gelu)
    assert torch.equal(u2, out)
