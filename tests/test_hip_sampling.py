"""GPU tests for the fused radix-top-k + Gumbel sampling kernel."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.fixture(scope="module")
def ops():
    from mdi_llm_amd.ops import require_hip_ops

    return require_hip_ops()


def _call(ops, logits, temperature, top_k, noise=True, seed=7, ctr_val=0):
    # ctr_val is realised as the (pos, slot) RNG key of the draw
    scratch = torch.zeros(520, device=DEV, dtype=torch.int32)
    out = torch.zeros(1, device=DEV, dtype=torch.int32)
    pos = torch.tensor([ctr_val], device=DEV, dtype=torch.int32)
    slot = torch.zeros(1, device=DEV, dtype=torch.int32)
    ops.sample(out, logits, scratch, temperature, top_k, noise, seed,
               pos=pos, slot=slot)
    return int(out)


def test_greedy_matches_argmax(ops):
    torch.manual_seed(0)
    for V in (1000, 128256):
        logits = torch.randn(V, device=DEV).to(torch.bfloat16)
        got = _call(ops, logits, 0.0, 0, noise=False)
        assert got == int(logits.float().argmax())


def test_topk_support(ops):
    """Every sampled token must be inside the true top-k set."""
    torch.manual_seed(1)
    V, k = 128256, 200
    logits = torch.randn(V, device=DEV).to(torch.bfloat16)
    kth = torch.topk(logits.float(), k).values[-1]
    topset = set((logits.float() >= kth).nonzero().flatten().tolist())
    for ctr in range(50):
        tok = _call(ops, logits, 0.8, k, noise=True, ctr_val=ctr)
        assert tok in topset, (ctr, tok)


def test_deterministic_given_seed_pos(ops):
    torch.manual_seed(2)
    logits = torch.randn(5000, device=DEV).to(torch.bfloat16)
    a = _call(ops, logits, 1.0, 50, seed=11, ctr_val=3)
    b = _call(ops, logits, 1.0, 50, seed=11, ctr_val=3)
    c = _call(ops, logits, 1.0, 50, seed=11, ctr_val=4)
    assert a == b
    # different position should (almost surely) give a different draw
    # occasionally equal is fine; just check the call runs
    assert isinstance(c, int)


def test_distribution_roughly_matches(ops):
    """Gumbel-max over temperature-scaled logits == softmax sampling."""
    probs = torch.tensor([0.6, 0.25, 0.1, 0.03, 0.02])
    logits = torch.log(probs).to(DEV).to(torch.bfloat16)
    pad = torch.full((3,), -30.0, device=DEV, dtype=torch.bfloat16)
    logits = torch.cat([logits, pad])
    counts = torch.zeros(8)
    n = 4000
    for ctr in range(n):
        tok = _call(ops, logits, 1.0, 0, noise=True, seed=5, ctr_val=ctr)
        counts[tok] += 1
    freq = counts / n
    assert torch.allclose(freq[:5], probs, atol=0.04), freq


def _call_p(ops, logits, temperature, top_k, top_p, seed=7, ctr_val=0):
    scratch = torch.zeros(520, device=DEV, dtype=torch.int32)
    out = torch.zeros(1, device=DEV, dtype=torch.int32)
    pos = torch.tensor([ctr_val], device=DEV, dtype=torch.int32)
    ops.sample(out, logits, scratch, temperature, top_k, True, seed,
               pos=pos, top_p=top_p)
    return int(out)


def test_top_p_nucleus_support(ops):
    """Every draw must come from the nucleus (smallest top set with
    cumulative probability >= p, boundary token included)."""
    torch.manual_seed(4)
    V = 4096
    logits = (torch.randn(V, device=DEV) * 3).to(torch.bfloat16)
    T, P = 1.0, 0.6
    probs = torch.softmax(logits.float() / T, dim=0)
    sp, si = probs.sort(descending=True)
    cum = sp.cumsum(0)
    n_keep = int((cum < P).sum()) + 1  # boundary token included
    nucleus = set(si[:n_keep].tolist())
    # allow bf16-level ties at the boundary value
    bval = sp[n_keep - 1]
    nucleus |= set((probs >= bval * 0.999).nonzero().flatten().tolist())
    for ctr in range(60):
        tok = _call_p(ops, logits, T, 0, P, ctr_val=ctr)
        assert tok in nucleus, (ctr, tok, n_keep)


def test_top_p_and_top_k_combined(ops):
    torch.manual_seed(5)
    V = 8192
    logits = (torch.randn(V, device=DEV) * 2).to(torch.bfloat16)
    kth = torch.topk(logits.float(), 50).values[-1]
    topset = set((logits.float() >= kth).nonzero().flatten().tolist())
    for ctr in range(30):
        tok = _call_p(ops, logits, 0.8, 50, 0.9, ctr_val=ctr)
        assert tok in topset


def test_small_topk(ops):
    torch.manual_seed(3)
    logits = torch.randn(1000, device=DEV).to(torch.bfloat16)
    top2 = set(torch.topk(logits.float(), 2).indices.tolist())
    for ctr in range(20):
        tok = _call(ops, logits, 2.0, 2, noise=True, ctr_val=ctr)
        # allow bf16 ties at the threshold
        kth = torch.topk(logits.float(), 2).values[-1]
        assert logits.float()[tok] >= kth


def test_stage_slot_op(ops):
    """stage_slot: pos/token staging + pos advance in one launch."""
    pos_table = torch.tensor([5, 9, 2], device=DEV, dtype=torch.int32)
    token_table = torch.tensor([11, 22, 33], device=DEV, dtype=torch.int32)
    slot = torch.tensor([1], device=DEV, dtype=torch.int32)
    pos = torch.zeros(1, device=DEV, dtype=torch.int32)
    tok = torch.zeros(1, device=DEV, dtype=torch.int32)
    ops.stage_slot(slot, pos_out=pos, token_out=tok, pos_table=pos_table,
                   token_table=token_table)
    assert int(pos) == 9 and int(tok) == 22
    ops.stage_slot(slot, pos_table_mut=pos_table, adv_pos=1)
    assert pos_table.tolist() == [5, 10, 2]


def test_sample_unpack_bookkeeping(ops):
    """The sampler's unpack launch writes token_table[slot] and advances
    pos_table[slot] when asked (the in-graph step bookkeeping)."""
    torch.manual_seed(7)
    logits = torch.randn(1000, device=DEV).to(torch.bfloat16)
    scratch = torch.zeros(520, device=DEV, dtype=torch.int32)
    out = torch.zeros(1, device=DEV, dtype=torch.int32)
    pos = torch.tensor([4], device=DEV, dtype=torch.int32)
    slot = torch.tensor([2], device=DEV, dtype=torch.int32)
    pos_table = torch.tensor([0, 0, 4, 0], device=DEV, dtype=torch.int32)
    token_table = torch.zeros(4, device=DEV, dtype=torch.int32)
    ops.sample(out, logits, scratch, 0.0, 0, False, 1,
               pos=pos, slot=slot, token_table=token_table,
               pos_table=pos_table, adv_slot=slot, adv_pos=1)
    argmax = int(logits.float().argmax())
    assert int(out) == argmax
    assert token_table.tolist() == [0, 0, argmax, 0]
    assert pos_table.tolist() == [0, 0, 5, 0]
