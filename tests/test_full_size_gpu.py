"""Full-size numerics on the GPU (VERDICT round-1 item 7): random-init
Llama-3-8B, HIP engine vs the torch stage — per-step logits within bf16
tolerance and 32-token greedy agreement.  Runs on one MI355X (two 16 GB
bf16 copies fit easily in 288 GB)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@torch.inference_mode()
def test_llama3_8b_engine_matches_torch():
    from mdi_llm_amd import GPT, ModelConfig
    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.ops.engine import DecodeEngine

    cfg = ModelConfig.from_name("Meta-Llama-3-8B-Instruct")
    torch.manual_seed(7)
    torch.set_default_dtype(torch.bfloat16)
    try:
        with torch.device(DEV):
            m = GPT(cfg)
    finally:
        torch.set_default_dtype(torch.float32)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0.0, 0.02)
    m.eval()
    m.max_seq_length = 2048
    m.set_kv_cache(1)

    stage = StarterStage(cfg, cfg.n_layer).to(DEV, dtype=torch.bfloat16)
    stage.load_state_dict(m.state_dict())
    stage.max_seq_length = 2048
    stage.eval()
    stage.set_kv_cache(1)

    torch.manual_seed(8)
    prompt = torch.randint(0, cfg.vocab_size - 1, (64,), device=DEV)
    ref_logits = m(prompt.view(1, -1), input_pos=0, slot=0)

    eng = DecodeEngine(stage, stage.kv_pool, use_graphs=True)
    eng.capture_graphs()
    stage.kv_pool.reset()
    # prefill through the engine's own HIP prefill path
    x = eng.prefill_prompt(prompt, 0, 0)
    eng.set_slot_pos(0, prompt.numel())
    pre_logits = eng.tail(x[-1])
    diff0 = (pre_logits.float() - ref_logits[0, -1].float()).abs().max()
    assert diff0 < 1.0, float(diff0)

    tok = ref_logits[0, -1].float().argmax()
    pos = prompt.numel()
    agree = 0
    for i in range(32):
        ref = m(tok.view(1, 1), input_pos=pos, slot=0)[0, -1].float()
        x = eng.decode_step_head(tok.to(torch.int32), slot=0)
        got = eng.tail(x).float()
        diff = (got - ref).abs().max()
        assert diff < 1.0, (i, float(diff))
        r_top = int(ref.argmax())
        g_top = int(got.argmax())
        if g_top == r_top:
            agree += 1
        else:
            # tolerate only genuine bf16 near-ties
            top2 = torch.topk(ref, 2).values
            assert float(top2[0] - top2[1]) < 0.05, (i, r_top, g_top)
        tok = ref.argmax()
        pos += 1
    assert agree >= 30, agree
