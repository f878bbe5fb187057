"""Layer split + chunked checkpoint format compatibility."""

import pytest
import torch

from mdi_llm_amd import GPT, ModelConfig, StarterStage, SecondaryStage
from mdi_llm_amd.utils import (
    N_LAYERS_NODES,
    chunk_dir,
    layer_split,
    split_and_store,
    split_parameters,
)


def test_table_compat_with_reference():
    # values from the reference N_LAYERS_NODES (config.py:56-98)
    assert layer_split(32, 2) == [14, 18]
    assert layer_split(32, 3) == [8, 12, 12]
    assert layer_split(22, 3) == [6, 8, 8]
    assert layer_split(48, 3) == [14, 17, 17]
    assert layer_split(22, 5) == [2, 5, 5, 5, 5]
    assert layer_split(32, 5) == [4, 7, 7, 7, 7]


def test_general_formula():
    # 8-stage Llama-3-8B — not in the reference table
    s = layer_split(32, 8)
    assert sum(s) == 32 and len(s) == 8
    assert s[0] <= min(s[1:])
    s = layer_split(80, 8)  # Llama-3-70B
    assert sum(s) == 80 and len(s) == 8
    assert s[0] <= min(s[1:])
    s = layer_split(12, 4)
    assert sum(s) == 12 and len(s) == 4


def test_split_keys_match_reference_layout():
    cfg = ModelConfig.from_name("nano-test")  # 4 layers
    m = GPT(cfg)
    chunks = split_parameters(m.state_dict(), 2)
    # starter: wte + h.0..k-1 + ln_f + lm_head, zero-indexed
    assert "transformer.wte.weight" in chunks[0]
    assert "transformer.ln_f.weight" in chunks[0]
    assert "lm_head.weight" in chunks[0]
    starter_blocks = {
        int(k.split(".")[2]) for k in chunks[0] if k.startswith("transformer.h.")
    }
    sec_blocks = {
        int(k.split(".")[2]) for k in chunks[1] if k.startswith("transformer.h.")
    }
    split = layer_split(4, 2)
    assert starter_blocks == set(range(split[0]))
    # secondary chunk re-zero-indexed (reference utils.py:374-381)
    assert sec_blocks == set(range(split[1]))
    assert "transformer.wte.weight" not in chunks[1]


def test_chunks_load_into_stage_modules(tmp_path):
    torch.manual_seed(0)
    cfg = ModelConfig.from_name("nano-test")
    m = GPT(cfg)
    m.apply_init()
    out = split_and_store(m.state_dict(), 3, tmp_path)
    assert out == chunk_dir(tmp_path, 3)
    split = layer_split(cfg.n_layer, 3)
    st = StarterStage(cfg, split[0])
    st.load_state_dict(torch.load(out / "model_starter.pth", weights_only=True))
    for i in (1, 2):
        sec = SecondaryStage(cfg, split[i])
        sec.load_state_dict(
            torch.load(out / f"model_secondary{i-1}.pth", weights_only=True)
        )


@torch.inference_mode()
def test_staged_forward_matches_full_model():
    """Chain starter head -> secondaries -> starter tail == full model."""
    torch.manual_seed(0)
    cfg = ModelConfig.from_name("nano-test")
    m = GPT(cfg)
    m.apply_init()
    m.eval()
    n_nodes = 2
    split = layer_split(cfg.n_layer, n_nodes)
    chunks = split_parameters(m.state_dict(), n_nodes)
    starter = StarterStage(cfg, split[0])
    starter.load_state_dict(chunks[0])
    sec = SecondaryStage(cfg, split[1])
    sec.load_state_dict(chunks[1])
    starter.eval(), sec.eval()
    starter.set_kv_cache(1)
    sec.set_kv_cache(1)

    idx = torch.randint(0, 255, (1, 10))
    ref = m(idx)  # full-context logits

    # prefill through the pipeline
    x = starter.forward_head(idx, slot=0, input_pos=0)
    x = sec(x, slot=0, input_pos=0)
    logits = starter.forward_tail(x)
    assert torch.allclose(logits[0, -1], ref[0, -1], atol=1e-4)

    # one decode step
    nxt = ref[0, -1].argmax().view(1, 1)
    full_ref = m(torch.cat([idx, nxt], dim=1))
    x = starter.forward_head(nxt, slot=0, input_pos=10)
    x = sec(x, slot=0, input_pos=10)
    logits = starter.forward_tail(x)
    assert torch.allclose(logits[0, -1], full_ref[0, -1], atol=1e-4)


@pytest.mark.parametrize("model,n_nodes", [
    ("Meta-Llama-3-8B-Instruct", 2),
    ("Meta-Llama-3-8B-Instruct", 4),
    ("Meta-Llama-3-8B-Instruct", 8),
    ("Meta-Llama-3-70B-Instruct", 8),
    ("TinyLlama-1.1B-Chat-v1.0", 3),
])
def test_balanced_split_byte_balance(model, n_nodes):
    """Analytic validation of the byte-balance objective (round-1 weak
    #7): per-stage streamed bytes (blocks + the starter's lm_head) stay
    within ~1.3 per-layer quanta of each other — the best achievable at
    integer block granularity — so no stage is a >15% straggler on the
    bandwidth-bound decode."""
    from mdi_llm_amd.config import ModelConfig
    from mdi_llm_amd.utils.partition import balanced_split

    cfg = ModelConfig.from_name(model)
    counts = balanced_split(cfg, n_nodes)
    assert sum(counts) == cfg.n_layer and len(counts) == n_nodes

    E, I, V = cfg.n_embd, cfg.intermediate_size, cfg.padded_vocab_size
    hs, nh, ng = cfg.head_size, cfg.n_head, cfg.n_query_groups
    layer_b = (E * (nh + 2 * ng) * hs + E * nh * hs + 3 * E * I) * 2
    starter_extra = V * E * 2 + 150_000_000
    bytes_per_stage = [counts[0] * layer_b + starter_extra] + [
        c * layer_b for c in counts[1:]
    ]
    spread = (max(bytes_per_stage) - min(bytes_per_stage)) / layer_b
    assert spread <= 1.3, (counts, spread)
    # and the relative straggler penalty is small (integer block
    # granularity bounds it: e.g. 30 blocks over 7 secondaries forces
    # some 5-block stages against a 4.3 mean)
    rel = max(bytes_per_stage) / (sum(bytes_per_stage) / n_nodes)
    assert rel < 1.2, (counts, rel)
