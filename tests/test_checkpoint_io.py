import torch

from helpers import make_toy_checkpoint, make_toy_tokenizer

from mdi_llm_amd import GPT, ModelConfig
from mdi_llm_amd.utils.checkpoint import (
    load_from_pt,
    load_state_dict_lazy,
    save_checkpoint,
)


def test_save_load_roundtrip(tmp_path):
    ckpt = make_toy_checkpoint(tmp_path / "ck")
    config, sd = load_from_pt(ckpt)
    assert config.name == "nano-test"
    m = GPT(config)
    m.load_state_dict(sd)


def test_lazy_load_is_mmap(tmp_path):
    ckpt = make_toy_checkpoint(tmp_path / "ck")
    sd = load_state_dict_lazy(ckpt / "lit_model.pth")
    t = sd["transformer.wte.weight"]
    assert t.shape[1] == 64
    _ = t.sum()  # materializes fine


def test_tokenizer_roundtrip(tmp_path):
    from mdi_llm_amd.tokenizer import Tokenizer

    make_toy_tokenizer(tmp_path)
    tok = Tokenizer(tmp_path)
    ids = tok.encode("hello world")
    assert ids.numel() > 0
    text = tok.decode(ids)
    assert "hello" in text and "world" in text
    assert tok.eos_id is not None


def test_tokenizer_bos_eos(tmp_path):
    from mdi_llm_amd.tokenizer import Tokenizer

    make_toy_tokenizer(tmp_path)
    tok = Tokenizer(tmp_path)
    ids = tok.encode("hi", bos=True, eos=True)
    assert ids[0] == tok.bos_id
    assert ids[-1] == tok.eos_id


def test_convert_llama_roundtrip(tmp_path):
    """lit -> HF -> lit must reproduce the weights exactly."""
    from mdi_llm_amd.utils.convert_hf import (
        convert_hf_checkpoint,
        convert_lit_checkpoint,
    )

    ckpt = make_toy_checkpoint(tmp_path / "nano-test")
    config, sd0 = load_from_pt(ckpt)
    hf_path = tmp_path / "hf" / "pytorch_model.bin"
    convert_lit_checkpoint(ckpt, hf_path, model_name="nano-test")

    out = tmp_path / "back"
    convert_hf_checkpoint(tmp_path / "hf", out, model_name="nano-test")
    _, sd1 = load_from_pt(out, config)
    for k in sd0:
        assert k in sd1, k
        # wte/lm_head were sliced to vocab_size then re-padded with zeros
        assert torch.equal(sd0[k][: sd1[k].shape[0]], sd1[k]), k


def test_convert_gpt2_synthetic(tmp_path):
    """A synthetic HF-gpt2-layout dict converts and loads into our model."""
    from mdi_llm_amd.utils.convert_hf import convert_hf_checkpoint

    torch.manual_seed(0)
    cfg = ModelConfig.from_name("nano-test-gpt2")
    E, L = cfg.n_embd, cfg.n_layer
    hf = {}
    hf["wte.weight"] = torch.randn(cfg.vocab_size, E)
    hf["wpe.weight"] = torch.randn(cfg.block_size, E)
    for l in range(L):
        hf[f"h.{l}.ln_1.weight"] = torch.randn(E)
        hf[f"h.{l}.ln_1.bias"] = torch.randn(E)
        hf[f"h.{l}.attn.c_attn.weight"] = torch.randn(E, 3 * E)  # Conv1D
        hf[f"h.{l}.attn.c_attn.bias"] = torch.randn(3 * E)
        hf[f"h.{l}.attn.c_proj.weight"] = torch.randn(E, E)
        hf[f"h.{l}.attn.c_proj.bias"] = torch.randn(E)
        hf[f"h.{l}.ln_2.weight"] = torch.randn(E)
        hf[f"h.{l}.ln_2.bias"] = torch.randn(E)
        hf[f"h.{l}.mlp.c_fc.weight"] = torch.randn(E, 4 * E)
        hf[f"h.{l}.mlp.c_fc.bias"] = torch.randn(4 * E)
        hf[f"h.{l}.mlp.c_proj.weight"] = torch.randn(4 * E, E)
        hf[f"h.{l}.mlp.c_proj.bias"] = torch.randn(E)
    hf["ln_f.weight"] = torch.randn(E)
    hf["ln_f.bias"] = torch.randn(E)
    src = tmp_path / "hf2"
    src.mkdir()
    torch.save(hf, src / "pytorch_model.bin")

    out = tmp_path / "lit2"
    convert_hf_checkpoint(src, out, model_name="nano-test-gpt2")
    config, sd = load_from_pt(out)
    m = GPT(config)
    m.load_state_dict(sd)
    m.eval()
    with torch.inference_mode():
        logits = m(torch.randint(0, 255, (1, 8)))
    assert torch.isfinite(logits).all()
    # qkv weave: q rows of head 0 must equal the first hs rows of c_attn^T
    qkv = sd["transformer.h.0.attn.attn.weight"]
    cattn = hf["h.0.attn.c_attn.weight"].t()
    hs, qpk = config.head_size, config.q_per_kv
    assert torch.equal(qkv[:hs], cattn[:hs])  # q head 0
    assert torch.equal(qkv[qpk * hs: qpk * hs + hs], cattn[E: E + hs])  # k0


def test_convert_phi_synthetic(tmp_path):
    """Synthetic HF phi-layout dict converts, loads, and runs."""
    from mdi_llm_amd.utils.convert_hf import convert_hf_checkpoint

    torch.manual_seed(3)
    # a tiny phi-style config: LayerNorm + shared attention norm handled by
    # the model; use nano-test-neox-like geometry under a phi name
    from mdi_llm_amd.config import ModelConfig, name_to_config

    name_to_config.setdefault(
        "phi-nano",
        dict(
            name="phi-nano", block_size=128, vocab_size=256,
            padding_multiple=64, n_layer=2, n_head=4, n_embd=64,
            rotary_percentage=0.5, parallel_residual=True,
            shared_attention_norm=True, bias=True, lm_head_bias=True,
            norm_class_name="LayerNorm", mlp_class_name="GptNeoxMLP",
            gelu_approximate="tanh",
        ),
    )
    cfg = ModelConfig.from_name("phi-nano")
    E = cfg.n_embd
    hf = {"model.embed_tokens.weight": torch.randn(cfg.vocab_size, E),
          "model.final_layernorm.weight": torch.randn(E),
          "model.final_layernorm.bias": torch.randn(E),
          "lm_head.weight": torch.randn(cfg.vocab_size, E),
          "lm_head.bias": torch.randn(cfg.vocab_size)}
    for l in range(cfg.n_layer):
        p = f"model.layers.{l}"
        hf[f"{p}.input_layernorm.weight"] = torch.randn(E)
        hf[f"{p}.input_layernorm.bias"] = torch.randn(E)
        for w in ("q_proj", "k_proj", "v_proj"):
            hf[f"{p}.self_attn.{w}.weight"] = torch.randn(E, E)
            hf[f"{p}.self_attn.{w}.bias"] = torch.randn(E)
        hf[f"{p}.self_attn.dense.weight"] = torch.randn(E, E)
        hf[f"{p}.self_attn.dense.bias"] = torch.randn(E)
        hf[f"{p}.mlp.fc1.weight"] = torch.randn(4 * E, E)
        hf[f"{p}.mlp.fc1.bias"] = torch.randn(4 * E)
        hf[f"{p}.mlp.fc2.weight"] = torch.randn(E, 4 * E)
        hf[f"{p}.mlp.fc2.bias"] = torch.randn(E)
    src = tmp_path / "phi"
    src.mkdir()
    torch.save(hf, src / "pytorch_model.bin")
    out = tmp_path / "phi_lit"
    convert_hf_checkpoint(src, out, model_name="phi-nano")
    config, sd = load_from_pt(out)
    m = GPT(config)
    m.load_state_dict(sd)
    m.eval()
    with torch.inference_mode():
        logits = m(torch.randint(0, 255, (1, 8)))
    assert torch.isfinite(logits).all()
    assert "transformer.h.0.attn.attn.bias" in sd


def test_convert_falcon_synthetic(tmp_path):
    """Synthetic HF falcon-layout dict (7b-style: MQA, shared norm)
    converts with NO qkv weave (layouts identical) and runs; the lit->HF
    reverse map reproduces the original keys."""
    from mdi_llm_amd.utils.convert_hf import (
        convert_hf_checkpoint,
        convert_lit_checkpoint,
    )

    torch.manual_seed(6)
    cfg = ModelConfig.from_name("nano-test-falcon")
    E, L = cfg.n_embd, cfg.n_layer
    hf = {"transformer.word_embeddings.weight":
          torch.randn(cfg.vocab_size, E),
          "transformer.ln_f.weight": torch.randn(E),
          "transformer.ln_f.bias": torch.randn(E),
          "lm_head.weight": torch.randn(cfg.vocab_size, E)}
    for l in range(L):
        p = f"transformer.h.{l}"
        hf[f"{p}.input_layernorm.weight"] = torch.randn(E)
        hf[f"{p}.input_layernorm.bias"] = torch.randn(E)
        hf[f"{p}.self_attention.query_key_value.weight"] = \
            torch.randn(cfg.qkv_dim, E)
        hf[f"{p}.self_attention.dense.weight"] = torch.randn(E, E)
        hf[f"{p}.mlp.dense_h_to_4h.weight"] = torch.randn(4 * E, E)
        hf[f"{p}.mlp.dense_4h_to_h.weight"] = torch.randn(E, 4 * E)
    src = tmp_path / "falcon"
    src.mkdir()
    torch.save(hf, src / "pytorch_model.bin")

    out = tmp_path / "falcon_lit"
    convert_hf_checkpoint(src, out, model_name="nano-test-falcon")
    config, sd = load_from_pt(out)
    # qkv copied straight through, no weave
    assert torch.equal(
        sd["transformer.h.0.attn.attn.weight"],
        hf["transformer.h.0.self_attention.query_key_value.weight"])
    m = GPT(config)
    m.load_state_dict(sd)
    m.eval()
    with torch.inference_mode():
        logits = m(torch.randint(0, 255, (1, 8)))
    assert torch.isfinite(logits).all()

    back = tmp_path / "back" / "pytorch_model.bin"
    convert_lit_checkpoint(out, back, model_name="nano-test-falcon")
    hf2 = torch.load(back, weights_only=True)
    assert set(hf2) == set(hf)
    for k in hf:
        assert torch.equal(hf[k], hf2[k]), k


def test_falcon_registry_matches_reference():
    cfg = ModelConfig.from_name("falcon-7b")
    assert (cfg.n_layer, cfg.n_head, cfg.n_embd) == (32, 71, 4544)
    assert cfg.n_query_groups == 1 and cfg.shared_attention_norm
    assert cfg.head_size == 64 and cfg.qkv_dim == (71 + 2) * 64
    cfg40 = ModelConfig.from_name("falcon-40b-instruct")
    assert cfg40.n_query_groups == 8 and not cfg40.shared_attention_norm
    assert ModelConfig.from_name("falcon-180B-chat").n_layer == 80


def test_convert_mixtral_moe_synthetic(tmp_path):
    """Synthetic Mixtral-layout (block_sparse_moe) dict converts and loads
    into the local MoE model (reference convert_hf_checkpoint.py:139-142)."""
    from mdi_llm_amd.utils.convert_hf import convert_hf_checkpoint

    torch.manual_seed(9)
    cfg = ModelConfig.from_name("nano-test-moe")
    E, I, L = cfg.n_embd, cfg.intermediate_size, cfg.n_layer
    hf = {"model.embed_tokens.weight": torch.randn(cfg.vocab_size, E),
          "model.norm.weight": torch.randn(E),
          "lm_head.weight": torch.randn(cfg.vocab_size, E)}
    for l in range(L):
        p = f"model.layers.{l}"
        hf[f"{p}.input_layernorm.weight"] = torch.randn(E)
        hf[f"{p}.post_attention_layernorm.weight"] = torch.randn(E)
        for w in ("q_proj", "k_proj", "v_proj"):
            hf[f"{p}.self_attn.{w}.weight"] = torch.randn(E, E)
        hf[f"{p}.self_attn.o_proj.weight"] = torch.randn(E, E)
        hf[f"{p}.block_sparse_moe.gate.weight"] = \
            torch.randn(cfg.n_expert, E)
        for e in range(cfg.n_expert):
            q = f"{p}.block_sparse_moe.experts.{e}"
            hf[f"{q}.w1.weight"] = torch.randn(I, E)
            hf[f"{q}.w3.weight"] = torch.randn(I, E)
            hf[f"{q}.w2.weight"] = torch.randn(E, I)
    src = tmp_path / "mixtral"
    src.mkdir()
    torch.save(hf, src / "pytorch_model.bin")
    out = tmp_path / "mixtral_lit"
    convert_hf_checkpoint(src, out, model_name="nano-test-moe")
    config, sd = load_from_pt(out)
    assert torch.equal(sd["transformer.h.0.mlp.experts.2.fc_1.weight"],
                       hf["model.layers.0.block_sparse_moe.experts.2.w1.weight"])
    m = GPT(config)
    m.load_state_dict(sd)
    m.eval()
    with torch.inference_mode():
        logits = m(torch.randint(0, 255, (1, 8)))
    assert torch.isfinite(logits).all()


def test_convert_falcon_split_norm_synthetic(tmp_path):
    """40b/180B-style falcon (separate ln_attn/ln_mlp, GQA) conversion
    and reverse map (reference convert_hf_checkpoint.py:86-94)."""
    from mdi_llm_amd.config import name_to_config
    from mdi_llm_amd.utils.convert_hf import (
        convert_hf_checkpoint,
        convert_lit_checkpoint,
    )

    name_to_config.setdefault(
        "falcon-nano40",
        dict(
            name="falcon-nano40", block_size=128, vocab_size=256,
            padding_multiple=64, n_layer=2, n_head=8, n_embd=64,
            rotary_percentage=1.0, n_query_groups=2, bias=False,
            parallel_residual=True, norm_class_name="LayerNorm",
            mlp_class_name="GptNeoxMLP",
        ),
    )
    torch.manual_seed(12)
    cfg = ModelConfig.from_name("falcon-nano40")
    E, L = cfg.n_embd, cfg.n_layer
    hf = {"transformer.word_embeddings.weight":
          torch.randn(cfg.vocab_size, E),
          "transformer.ln_f.weight": torch.randn(E),
          "transformer.ln_f.bias": torch.randn(E),
          "lm_head.weight": torch.randn(cfg.vocab_size, E)}
    for l in range(L):
        p = f"transformer.h.{l}"
        hf[f"{p}.ln_attn.weight"] = torch.randn(E)
        hf[f"{p}.ln_attn.bias"] = torch.randn(E)
        hf[f"{p}.ln_mlp.weight"] = torch.randn(E)
        hf[f"{p}.ln_mlp.bias"] = torch.randn(E)
        hf[f"{p}.self_attention.query_key_value.weight"] = \
            torch.randn(cfg.qkv_dim, E)
        hf[f"{p}.self_attention.dense.weight"] = torch.randn(E, E)
        hf[f"{p}.mlp.dense_h_to_4h.weight"] = torch.randn(4 * E, E)
        hf[f"{p}.mlp.dense_4h_to_h.weight"] = torch.randn(E, 4 * E)
    src = tmp_path / "f40"
    src.mkdir()
    torch.save(hf, src / "pytorch_model.bin")
    out = tmp_path / "f40_lit"
    convert_hf_checkpoint(src, out, model_name="falcon-nano40")
    config, sd = load_from_pt(out)
    assert "transformer.h.0.norm_2.weight" in sd  # split norms mapped
    m = GPT(config)
    m.load_state_dict(sd)
    m.eval()
    with torch.inference_mode():
        logits = m(torch.randint(0, 255, (1, 6)))
    assert torch.isfinite(logits).all()
    back = tmp_path / "f40_back" / "pytorch_model.bin"
    convert_lit_checkpoint(out, back, model_name="falcon-nano40")
    hf2 = torch.load(back, weights_only=True)
    assert set(hf2) == set(hf)
    for k in hf:
        assert torch.equal(hf[k], hf2[k]), k
