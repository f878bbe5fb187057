"""Converter round trips (HF -> lit -> HF) for the neox/gpt2/phi families
(reference convert_lit_checkpoint.py:15-214) and the
bounded-RAM streaming conversion (reference streams via
litgpt_utils.py:304-343 — here via IncrementalSaver)."""

import json
import os
import subprocess
import sys

import pytest
import torch

from mdi_llm_amd.config import ModelConfig
from mdi_llm_amd.utils.convert_hf import (
    convert_hf_checkpoint,
    convert_lit_checkpoint,
    weave_qkv,
)


def _hf_neox_sd(cfg):
    torch.manual_seed(0)
    sd = {"gpt_neox.embed_in.weight": torch.randn(cfg.vocab_size,
                                                  cfg.n_embd),
          "gpt_neox.final_layer_norm.weight": torch.randn(cfg.n_embd),
          "gpt_neox.final_layer_norm.bias": torch.randn(cfg.n_embd),
          "embed_out.weight": torch.randn(cfg.vocab_size, cfg.n_embd)}
    for i in range(cfg.n_layer):
        p = f"gpt_neox.layers.{i}"
        sd.update({
            f"{p}.input_layernorm.weight": torch.randn(cfg.n_embd),
            f"{p}.input_layernorm.bias": torch.randn(cfg.n_embd),
            f"{p}.attention.query_key_value.weight":
                torch.randn(3 * cfg.n_embd, cfg.n_embd),
            f"{p}.attention.query_key_value.bias":
                torch.randn(3 * cfg.n_embd),
            f"{p}.attention.dense.weight":
                torch.randn(cfg.n_embd, cfg.n_embd),
            f"{p}.attention.dense.bias": torch.randn(cfg.n_embd),
            f"{p}.post_attention_layernorm.weight": torch.randn(cfg.n_embd),
            f"{p}.post_attention_layernorm.bias": torch.randn(cfg.n_embd),
            f"{p}.mlp.dense_h_to_4h.weight":
                torch.randn(cfg.intermediate_size, cfg.n_embd),
            f"{p}.mlp.dense_h_to_4h.bias":
                torch.randn(cfg.intermediate_size),
            f"{p}.mlp.dense_4h_to_h.weight":
                torch.randn(cfg.n_embd, cfg.intermediate_size),
            f"{p}.mlp.dense_4h_to_h.bias": torch.randn(cfg.n_embd),
        })
    return sd


def _hf_gpt2_sd(cfg):
    torch.manual_seed(1)
    sd = {"transformer.wte.weight": torch.randn(cfg.vocab_size, cfg.n_embd),
          "transformer.wpe.weight": torch.randn(cfg.block_size, cfg.n_embd),
          "transformer.ln_f.weight": torch.randn(cfg.n_embd),
          "transformer.ln_f.bias": torch.randn(cfg.n_embd)}
    for i in range(cfg.n_layer):
        p = f"transformer.h.{i}"
        sd.update({
            f"{p}.ln_1.weight": torch.randn(cfg.n_embd),
            f"{p}.ln_1.bias": torch.randn(cfg.n_embd),
            f"{p}.attn.c_attn.weight":
                torch.randn(cfg.n_embd, 3 * cfg.n_embd),  # Conv1D
            f"{p}.attn.c_attn.bias": torch.randn(3 * cfg.n_embd),
            f"{p}.attn.c_proj.weight": torch.randn(cfg.n_embd, cfg.n_embd),
            f"{p}.attn.c_proj.bias": torch.randn(cfg.n_embd),
            f"{p}.ln_2.weight": torch.randn(cfg.n_embd),
            f"{p}.ln_2.bias": torch.randn(cfg.n_embd),
            f"{p}.mlp.c_fc.weight":
                torch.randn(cfg.n_embd, cfg.intermediate_size),
            f"{p}.mlp.c_fc.bias": torch.randn(cfg.intermediate_size),
            f"{p}.mlp.c_proj.weight":
                torch.randn(cfg.intermediate_size, cfg.n_embd),
            f"{p}.mlp.c_proj.bias": torch.randn(cfg.n_embd),
        })
    return sd


def _hf_phi_sd(cfg):
    torch.manual_seed(2)
    hs = cfg.head_size
    sd = {"model.embed_tokens.weight": torch.randn(cfg.vocab_size,
                                                   cfg.n_embd),
          "model.final_layernorm.weight": torch.randn(cfg.n_embd),
          "model.final_layernorm.bias": torch.randn(cfg.n_embd),
          "lm_head.weight": torch.randn(cfg.vocab_size, cfg.n_embd),
          "lm_head.bias": torch.randn(cfg.vocab_size)}
    for i in range(cfg.n_layer):
        p = f"model.layers.{i}"
        sd.update({
            f"{p}.input_layernorm.weight": torch.randn(cfg.n_embd),
            f"{p}.input_layernorm.bias": torch.randn(cfg.n_embd),
            f"{p}.self_attn.q_proj.weight":
                torch.randn(cfg.n_head * hs, cfg.n_embd),
            f"{p}.self_attn.q_proj.bias": torch.randn(cfg.n_head * hs),
            f"{p}.self_attn.k_proj.weight":
                torch.randn(cfg.n_query_groups * hs, cfg.n_embd),
            f"{p}.self_attn.k_proj.bias":
                torch.randn(cfg.n_query_groups * hs),
            f"{p}.self_attn.v_proj.weight":
                torch.randn(cfg.n_query_groups * hs, cfg.n_embd),
            f"{p}.self_attn.v_proj.bias":
                torch.randn(cfg.n_query_groups * hs),
            f"{p}.self_attn.dense.weight":
                torch.randn(cfg.n_embd, cfg.n_head * hs),
            f"{p}.self_attn.dense.bias": torch.randn(cfg.n_embd),
            f"{p}.mlp.fc1.weight":
                torch.randn(cfg.intermediate_size, cfg.n_embd),
            f"{p}.mlp.fc1.bias": torch.randn(cfg.intermediate_size),
            f"{p}.mlp.fc2.weight":
                torch.randn(cfg.n_embd, cfg.intermediate_size),
            f"{p}.mlp.fc2.bias": torch.randn(cfg.n_embd),
        })
    return sd


@pytest.mark.parametrize("family,conf_name,builder", [
    ("neox", "nano-test-neox", _hf_neox_sd),
    ("gpt2", "nano-test-gpt2", _hf_gpt2_sd),
    ("phi", "nano-phi-test", _hf_phi_sd),
])
def test_hf_lit_hf_roundtrip(family, conf_name, builder, tmp_path):
    cfg = ModelConfig.from_name(conf_name)
    hf_sd = builder(cfg)
    hf_dir = tmp_path / "hf"
    hf_dir.mkdir()
    torch.save(hf_sd, hf_dir / "pytorch_model.bin")

    out = tmp_path / "lit"
    convert_hf_checkpoint(hf_dir, out, model_name=conf_name)
    lit_sd = torch.load(out / "lit_model.pth", weights_only=True)

    back_path = tmp_path / "back.bin"
    convert_lit_checkpoint(out, back_path, model_name=conf_name)
    back = torch.load(back_path, weights_only=True)

    skip_tied = {"lm_head.weight"} if family == "gpt2" else set()
    for k, v in hf_sd.items():
        if k in skip_tied:
            continue
        assert k in back, k
        assert torch.equal(back[k], v), k
    # nothing invented on the way back
    extra = set(back) - set(hf_sd)
    assert not extra, extra
    # the lit side really used the grouped-interleaved fused layout
    n_fused = sum(1 for k in lit_sd if k.endswith(".attn.attn.weight"))
    assert n_fused == cfg.n_layer


def test_streaming_convert_bounded_rss(tmp_path):
    """Converting a ~380 MB sharded checkpoint must not hold the model in
    RAM: the child's RSS growth during conversion stays a small fraction
    of the model size (streamed shard reads + incremental zip writes)."""
    gen = f"""
import torch, json, os
d = {str(repr(str(tmp_path)))}
hf = os.path.join(d, "hf"); os.makedirs(hf, exist_ok=True)
torch.manual_seed(0)
E, I, V = 1024, 2752, 8192
weight_map = {{}}
for i in range(6):
    p = f"model.layers.{{i}}"
    sd = {{
        f"{{p}}.input_layernorm.weight": torch.randn(E),
        f"{{p}}.self_attn.q_proj.weight": torch.randn(E, E),
        f"{{p}}.self_attn.k_proj.weight": torch.randn(E, E),
        f"{{p}}.self_attn.v_proj.weight": torch.randn(E, E),
        f"{{p}}.self_attn.o_proj.weight": torch.randn(E, E),
        f"{{p}}.post_attention_layernorm.weight": torch.randn(E),
        f"{{p}}.mlp.gate_proj.weight": torch.randn(I, E),
        f"{{p}}.mlp.up_proj.weight": torch.randn(I, E),
        f"{{p}}.mlp.down_proj.weight": torch.randn(E, I),
    }}
    fn = f"pytorch_model-{{i}}.bin"
    torch.save(sd, os.path.join(hf, fn))
    for k in sd: weight_map[k] = fn
sd = {{"model.embed_tokens.weight": torch.randn(V, E),
      "model.norm.weight": torch.randn(E),
      "lm_head.weight": torch.randn(V, E)}}
torch.save(sd, os.path.join(hf, "pytorch_model-t.bin"))
for k in sd: weight_map[k] = "pytorch_model-t.bin"
json.dump({{"weight_map": weight_map}},
          open(os.path.join(hf, "pytorch_model.bin.index.json"), "w"))
"""
    subprocess.run([sys.executable, "-c", gen], check=True)

    conv = f"""
import resource, torch, sys, os
sys.path.insert(0, {str(repr(os.getcwd()))})
from mdi_llm_amd.utils.convert_hf import convert_hf_checkpoint
base = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
d = {str(repr(str(tmp_path)))}
convert_hf_checkpoint(os.path.join(d, "hf"), os.path.join(d, "lit"),
                      model_name="nano-test",
                      config_overrides=dict(
                          n_layer=6, n_embd=1024, n_head=16,
                          n_query_groups=16, intermediate_size=2752,
                          vocab_size=8192, padded_vocab_size=8192))
peak = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
print("RSS_DELTA_KB", peak - base)
"""
    r = subprocess.run([sys.executable, "-c", conv], check=True,
                       capture_output=True, text=True)
    delta_kb = int([ln for ln in r.stdout.splitlines()
                    if ln.startswith("RSS_DELTA_KB")][0].split()[1])
    model_bytes = sum(
        os.path.getsize(tmp_path / "hf" / f)
        for f in os.listdir(tmp_path / "hf") if f.endswith(".bin"))
    # streamed: well under half the model (largest tensors are ~32 MB)
    assert delta_kb * 1024 < model_bytes * 0.5, (delta_kb, model_bytes)

    # and the product is a loadable lit checkpoint
    sd = torch.load(tmp_path / "lit" / "lit_model.pth", weights_only=True,
                    mmap=True)
    assert sd["transformer.wte.weight"].shape == (8192, 1024)
    assert "transformer.h.5.attn.attn.weight" in sd


def test_moe_lit_hf_roundtrip(tmp_path):
    """Mixtral-layout (block_sparse_moe experts) HF -> lit -> HF."""
    cfg = ModelConfig.from_name("nano-test-moe")
    torch.manual_seed(3)
    E, I = cfg.n_embd, cfg.intermediate_size
    hs = cfg.head_size
    sd = {"model.embed_tokens.weight": torch.randn(cfg.vocab_size, E),
          "model.norm.weight": torch.randn(E),
          "lm_head.weight": torch.randn(cfg.vocab_size, E)}
    for i in range(cfg.n_layer):
        p = f"model.layers.{i}"
        sd.update({
            f"{p}.input_layernorm.weight": torch.randn(E),
            f"{p}.self_attn.q_proj.weight":
                torch.randn(cfg.n_head * hs, E),
            f"{p}.self_attn.k_proj.weight":
                torch.randn(cfg.n_query_groups * hs, E),
            f"{p}.self_attn.v_proj.weight":
                torch.randn(cfg.n_query_groups * hs, E),
            f"{p}.self_attn.o_proj.weight": torch.randn(E, cfg.n_head * hs),
            f"{p}.post_attention_layernorm.weight": torch.randn(E),
            f"{p}.block_sparse_moe.gate.weight":
                torch.randn(cfg.n_expert, E),
        })
        for e in range(cfg.n_expert):
            q = f"{p}.block_sparse_moe.experts.{e}"
            sd.update({
                f"{q}.w1.weight": torch.randn(I, E),
                f"{q}.w3.weight": torch.randn(I, E),
                f"{q}.w2.weight": torch.randn(E, I),
            })
    hf_dir = tmp_path / "hf"
    hf_dir.mkdir()
    torch.save(sd, hf_dir / "pytorch_model.bin")
    out = tmp_path / "lit"
    convert_hf_checkpoint(hf_dir, out, model_name="nano-test-moe")
    lit_sd = torch.load(out / "lit_model.pth", weights_only=True)
    assert "transformer.h.0.mlp.experts.0.fc_1.weight" in lit_sd

    back_path = tmp_path / "back.bin"
    convert_lit_checkpoint(out, back_path, model_name="nano-test-moe")
    back = torch.load(back_path, weights_only=True)
    for k, v in sd.items():
        assert k in back, k
        assert torch.equal(back[k], v), k
    assert not (set(back) - set(sd))
