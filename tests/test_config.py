import pytest
import torch

from mdi_llm_amd.config import ModelConfig, name_to_config


def test_registry_has_required_families():
    required = [
        "NanoLlama",
        "TinyLlama-1.1B-Chat-v1.0",
        "Llama-2-7b-hf",
        "Meta-Llama-3-8B-Instruct",
        "Meta-Llama-3-70B-Instruct",
        "gpt2",
        "gpt2-xl",
        "pythia-160m",
        "phi-2",
        "Mistral-7B-Instruct-v0.2",
        "gemma-2b",
        "Mixtral-8x7B-v0.1",
    ]
    for name in required:
        assert name in name_to_config, name
        cfg = ModelConfig.from_name(name)
        assert cfg.n_layer > 0


def test_llama3_8b_shapes():
    cfg = ModelConfig.from_name("Meta-Llama-3-8B-Instruct")
    assert cfg.n_layer == 32
    assert cfg.n_embd == 4096
    assert cfg.n_query_groups == 8
    assert cfg.head_size == 128
    assert cfg.padded_vocab_size == 128256
    assert cfg.intermediate_size == 14336
    assert cfg.qkv_dim == (32 + 16) * 128
    assert cfg.rope_n_elem == 128


def test_padded_vocab_rounding():
    cfg = ModelConfig(name="x", vocab_size=50254, padding_multiple=512)
    assert cfg.padded_vocab_size == 50688


def test_dict_roundtrip():
    cfg = ModelConfig.from_name("TinyLlama-1.1B-Chat-v1.0")
    d = cfg.to_dict()
    cfg2 = ModelConfig.from_dict(d)
    assert cfg2 == cfg


def test_yaml_roundtrip(tmp_path):
    cfg = ModelConfig.from_name("NanoLlama")
    cfg.save(tmp_path / "model_config.yaml")
    cfg2 = ModelConfig.from_checkpoint(tmp_path)
    assert cfg2 == cfg


def test_unknown_name_raises():
    with pytest.raises(ValueError):
        ModelConfig.from_name("no-such-model-xyz")


def test_overrides():
    cfg = ModelConfig.from_name("NanoLlama", block_size=256)
    assert cfg.block_size == 256
