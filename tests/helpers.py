"""Shared test fixtures: build a complete toy litGPT checkpoint dir
(model_config.yaml + lit_model.pth + tokenizer.json) on the fly."""

from pathlib import Path

import torch


def make_toy_tokenizer(out_dir: Path, vocab_size: int = 256) -> Path:
    """Train a tiny byte-level BPE tokenizer and save tokenizer.json."""
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers
    from tokenizers.processors import TemplateProcessing

    tok = Tokenizer(models.BPE(unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    trainer = trainers.BpeTrainer(
        vocab_size=vocab_size,
        special_tokens=["<unk>", "<s>", "</s>"],
    )
    corpus = [
        "the quick brown fox jumps over the lazy dog",
        "hello world this is a tiny tokenizer for tests",
        "who are you? I am a language model running on MI355X",
        "0123456789 abcdefghijklmnopqrstuvwxyz",
    ] * 8
    tok.train_from_iterator(corpus, trainer)
    out_dir.mkdir(parents=True, exist_ok=True)
    tok.save(str(out_dir / "tokenizer.json"))
    (out_dir / "tokenizer_config.json").write_text(
        '{"bos_token": "<s>", "eos_token": "</s>", "add_bos_token": false}'
    )
    return out_dir / "tokenizer.json"


def make_toy_checkpoint(out_dir: Path, name: str = "nano-test",
                        seed: int = 0) -> Path:
    from mdi_llm_amd import GPT, ModelConfig
    from mdi_llm_amd.utils.checkpoint import save_checkpoint

    torch.manual_seed(seed)
    cfg = ModelConfig.from_name(name)
    m = GPT(cfg)
    m.apply_init()
    save_checkpoint(out_dir, cfg, m.state_dict())
    make_toy_tokenizer(out_dir)
    return out_dir
