"""Prompt styles: per-model chat templating and stop-token sequences.

Capability parity with the reference prompt machinery
(/root/reference/src/sub/prompts.py: ``PromptStyle`` 17, model styles
195-323, ``model_name_to_prompt_style`` 325, ``save/load/has_prompt_style``
369-389, ``get_user_prompt`` 392-445 incl. ``FILE:`` multi-prompt fan-out).
"""

from __future__ import annotations

import json
import re
from pathlib import Path
from typing import List, Optional, Tuple, Type, Union

import yaml

from .tokenizer import Tokenizer

__all__ = [
    "PromptStyle",
    "model_name_to_prompt_style",
    "save_prompt_style",
    "load_prompt_style",
    "has_prompt_style",
    "get_user_prompt",
]


class PromptStyle:
    """Base class: how a user prompt is wrapped for a given model family."""

    def apply(self, prompt: str, **kwargs) -> str:
        return prompt

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        return ([tokenizer.eos_id],) if tokenizer.eos_id is not None else ()

    @classmethod
    def from_name(cls, name: str) -> "PromptStyle":
        return prompt_styles[name]()

    @classmethod
    def from_config(cls, config) -> "PromptStyle":
        return model_name_to_prompt_style(getattr(config, "name", str(config)))


class Default(PromptStyle):
    pass


class NoPrompt(PromptStyle):
    """Raw pass-through (reference prompts.py:302)."""


class Alpaca(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return (
            "Below is an instruction that describes a task. Write a response "
            "that appropriately completes the request.\n\n"
            f"### Instruction:\n{prompt}\n\n### Response:\n"
        )


class Llama2(PromptStyle):
    def apply(self, prompt: str, *, sys_prompt: Optional[str] = None, **kw) -> str:
        sys_prompt = sys_prompt or (
            "You are a helpful assistant. Always answer as helpfully as "
            "possible and follow ALL given instructions."
        )
        return f"<s>[INST] <<SYS>>\n{sys_prompt}\n<</SYS>>\n\n{prompt} [/INST] "


class Llama3(PromptStyle):
    def apply(self, prompt: str, *, sys_prompt: Optional[str] = None, **kw) -> str:
        sys_prompt = sys_prompt or "You are a helpful assistant."
        return (
            "<|begin_of_text|><|start_header_id|>system<|end_header_id|>\n\n"
            f"{sys_prompt}<|eot_id|><|start_header_id|>user<|end_header_id|>\n\n"
            f"{prompt}<|eot_id|><|start_header_id|>assistant<|end_header_id|>\n\n"
        )

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        stops: list = []
        if tokenizer.eos_id is not None:
            stops.append([tokenizer.eos_id])
        eot = tokenizer.token_to_id("<|eot_id|>")
        if eot is not None:
            stops.append([eot])
        return tuple(stops)


class TinyLlama(PromptStyle):
    def apply(self, prompt: str, *, sys_prompt: Optional[str] = None, **kw) -> str:
        sys_prompt = sys_prompt or (
            "You are a friendly chatbot who always gives helpful, detailed, "
            "and polite answers."
        )
        return (
            f"<|system|>\n{sys_prompt}</s>\n<|user|>\n{prompt}</s>\n<|assistant|>\n"
        )


class ChatML(PromptStyle):
    def apply(self, prompt: str, *, sys_prompt: Optional[str] = None, **kw) -> str:
        sys_prompt = sys_prompt or "You are a helpful assistant."
        return (
            f"<|im_start|>system\n{sys_prompt}<|im_end|>\n"
            f"<|im_start|>user\n{prompt}<|im_end|>\n<|im_start|>assistant\n"
        )

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        stops: list = []
        if tokenizer.eos_id is not None:
            stops.append([tokenizer.eos_id])
        im_end = tokenizer.token_to_id("<|im_end|>")
        if im_end is not None:
            stops.append([im_end])
        return tuple(stops)


class Phi2(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"Instruct:{prompt}\nOutput:"


class Gemma(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"<start_of_turn>user\n{prompt}<end_of_turn>\n<start_of_turn>model\n"

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        stops: list = []
        if tokenizer.eos_id is not None:
            stops.append([tokenizer.eos_id])
        eot = tokenizer.token_to_id("<end_of_turn>")
        if eot is not None:
            stops.append([eot])
        return tuple(stops)


class MistralInstruct(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"<s>[INST] {prompt} [/INST]"


prompt_styles: dict[str, Type[PromptStyle]] = {
    "default": Default,
    "noprompt": NoPrompt,
    "alpaca": Alpaca,
    "llama2": Llama2,
    "llama3": Llama3,
    "tinyllama": TinyLlama,
    "chatml": ChatML,
    "phi-2": Phi2,
    "gemma": Gemma,
    "mistral": MistralInstruct,
}


def model_name_to_prompt_style(model_name: str) -> PromptStyle:
    """Map a model/config name to its chat style
    (reference prompts.py:325-366)."""
    if re.search(r"Llama-2.*-chat", model_name):
        return Llama2()
    if re.search(r"Llama-3.*-Instruct", model_name, re.IGNORECASE):
        return Llama3()
    if re.search(r"TinyLlama.*Chat", model_name):
        return TinyLlama()
    if re.search("phi-2", model_name):
        return Phi2()
    if re.search(r"gemma.*-it", model_name):
        return Gemma()
    if re.search(r"Mistral.*Instruct", model_name):
        return MistralInstruct()
    return Default()


def save_prompt_style(style: PromptStyle, checkpoint_dir: Union[str, Path]) -> None:
    cls = type(style)
    config = {"class_path": f"{cls.__module__}.{cls.__name__}"}
    with open(Path(checkpoint_dir) / "prompt_style.yaml", "w") as fp:
        yaml.safe_dump(config, fp)


def load_prompt_style(checkpoint_dir: Union[str, Path]) -> PromptStyle:
    with open(Path(checkpoint_dir) / "prompt_style.yaml") as fp:
        config = yaml.safe_load(fp)
    import importlib

    module_path, _, name = config["class_path"].rpartition(".")
    mod = importlib.import_module(module_path)
    return getattr(mod, name)()


def has_prompt_style(checkpoint_dir: Union[str, Path]) -> bool:
    return (Path(checkpoint_dir) / "prompt_style.yaml").is_file()


def get_user_prompt(
    prompt: str, n_samples: int, custom_system_prompt: Optional[str] = None
) -> List[str]:
    """Expand the CLI ``--prompt`` argument into one prompt per sample.

    ``FILE:<path>`` reads a text file whose paragraphs (blank-line
    separated) map to samples, cycling if fewer than ``n_samples``
    (reference prompts.py:392-445).
    """
    if prompt.startswith("FILE:"):
        path = Path(prompt[len("FILE:") :])
        text = path.read_text(encoding="utf-8")
        paragraphs = [p.strip() for p in re.split(r"\n\s*\n", text) if p.strip()]
        if not paragraphs:
            raise ValueError(f"no prompts found in {path}")
        return [paragraphs[i % len(paragraphs)] for i in range(n_samples)]
    return [prompt for _ in range(n_samples)]
