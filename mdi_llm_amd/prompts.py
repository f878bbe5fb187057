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


def _ids(tokenizer: Tokenizer, *seqs) -> Tuple[List[int], ...]:
    """Build stop sequences, dropping any whose tokens don't exist in
    this tokenizer (the reference indexes blindly and would crash)."""
    out: list = []
    if tokenizer.eos_id is not None:
        out.append([tokenizer.eos_id])
    for seq in seqs:
        ids = []
        ok = True
        for t in seq:
            tid = t if isinstance(t, int) else tokenizer.token_to_id(t)
            if tid is None:
                ok = False
                break
            ids.append(tid)
        if ok and ids:
            out.append(ids)
    return tuple(out)


# ---- remaining reference families (ref prompts.py:59-300) --------------
class FLAN(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return (
            "Below is an instruction that describes a task. Write a response "
            "that appropriately completes the request.\n\n"
            f"### Instruction:\n{prompt}\n\n### Response:\n"
        )


class Longform(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return (
            "Below is an instruction that describes a task, paired with an "
            "input that provides further context. Write a response that "
            "appropriately completes the request.\n\n"
            f"### Instruction:\n{prompt}\n\n### Response:\n"
        )


class StableLMAlpha(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return (
            "<|SYSTEM|># StableLM Tuned (Alpha version)\n- StableLM is a "
            "helpful and harmless open-source AI language model developed by "
            "StabilityAI.\n- StableLM is excited to be able to help the "
            "user, but will refuse to do anything that could be considered "
            "harmful to the user.\n- StableLM is more than just an "
            "information source, StableLM is also able to write poetry, "
            "short stories, and make jokes.\n- StableLM will refuse to "
            f"participate in anything that could harm a human.<|USER|>"
            f"{prompt}<|ASSISTANT|>"
        )

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        return _ids(tokenizer, ["<|SYSTEM|>"], ["<|ASSISTANT|>"],
                    ["<|USER|>"])


class StableLMZephyr(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"<|user|>\n{prompt}<|endoftext|>\n<|assistant|>\n"


class TogetherComputerChat(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"<human>: {prompt}\n<bot>:"

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        return _ids(tokenizer, ["<", "human", ">:"], ["<", "bot", ">:"])


class TogetherComputerInstruct(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"Q: {prompt}\nA:"

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        # NeoX tokenizer ids 187/535/2756 are '\n', '\n\n', '\n\n\n'
        return _ids(tokenizer, ["Q", ":"], ["Question"], ["A", ":"],
                    ["Label", ":"], [187, 187], [535], [2756])


class Falcon(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"Do not prefix your replies with 'Bot: '\nUser: {prompt}\n"

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        # 193 is '\n' in the falcon tokenizer
        return _ids(tokenizer, ["User", ":"], [193, "User"])


class Vicuna(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return (
            "A chat between a curious user and an artificial intelligence "
            "assistant. The assistant gives helpful, detailed, and polite "
            f"answers to the user's questions. USER: {prompt} ASSISTANT:"
        )


class Llama2FunctionCalling(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        # single search_bing example function, as the published adapter
        # expects (curly braces doubled to stay format-safe)
        function_list = json.dumps({
            "function": "search_bing",
            "description": (
                "Search the web for content on Bing. This allows users to "
                "search online/the internet/the web for content."
            ),
            "arguments": [{
                "name": "query",
                "type": "string",
                "description": "The search query string",
            }],
        }).replace("{", "{{").replace("}", "}}")
        sys_prompt = (
            "You are a helpful, respectful and honest assistant. Always "
            "answer as helpfully aspossible. Your only response should be "
            "JSON formatted functions"
        )
        return (
            f"<FUNCTIONS>{function_list.strip()}</FUNCTIONS>\n\n"
            f"[INST]<<SYS>>\n{sys_prompt}\n<</SYS>>\n\n{prompt}[/INST]\n\n"
        )


class FreeWilly2(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return (
            "### System:\nThis is a system prompt, please behave and help "
            "the user.\n\n"
            f"### User:\n{prompt}\n\n### Assistant:\n"
        )


class Platypus(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"### Instruction:\n\n{prompt}\n\n### Response:\n"


class NousResearch(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"### Instruction:\n{prompt}\n\n### Response:\n"


class StableCode(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"###Instruction\n{prompt}###Response\n"


class CodeLlama(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        # no default system prompt (HF conversational-instructions doc)
        return f"<s>[INST] {prompt} [/INST]"


class Phi1(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"{prompt}\n\nAnswer:"

    def stop_tokens(self, tokenizer: Tokenizer) -> Tuple[List[int], ...]:
        # 198 is '\n' in the codegen tokenizer
        return _ids(tokenizer, ["Answer", ":"], [198, "Answer", ":"])


class H2Oai(PromptStyle):
    def apply(self, prompt: str, **kwargs) -> str:
        return f"<|prompt|>{prompt}</s><|answer|>"


prompt_styles: dict[str, Type[PromptStyle]] = {
    "default": Default,
    "noprompt": NoPrompt,
    "alpaca": Alpaca,
    "flan": FLAN,
    "longform": Longform,
    "stablelm-alpha": StableLMAlpha,
    "stablelm-zephyr": StableLMZephyr,
    "togethercomputer-chat": TogetherComputerChat,
    "togethercomputer-instruct": TogetherComputerInstruct,
    "falcon": Falcon,
    "vicuna": Vicuna,
    "llama2-function-calling": Llama2FunctionCalling,
    "llama2": Llama2,
    "llama3": Llama3,
    "freewilly2": FreeWilly2,
    "platypus": Platypus,
    "nous-research": NousResearch,
    "stablecode": StableCode,
    "codellama": CodeLlama,
    "phi-1": Phi1,
    "phi-2": Phi2,
    "tinyllama": TinyLlama,
    "chatml": ChatML,
    "gemma": Gemma,
    "h2oai": H2Oai,
    "mistral": MistralInstruct,
}


def model_name_to_prompt_style(model_name: str) -> PromptStyle:
    """Map a model/config name to its chat style
    (reference prompts.py:325-366, same precedence order)."""
    if re.search(r"stablelm-tuned-alpha", model_name):
        return StableLMAlpha()
    if re.search(r"stablelm-zephyr-3b", model_name):
        return StableLMZephyr()
    if re.search("stablecode-instruct", model_name):
        return StableCode()
    if re.search(r"RedPajama-INCITE.*-Chat", model_name):
        return TogetherComputerChat()
    if re.search(r"RedPajama-INCITE.*-Instruct", model_name):
        return TogetherComputerInstruct()
    if re.search(r"falcon.*-instruct", model_name):
        return Falcon()
    if re.search(r"vicuna|longchat", model_name):
        return Vicuna()
    if re.search("Llama-2-7b-chat-hf-function-calling-v2", model_name):
        return Llama2FunctionCalling()
    if re.search("Llama-2.*-chat", model_name):
        return Llama2()
    if re.search(r"Llama-3.*-Instruct", model_name, re.IGNORECASE):
        return Llama3()
    if re.search("FreeWilly2", model_name):
        return FreeWilly2()
    if re.search("Platypus", model_name):
        return Platypus()
    if re.search("Nous-Hermes", model_name):
        return NousResearch()
    if re.search("CodeLlama", model_name):
        return CodeLlama()
    if re.search("Mistral.*Instruct", model_name):
        return MistralInstruct()
    if re.search("phi-1", model_name):
        return Phi1()
    if re.search("phi-2", model_name):
        return Phi2()
    if re.search(r"TinyLlama.*Chat|tiny-llama.*chat", model_name):
        return TinyLlama()
    if re.search(r"(Code)?[Gg]emma.*-it", model_name):
        return Gemma()
    if re.search(r"Danube2.*-chat", model_name):
        return H2Oai()
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
