"""Registry parity: every named config of the reference registry
(/root/reference/src/sub/config.py:180-1667, 112 concrete names after
template expansion) resolves here with the same architecture.  The name
list is frozen below so the test runs without the reference checkout."""

import pytest

REFERENCE_NAMES = [
    "Camel-Platypus2-13B",
    "Camel-Platypus2-70B",
    "CodeGemma-7b-it",
    "CodeLlama-13b-Instruct-hf",
    "CodeLlama-13b-Python-hf",
    "CodeLlama-13b-hf",
    "CodeLlama-34b-Instruct-hf",
    "CodeLlama-34b-Python-hf",
    "CodeLlama-34b-hf",
    "CodeLlama-70b-Instruct-hf",
    "CodeLlama-70b-Python-hf",
    "CodeLlama-70b-hf",
    "CodeLlama-7b-Instruct-hf",
    "CodeLlama-7b-Python-hf",
    "CodeLlama-7b-hf",
    "Danube2-1.8b-chat",
    "FreeWilly2",
    "Gemma-2b",
    "Gemma-2b-it",
    "Gemma-7b",
    "Gemma-7b-it",
    "LLaMA-2-7B-32K",
    "Llama-2-13b-chat-hf",
    "Llama-2-13b-hf",
    "Llama-2-70b-chat-hf",
    "Llama-2-70b-hf",
    "Llama-2-7b-chat-hf",
    "Llama-2-7b-chat-hf-function-calling-v2",
    "Llama-2-7b-hf",
    "Llama-3-70B",
    "Llama-3-70B-Instruct",
    "Llama-3-8B",
    "Llama-3-8B-Instruct",
    "Mistral-7B-Instruct-v0.1",
    "Mistral-7B-Instruct-v0.2",
    "Mistral-7B-Instruct-v0.3",
    "Mistral-7B-v0.1",
    "Mistral-7B-v0.2",
    "Mistral-7B-v0.3",
    "Mixtral-8x7B-Instruct-v0.1",
    "Mixtral-8x7B-v0.1",
    "Nous-Hermes-13b",
    "Nous-Hermes-Llama2-13b",
    "Nous-Hermes-llama-2-7b",
    "Platypus-30B",
    "Platypus2-13B",
    "Platypus2-70B",
    "Platypus2-70B-instruct",
    "Platypus2-7B",
    "RedPajama-INCITE-7B-Base",
    "RedPajama-INCITE-7B-Chat",
    "RedPajama-INCITE-7B-Instruct",
    "RedPajama-INCITE-Base-3B-v1",
    "RedPajama-INCITE-Base-7B-v0.1",
    "RedPajama-INCITE-Chat-3B-v1",
    "RedPajama-INCITE-Chat-7B-v0.1",
    "RedPajama-INCITE-Instruct-3B-v1",
    "RedPajama-INCITE-Instruct-7B-v0.1",
    "Stable-Platypus2-13B",
    "dolly-v2-12b",
    "dolly-v2-3b",
    "dolly-v2-7b",
    "falcon-180B",
    "falcon-180B-chat",
    "falcon-40b",
    "falcon-40b-instruct",
    "falcon-7b",
    "falcon-7b-instruct",
    "longchat-13b-16k",
    "longchat-7b-16k",
    "open_llama_13b",
    "open_llama_3b",
    "open_llama_7b",
    "phi-1_5",
    "phi-2",
    "pythia-1.4b",
    "pythia-1.4b-deduped",
    "pythia-12b",
    "pythia-12b-deduped",
    "pythia-14m",
    "pythia-160m",
    "pythia-160m-deduped",
    "pythia-1b",
    "pythia-1b-deduped",
    "pythia-2.8b",
    "pythia-2.8b-deduped",
    "pythia-31m",
    "pythia-410m",
    "pythia-410m-deduped",
    "pythia-6.9b",
    "pythia-6.9b-deduped",
    "pythia-70m",
    "pythia-70m-deduped",
    "stable-code-3b",
    "stablecode-completion-alpha-3b",
    "stablecode-completion-alpha-3b-4k",
    "stablecode-instruct-alpha-3b",
    "stablelm-3b-4e1t",
    "stablelm-base-alpha-3b",
    "stablelm-base-alpha-7b",
    "stablelm-tuned-alpha-3b",
    "stablelm-tuned-alpha-7b",
    "stablelm-zephyr-3b",
    "tiny-llama-1.1b",
    "tiny-llama-1.1b-chat",
    "vicuna-13b-v1.3",
    "vicuna-13b-v1.5",
    "vicuna-13b-v1.5-16k",
    "vicuna-33b-v1.3",
    "vicuna-7b-v1.3",
    "vicuna-7b-v1.5",
    "vicuna-7b-v1.5-16k",
]


def test_reference_name_count():
    assert len(REFERENCE_NAMES) == 112


@pytest.mark.parametrize("name", REFERENCE_NAMES)
def test_reference_config_resolves(name):
    from mdi_llm_amd.config import ModelConfig

    cfg = ModelConfig.from_name(name)
    assert cfg.n_layer > 0 and cfg.n_embd > 0 and cfg.n_head > 0
    assert cfg.padded_vocab_size is None or \
        cfg.padded_vocab_size >= cfg.vocab_size


SPOT = {
    # name: (n_layer, n_head, n_embd, n_query_groups, mlp, norm)
    "stablelm-zephyr-3b": (32, 32, 2560, 32, "LLaMAMLP", "LayerNorm"),
    "falcon-40b": (60, 128, 8192, 8, "GptNeoxMLP", "LayerNorm"),
    "CodeLlama-70b-Instruct-hf": (80, 64, 8192, 8, "LLaMAMLP", "RMSNorm"),
    "vicuna-13b-v1.5-16k": (40, 40, 5120, 40, "LLaMAMLP", "RMSNorm"),
    "longchat-7b-16k": (32, 32, 4096, 32, "LLaMAMLP", "RMSNorm"),
    "dolly-v2-12b": (36, 40, 5120, 40, "GptNeoxMLP", "LayerNorm"),
    "RedPajama-INCITE-7B-Chat": (32, 32, 4096, 32, "GptNeoxMLP",
                                 "LayerNorm"),
    "open_llama_13b": (40, 40, 5120, 40, "LLaMAMLP", "RMSNorm"),
    "Platypus2-70B-instruct": (80, 64, 8192, 8, "LLaMAMLP", "RMSNorm"),
    "Danube2-1.8b-chat": (24, 32, 2560, 8, "LLaMAMLP", "RMSNorm"),
    "CodeGemma-7b-it": (28, 16, 3072, 16, "GemmaMLP", "RMSNorm"),
    "Mistral-7B-Instruct-v0.3": (32, 32, 4096, 8, "LLaMAMLP", "RMSNorm"),
    "FreeWilly2": (80, 64, 8192, 8, "LLaMAMLP", "RMSNorm"),
}


@pytest.mark.parametrize("name", sorted(SPOT))
def test_reference_config_spot_fields(name):
    from mdi_llm_amd.config import ModelConfig

    cfg = ModelConfig.from_name(name)
    nl, nh, ne, ng, mlp, norm = SPOT[name]
    assert cfg.n_layer == nl
    assert cfg.n_head == nh
    assert cfg.n_embd == ne
    assert (cfg.n_query_groups or cfg.n_head) == ng
    assert cfg.mlp_class_name == mlp
    assert cfg.norm_class_name == norm
