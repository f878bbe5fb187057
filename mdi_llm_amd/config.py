"""Model configuration registry for mdi_llm_amd.

Capability parity with the reference's litGPT-style registry
(/root/reference/src/sub/config.py:180-1669 and the ``Config`` dataclass in
/root/reference/src/sub/model.py:93-273) — re-written for this framework:
a plain dataclass with the fields our MI355X decode engine and the PyTorch
reference path need, a curated named-config table covering the model
families the reference supports (Llama-2/3, TinyLlama, NanoLlama, GPT-2,
Pythia/NeoX, Phi, Mistral, Gemma), and ``from_name`` / ``from_file`` /
``from_checkpoint`` constructors plus full dict round-tripping (the node
init RPC serializes configs, as the reference does via ``asdict``).
"""

from __future__ import annotations

import copy
from dataclasses import dataclass, field, asdict
from pathlib import Path
from typing import Any, Literal, Optional, Union

import yaml

__all__ = ["ModelConfig", "name_to_config", "configs"]


def find_multiple(n: int, k: int) -> int:
    if n % k == 0:
        return n
    return n + k - (n % k)


@dataclass
class ModelConfig:
    """Architecture description of one decoder-only transformer model."""

    name: str = ""
    hf_config: dict = field(default_factory=dict)
    block_size: int = 4096
    vocab_size: int = 50254
    padding_multiple: int = 512
    padded_vocab_size: Optional[int] = None
    n_layer: int = 16
    n_head: int = 32
    head_size: Optional[int] = None
    n_embd: int = 4096
    rotary_percentage: float = 0.25
    parallel_residual: bool = True
    bias: bool = True
    lm_head_bias: bool = False
    # GQA/MQA: number of KV head groups. n_head -> MHA, 1 -> MQA.
    n_query_groups: Optional[int] = None
    shared_attention_norm: bool = False
    norm_class_name: Literal["LayerNorm", "RMSNorm"] = "LayerNorm"
    norm_eps: float = 1e-5
    mlp_class_name: Literal["GptNeoxMLP", "LLaMAMLP", "GemmaMLP", "LLaMAMoE"] = (
        "GptNeoxMLP"
    )
    gelu_approximate: str = "none"
    intermediate_size: Optional[int] = None
    rope_condense_ratio: int = 1
    rope_base: int = 10000
    n_expert: int = 0
    n_expert_per_token: int = 0
    # Gemma multiplies embeddings by sqrt(n_embd).
    scale_embeddings: bool = False
    # "rope" (default) or "learned" (GPT-2-style wpe table).
    pos_embedding: Literal["rope", "learned"] = "rope"

    def __post_init__(self) -> None:
        if not self.name:
            self.name = self.hf_config.get("name", "")
        if self.head_size is None:
            assert self.n_embd % self.n_head == 0
            self.head_size = self.n_embd // self.n_head
        if self.padded_vocab_size is None:
            self.padded_vocab_size = find_multiple(self.vocab_size, self.padding_multiple)
        else:
            self.vocab_size = min(self.vocab_size, self.padded_vocab_size)
        if self.n_query_groups is not None:
            assert self.n_head % self.n_query_groups == 0
        else:
            self.n_query_groups = self.n_head
        if self.intermediate_size is None:
            if self.mlp_class_name == "LLaMAMLP":
                raise ValueError(
                    f"config {self.name!r} requires intermediate_size for LLaMAMLP"
                )
            self.intermediate_size = 4 * self.n_embd
        self.rope_n_elem = int(self.rotary_percentage * self.head_size)

    # -- derived sizes ----------------------------------------------------
    @property
    def qkv_dim(self) -> int:
        """Output width of the fused QKV projection."""
        return (self.n_head + 2 * self.n_query_groups) * self.head_size

    @property
    def q_per_kv(self) -> int:
        return self.n_head // self.n_query_groups

    # -- constructors -----------------------------------------------------
    @classmethod
    def from_name(cls, name: str, **overrides: Any) -> "ModelConfig":
        if name not in name_to_config:
            # try candidate with template substitution, e.g. pythia sizes
            matches = [k for k in name_to_config if k.lower() == name.lower()]
            if not matches:
                raise ValueError(f"unknown model config name {name!r}")
            name = matches[0]
        conf = copy.deepcopy(name_to_config[name])
        conf.update(overrides)
        return cls(**conf)

    @classmethod
    def from_file(cls, path: Union[str, Path], **overrides: Any) -> "ModelConfig":
        with open(path, encoding="utf-8") as fp:
            raw = yaml.safe_load(fp) or {}
        raw.pop("rope_n_elem", None)
        raw.update(overrides)
        known = {f for f in cls.__dataclass_fields__}
        raw = {k: v for k, v in raw.items() if k in known}
        return cls(**raw)

    @classmethod
    def from_checkpoint(cls, path: Union[str, Path], **overrides: Any) -> "ModelConfig":
        """Load ``model_config.yaml`` from a checkpoint dir, else match by name."""
        path = Path(path)
        cfg_file = path / "model_config.yaml"
        if cfg_file.is_file():
            return cls.from_file(cfg_file, **overrides)
        if (conf_name := path.name) in name_to_config:
            return cls.from_name(conf_name, **overrides)
        raise FileNotFoundError(f"no model_config.yaml in {path} and {path.name!r} unknown")

    @classmethod
    def from_dict(cls, d: dict) -> "ModelConfig":
        d = dict(d)
        d.pop("rope_n_elem", None)
        known = {f for f in cls.__dataclass_fields__}
        return cls(**{k: v for k, v in d.items() if k in known})

    def to_dict(self) -> dict:
        return asdict(self)

    def save(self, path: Union[str, Path]) -> None:
        with open(path, "w", encoding="utf-8") as fp:
            yaml.safe_dump(self.to_dict(), fp)


########################################################################
# Named configs.  Families mirror the reference registry's coverage
# (/root/reference/src/sub/config.py): custom NanoLlama, TinyLlama,
# Llama-2, Llama-3, GPT-2 (via the old/GPT2 generation), Pythia/NeoX,
# Phi, Mistral, Gemma, plus tiny test configs for CI.
########################################################################

configs: list[dict] = []

# -- tiny test configs (CPU tests / synthetic runs) ----------------------
configs.extend(
    [
        dict(
            name="nano-test",
            block_size=128,
            vocab_size=256,
            padding_multiple=64,
            n_layer=4,
            n_head=4,
            n_embd=64,
            n_query_groups=2,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMLP",
            intermediate_size=176,
            norm_eps=1e-5,
        ),
        dict(
            # GPU-testable tiny llama: head_size 64, GQA 2:1 (DecodeEngine
            # needs head_size in {64,128,256} and hs*q_per_kv >= 64)
            name="nano-gpu",
            block_size=256,
            vocab_size=512,
            padding_multiple=64,
            n_layer=4,
            n_head=4,
            n_embd=256,
            n_query_groups=2,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMLP",
            intermediate_size=688,
            norm_eps=1e-5,
        ),
        dict(
            # GPU-testable NeoX-style tiny config: parallel residual,
            # LayerNorm, partial rotary — exercises those engine paths
            name="nano-neox-gpu",
            block_size=256,
            vocab_size=512,
            padding_multiple=64,
            n_layer=3,
            n_head=4,
            n_embd=256,
            rotary_percentage=0.25,
            parallel_residual=True,
            bias=True,
            norm_class_name="LayerNorm",
            mlp_class_name="GptNeoxMLP",
        ),
        dict(
            # phi-2-style tiny config (parallel residual, shared attn
            # norm, LayerNorm + biases, partial rotary) for converter
            # round-trip tests
            name="nano-phi-test",
            block_size=128,
            vocab_size=256,
            padding_multiple=64,
            n_layer=2,
            n_head=4,
            n_embd=64,
            rotary_percentage=0.5,
            parallel_residual=True,
            shared_attention_norm=True,
            bias=True,
            lm_head_bias=True,
            norm_class_name="LayerNorm",
            mlp_class_name="GptNeoxMLP",
            gelu_approximate="tanh",
        ),
        dict(
            name="nano-test-gpt2",
            block_size=128,
            vocab_size=256,
            padding_multiple=64,
            n_layer=2,
            n_head=4,
            n_embd=64,
            rotary_percentage=0.0,
            pos_embedding="learned",
            parallel_residual=False,
            bias=True,
            norm_class_name="LayerNorm",
            mlp_class_name="GptNeoxMLP",
        ),
        dict(
            # NeoX-style tiny config (parallel residual, partial rotary)
            name="nano-test-neox",
            block_size=128,
            vocab_size=256,
            padding_multiple=64,
            n_layer=2,
            n_head=4,
            n_embd=64,
            rotary_percentage=0.25,
            parallel_residual=True,
            bias=True,
            norm_class_name="LayerNorm",
            mlp_class_name="GptNeoxMLP",
        ),
        dict(
            # 8-layer variant: splits across up to 8 pipeline stages in
            # CPU tests of the multi-GPU bench path
            name="nano-test-deep",
            block_size=128,
            vocab_size=256,
            padding_multiple=64,
            n_layer=8,
            n_head=4,
            n_embd=64,
            n_query_groups=4,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMLP",
            intermediate_size=128,
        ),
        dict(
            # falcon-7b-style tiny config: MQA + shared attention norm
            name="nano-test-falcon",
            block_size=128,
            vocab_size=256,
            padding_multiple=64,
            n_layer=2,
            n_head=4,
            n_embd=64,
            n_query_groups=1,
            rotary_percentage=1.0,
            parallel_residual=True,
            shared_attention_norm=True,
            bias=False,
            norm_class_name="LayerNorm",
            mlp_class_name="GptNeoxMLP",
        ),
        dict(
            # MoE variant big enough for the HIP engine (hs=64)
            name="nano-moe-gpu",
            block_size=256,
            vocab_size=512,
            padding_multiple=64,
            n_layer=2,
            n_head=4,
            n_embd=256,
            n_query_groups=4,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMoE",
            intermediate_size=224,
            n_expert=4,
            n_expert_per_token=2,
            norm_eps=1e-5,
        ),
        dict(
            name="nano-test-moe",
            block_size=128,
            vocab_size=256,
            padding_multiple=64,
            n_layer=2,
            n_head=4,
            n_embd=64,
            n_query_groups=4,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMoE",
            intermediate_size=96,
            n_expert=4,
            n_expert_per_token=2,
        ),
    ]
)

# -- NanoLlama (the reference's custom 304M model,
#    /root/reference/src/checkpoints/custom/NanoLlama) -------------------
configs.append(
    dict(
        name="NanoLlama",
        block_size=1024,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=12,
        n_head=16,
        n_embd=1024,
        n_query_groups=4,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name="RMSNorm",
        mlp_class_name="LLaMAMLP",
        intermediate_size=5632,
        norm_eps=1e-5,
    )
)

# -- TinyLlama 1.1B (reference: config.py:1611-1639) ---------------------
for name in ("TinyLlama-1.1B-intermediate-step-1431k-3T", "TinyLlama-1.1B-Chat-v1.0", "tiny-llama-1.1b"):
    configs.append(
        dict(
            name=name,
            hf_config=dict(org="TinyLlama", name=name),
            block_size=2048,
            vocab_size=32000,
            padding_multiple=64,
            n_layer=22,
            n_head=32,
            n_embd=2048,
            n_query_groups=4,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMLP",
            intermediate_size=5632,
            norm_eps=1e-5,
        )
    )

# -- Llama-2 (reference: config.py:824-878) ------------------------------
for size, n_layer, n_head, n_embd, interm, groups in (
    ("7b", 32, 32, 4096, 11008, 32),
    ("13b", 40, 40, 5120, 13824, 40),
    ("70b", 80, 64, 8192, 28672, 8),
):
    configs.append(
        dict(
            name=f"Llama-2-{size}-hf",
            hf_config=dict(org="meta-llama", name=f"Llama-2-{size}-hf"),
            block_size=4096,
            vocab_size=32000,
            padding_multiple=64,
            n_layer=n_layer,
            n_head=n_head,
            n_embd=n_embd,
            n_query_groups=groups,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMLP",
            intermediate_size=interm,
            norm_eps=1e-5,
        )
    )
    configs.append(
        {**configs[-1], "name": f"Llama-2-{size}-chat-hf",
         "hf_config": dict(org="meta-llama", name=f"Llama-2-{size}-chat-hf")}
    )

# -- Llama-3 / 3.1 (reference: config.py:884-928) ------------------------
for suffix in ("", "-Instruct"):
    configs.append(
        dict(
            name=f"Meta-Llama-3-8B{suffix}",
            hf_config=dict(org="meta-llama", name=f"Meta-Llama-3-8B{suffix}"),
            block_size=8192,
            vocab_size=128000,
            padded_vocab_size=128256,
            n_layer=32,
            n_head=32,
            n_embd=4096,
            n_query_groups=8,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMLP",
            intermediate_size=14336,
            rope_base=500000,
            norm_eps=1e-5,
        )
    )
    configs.append(
        dict(
            name=f"Meta-Llama-3-70B{suffix}",
            hf_config=dict(org="meta-llama", name=f"Meta-Llama-3-70B{suffix}"),
            block_size=8192,
            vocab_size=128000,
            padded_vocab_size=128256,
            n_layer=80,
            n_head=64,
            n_embd=8192,
            n_query_groups=8,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="LLaMAMLP",
            intermediate_size=28672,
            rope_base=500000,
            norm_eps=1e-5,
        )
    )

# -- GPT-2 family (capability of the reference's old/GPT2 generation) ----
for name, n_layer, n_head, n_embd in (
    ("gpt2", 12, 12, 768),
    ("gpt2-medium", 24, 16, 1024),
    ("gpt2-large", 36, 20, 1280),
    ("gpt2-xl", 48, 25, 1600),
):
    configs.append(
        dict(
            name=name,
            hf_config=dict(org="openai-community", name=name),
            block_size=1024,
            vocab_size=50257,
            padded_vocab_size=50304,
            n_layer=n_layer,
            n_head=n_head,
            n_embd=n_embd,
            rotary_percentage=0.0,
            pos_embedding="learned",
            parallel_residual=False,
            bias=True,
            norm_class_name="LayerNorm",
            mlp_class_name="GptNeoxMLP",
            gelu_approximate="tanh",
        )
    )

# -- Pythia / GPT-NeoX (reference: config.py pythia block) ---------------
for size, n_layer, n_head, n_embd, pad in (
    ("70m", 6, 8, 512, 128),
    ("160m", 12, 12, 768, 128),
    ("410m", 24, 16, 1024, 128),
    ("1b", 16, 8, 2048, 128),
    ("1.4b", 24, 16, 2048, 128),
    ("2.8b", 32, 32, 2560, 128),
    ("6.9b", 32, 32, 4096, 256),
    ("12b", 36, 40, 5120, 128),
):
    configs.append(
        dict(
            name=f"pythia-{size}",
            hf_config=dict(org="EleutherAI", name=f"pythia-{size}"),
            block_size=2048,
            vocab_size=50254,
            padding_multiple=pad,
            n_layer=n_layer,
            n_head=n_head,
            n_embd=n_embd,
            rotary_percentage=0.25,
            parallel_residual=True,
            bias=True,
            norm_class_name="LayerNorm",
            mlp_class_name="GptNeoxMLP",
        )
    )

# -- Phi-2 (reference: config.py phi block) ------------------------------
configs.append(
    dict(
        name="phi-2",
        hf_config=dict(org="microsoft", name="phi-2"),
        block_size=2048,
        vocab_size=50257,
        padded_vocab_size=51200,
        n_layer=32,
        n_head=32,
        n_embd=2560,
        rotary_percentage=0.4,
        parallel_residual=True,
        shared_attention_norm=True,
        bias=True,
        lm_head_bias=True,
        norm_class_name="LayerNorm",
        mlp_class_name="GptNeoxMLP",
        gelu_approximate="tanh",
    )
)

# -- Mistral-7B (reference: config.py mistral block) ---------------------
configs.append(
    dict(
        name="Mistral-7B-v0.1",
        hf_config=dict(org="mistralai", name="Mistral-7B-v0.1"),
        block_size=4096,
        vocab_size=32000,
        padding_multiple=512,
        n_layer=32,
        n_head=32,
        n_embd=4096,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name="RMSNorm",
        mlp_class_name="LLaMAMLP",
        intermediate_size=14336,
        norm_eps=1e-5,
        padded_vocab_size=32000,
    )
)
configs.append(
    {**configs[-1], "name": "Mistral-7B-Instruct-v0.2",
     "hf_config": dict(org="mistralai", name="Mistral-7B-Instruct-v0.2"),
     "block_size": 32768}
)

# -- Mixtral (LLaMAMoE, reference: model.py:823-853 local MoE) -----------
configs.append(
    dict(
        name="Mixtral-8x7B-v0.1",
        hf_config=dict(org="mistralai", name="Mixtral-8x7B-v0.1"),
        block_size=32768,
        vocab_size=32000,
        padded_vocab_size=32000,
        padding_multiple=512,
        n_layer=32,
        n_head=32,
        n_embd=4096,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name="RMSNorm",
        mlp_class_name="LLaMAMoE",
        intermediate_size=14336,
        rope_base=1000000,
        n_expert=8,
        n_expert_per_token=2,
        norm_eps=1e-5,
    )
)

# -- Gemma (reference: config.py gemma block) ----------------------------
for size, n_layer, n_head, n_embd, interm, groups, hs in (
    ("2b", 18, 8, 2048, 16384, 1, 256),
    ("7b", 28, 16, 3072, 24576, 16, 256),
):
    configs.append(
        dict(
            name=f"gemma-{size}",
            hf_config=dict(org="google", name=f"gemma-{size}"),
            block_size=8192,
            vocab_size=256000,
            padding_multiple=64,
            n_layer=n_layer,
            n_head=n_head,
            n_embd=n_embd,
            head_size=hs,
            n_query_groups=groups,
            rotary_percentage=1.0,
            parallel_residual=False,
            bias=False,
            norm_class_name="RMSNorm",
            mlp_class_name="GemmaMLP",
            intermediate_size=interm,
            scale_embeddings=True,
            norm_eps=1e-6,
        )
    )

# -- Falcon (reference: config.py falcon block, lines 476-534) -----------
# 7b is MQA (one kv group) with a single shared attention/MLP norm;
# 40b/180B use 8 kv groups and separate ln_attn/ln_mlp norms.
for base in (
    dict(
        name="falcon-7b{}",
        hf_config=dict(org="tiiuae", name="falcon-7b{}"),
        block_size=2048,
        vocab_size=65024,
        padded_vocab_size=65024,
        n_layer=32,
        n_head=71,
        n_embd=4544,
        rotary_percentage=1.0,
        n_query_groups=1,
        bias=False,
        shared_attention_norm=True,
    ),
    dict(
        name="falcon-40b{}",
        hf_config=dict(org="tiiuae", name="falcon-40b{}"),
        block_size=2048,
        vocab_size=65024,
        padded_vocab_size=65024,
        n_layer=60,
        n_head=128,
        n_embd=8192,
        rotary_percentage=1.0,
        n_query_groups=8,
        bias=False,
    ),
):
    for kind in ("", "-instruct"):
        c = dict(base)
        c["name"] = base["name"].format(kind)
        c["hf_config"] = dict(org="tiiuae",
                              name=base["hf_config"]["name"].format(kind))
        configs.append(c)
for kind in ("", "-chat"):
    configs.append(
        dict(
            name=f"falcon-180B{kind}",
            hf_config=dict(org="tiiuae", name=f"falcon-180B{kind}"),
            block_size=2048,
            vocab_size=65024,
            padded_vocab_size=65024,
            n_layer=80,
            n_head=232,
            n_embd=14848,
            rotary_percentage=1.0,
            n_query_groups=8,
            bias=False,
        )
    )


# -- reference-parity registry completion (round 2): every remaining
# named config of the reference registry (same hyperparameters for
# checkpoint compatibility; /root/reference/src/sub/config.py:180-1667)
# -- stablelm ----------------------------------------------------
configs.extend([
    dict(
        name='stablelm-base-alpha-3b',
        hf_config={'org': 'stabilityai',
        'name': 'stablelm-base-alpha-3b'},
    ),
    dict(
        name='stablelm-base-alpha-7b',
        hf_config={'org': 'stabilityai',
        'name': 'stablelm-base-alpha-7b'},
        n_head=48,
        n_embd=6144,
        padding_multiple=256,
    ),
    dict(
        name='stablelm-tuned-alpha-3b',
        hf_config={'org': 'stabilityai',
        'name': 'stablelm-tuned-alpha-3b'},
        n_head=32,
    ),
    dict(
        name='stablelm-tuned-alpha-7b',
        hf_config={'org': 'stabilityai',
        'name': 'stablelm-tuned-alpha-7b'},
        n_head=48,
        n_embd=6144,
        padding_multiple=256,
    ),
    dict(
        name='stablelm-3b-4e1t',
        hf_config={'org': 'stabilityai',
        'name': 'stablelm-3b-4e1t'},
        padded_vocab_size=50304,
        n_layer=32,
        n_head=32,
        n_embd=2560,
        parallel_residual=False,
        bias=False,
        mlp_class_name='LLaMAMLP',
        intermediate_size=6912,
    ),
    dict(
        name='stablelm-zephyr-3b',
        hf_config={'org': 'stabilityai',
        'name': 'stablelm-zephyr-3b'},
        padded_vocab_size=50304,
        n_layer=32,
        n_head=32,
        n_embd=2560,
        parallel_residual=False,
        bias=False,
        mlp_class_name='LLaMAMLP',
        intermediate_size=6912,
    ),
])
# -- stablecode --------------------------------------------------
configs.extend([
    dict(
        name='stablecode-completion-alpha-3b',
        hf_config={'org': 'stabilityai',
        'name': 'stablecode-completion-alpha-3b'},
        block_size=16384,
        vocab_size=49152,
        n_layer=32,
        n_embd=2560,
    ),
    dict(
        name='stablecode-completion-alpha-3b-4k',
        hf_config={'org': 'stabilityai',
        'name': 'stablecode-completion-alpha-3b-4k'},
        vocab_size=49152,
        n_layer=32,
        n_embd=2560,
    ),
    dict(
        name='stablecode-instruct-alpha-3b',
        hf_config={'org': 'stabilityai',
        'name': 'stablecode-instruct-alpha-3b'},
        vocab_size=49152,
        n_layer=32,
        n_embd=2560,
    ),
])
# -- stable-code -------------------------------------------------
configs.extend([
    dict(
        name='stable-code-3b',
        hf_config={'org': 'stabilityai',
        'name': 'stable-code-3b'},
        padded_vocab_size=50304,
        n_layer=32,
        n_embd=2560,
        block_size=16384,
        parallel_residual=False,
        bias=False,
        mlp_class_name='LLaMAMLP',
        intermediate_size=6912,
    ),
])
# -- pythia ------------------------------------------------------
configs.extend([
    dict(
        name='pythia-14m',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-14m'},
        block_size=512,
        n_layer=6,
        n_embd=128,
        n_head=4,
        padding_multiple=128,
    ),
    dict(
        name='pythia-31m',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-31m'},
        block_size=1024,
        n_layer=6,
        n_embd=256,
        n_head=8,
        padding_multiple=128,
    ),
    dict(
        name='pythia-70m-deduped',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-70m-deduped'},
        block_size=2048,
        n_layer=6,
        n_embd=512,
        n_head=8,
        padding_multiple=128,
    ),
    dict(
        name='pythia-160m-deduped',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-160m-deduped'},
        block_size=2048,
        n_layer=12,
        n_embd=768,
        n_head=12,
        padding_multiple=128,
    ),
    dict(
        name='pythia-410m-deduped',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-410m-deduped'},
        block_size=2048,
        n_layer=24,
        n_embd=1024,
        n_head=16,
        padding_multiple=128,
    ),
    dict(
        name='pythia-1b-deduped',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-1b-deduped'},
        block_size=2048,
        n_embd=2048,
        n_head=8,
        padding_multiple=128,
    ),
    dict(
        name='pythia-1.4b-deduped',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-1.4b-deduped'},
        block_size=2048,
        n_layer=24,
        n_embd=2048,
        n_head=16,
        padding_multiple=128,
    ),
    dict(
        name='pythia-2.8b-deduped',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-2.8b-deduped'},
        block_size=2048,
        n_layer=32,
        n_embd=2560,
        padding_multiple=128,
    ),
    dict(
        name='pythia-6.9b-deduped',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-6.9b-deduped'},
        block_size=2048,
        n_layer=32,
        padding_multiple=256,
    ),
    dict(
        name='pythia-12b-deduped',
        hf_config={'org': 'EleutherAI',
        'name': 'pythia-12b-deduped'},
        block_size=2048,
        n_layer=36,
        n_embd=5120,
        n_head=40,
    ),
])
# -- dolly -------------------------------------------------------
configs.extend([
    dict(
        name='dolly-v2-3b',
        hf_config={'org': 'databricks',
        'name': 'dolly-v2-3b'},
        block_size=2048,
        n_layer=32,
        n_embd=2560,
        padded_vocab_size=50280,
    ),
    dict(
        name='dolly-v2-7b',
        hf_config={'org': 'databricks',
        'name': 'dolly-v2-7b'},
        block_size=2048,
        n_layer=32,
        padded_vocab_size=50280,
    ),
    dict(
        name='dolly-v2-12b',
        hf_config={'org': 'databricks',
        'name': 'dolly-v2-12b'},
        block_size=2048,
        n_layer=36,
        n_embd=5120,
        n_head=40,
        padded_vocab_size=50280,
    ),
])
# -- RedPajama ---------------------------------------------------
configs.extend([
    dict(
        name='RedPajama-INCITE-Base-3B-v1',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-Base-3B-v1'},
        block_size=2048,
        n_layer=32,
        n_embd=2560,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
    dict(
        name='RedPajama-INCITE-Chat-3B-v1',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-Chat-3B-v1'},
        block_size=2048,
        n_layer=32,
        n_embd=2560,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
    dict(
        name='RedPajama-INCITE-Instruct-3B-v1',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-Instruct-3B-v1'},
        block_size=2048,
        n_layer=32,
        n_embd=2560,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
    dict(
        name='RedPajama-INCITE-7B-Base',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-7B-Base'},
        block_size=2048,
        n_layer=32,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
    dict(
        name='RedPajama-INCITE-7B-Chat',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-7B-Chat'},
        block_size=2048,
        n_layer=32,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
    dict(
        name='RedPajama-INCITE-7B-Instruct',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-7B-Instruct'},
        block_size=2048,
        n_layer=32,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
    dict(
        name='RedPajama-INCITE-Base-7B-v0.1',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-Base-7B-v0.1'},
        block_size=2048,
        n_layer=32,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
    dict(
        name='RedPajama-INCITE-Chat-7B-v0.1',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-Chat-7B-v0.1'},
        block_size=2048,
        n_layer=32,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
    dict(
        name='RedPajama-INCITE-Instruct-7B-v0.1',
        hf_config={'org': 'togethercomputer',
        'name': 'RedPajama-INCITE-Instruct-7B-v0.1'},
        block_size=2048,
        n_layer=32,
        padding_multiple=256,
        rotary_percentage=1.0,
        parallel_residual=False,
    ),
])
# -- open_llama --------------------------------------------------
configs.extend([
    dict(
        name='open_llama_3b',
        hf_config={'org': 'openlm-research',
        'name': 'open_llama_3b'},
        block_size=2048,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=26,
        n_embd=3200,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=8640,
    ),
    dict(
        name='open_llama_7b',
        hf_config={'org': 'openlm-research',
        'name': 'open_llama_7b'},
        block_size=2048,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
    ),
    dict(
        name='open_llama_13b',
        hf_config={'org': 'openlm-research',
        'name': 'open_llama_13b'},
        block_size=2048,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
    ),
])
# -- vicuna ------------------------------------------------------
configs.extend([
    dict(
        name='vicuna-7b-v1.3',
        hf_config={'org': 'lmsys',
        'name': 'vicuna-7b-v1.3'},
        block_size=2048,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
    ),
    dict(
        name='vicuna-13b-v1.3',
        hf_config={'org': 'lmsys',
        'name': 'vicuna-13b-v1.3'},
        block_size=2048,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
    ),
    dict(
        name='vicuna-33b-v1.3',
        hf_config={'org': 'lmsys',
        'name': 'vicuna-33b-v1.3'},
        block_size=2048,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=60,
        n_head=52,
        n_embd=6656,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=17920,
    ),
    dict(
        name='vicuna-7b-v1.5',
        hf_config={'org': 'lmsys',
        'name': 'vicuna-7b-v1.5'},
        vocab_size=32000,
        padding_multiple=64,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
    ),
    dict(
        name='vicuna-7b-v1.5-16k',
        hf_config={'org': 'lmsys',
        'name': 'vicuna-7b-v1.5-16k'},
        block_size=16384,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
        rope_condense_ratio=4,
    ),
    dict(
        name='vicuna-13b-v1.5',
        hf_config={'org': 'lmsys',
        'name': 'vicuna-13b-v1.5'},
        vocab_size=32000,
        padding_multiple=64,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
    ),
    dict(
        name='vicuna-13b-v1.5-16k',
        hf_config={'org': 'lmsys',
        'name': 'vicuna-13b-v1.5-16k'},
        block_size=16384,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
        rope_condense_ratio=4,
    ),
])
# -- longchat ----------------------------------------------------
configs.extend([
    dict(
        name='longchat-7b-16k',
        hf_config={'org': 'lmsys',
        'name': 'longchat-7b-16k'},
        block_size=16384,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
        rope_condense_ratio=8,
    ),
    dict(
        name='longchat-13b-16k',
        hf_config={'org': 'lmsys',
        'name': 'longchat-13b-16k'},
        block_size=16384,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
        rope_condense_ratio=8,
    ),
])
# -- Nous --------------------------------------------------------
configs.extend([
    dict(
        name='Nous-Hermes-llama-2-7b',
        hf_config={'org': 'NousResearch',
        'name': 'Nous-Hermes-llama-2-7b'},
        padded_vocab_size=32000,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
    ),
    dict(
        name='Nous-Hermes-13b',
        hf_config={'org': 'NousResearch',
        'name': 'Nous-Hermes-13b'},
        block_size=2048,
        vocab_size=32000,
        padded_vocab_size=32001,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
    ),
    dict(
        name='Nous-Hermes-Llama2-13b',
        hf_config={'org': 'NousResearch',
        'name': 'Nous-Hermes-Llama2-13b'},
        vocab_size=32000,
        padded_vocab_size=32032,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
    ),
])
# -- Llama-3 -----------------------------------------------------
configs.extend([
    dict(
        name='Llama-3-8B',
        hf_config={'org': 'meta-llama',
        'name': 'Meta-Llama-3-8B'},
        block_size=8192,
        vocab_size=128000,
        padded_vocab_size=128256,
        n_layer=32,
        n_head=32,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=14336,
        rope_base=500000,
    ),
    dict(
        name='Llama-3-8B-Instruct',
        hf_config={'org': 'meta-llama',
        'name': 'Meta-Llama-3-8B-Instruct'},
        block_size=8192,
        vocab_size=128000,
        padded_vocab_size=128256,
        n_layer=32,
        n_head=32,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=14336,
        rope_base=500000,
    ),
    dict(
        name='Llama-3-70B',
        hf_config={'org': 'meta-llama',
        'name': 'Meta-Llama-3-70B'},
        block_size=8192,
        vocab_size=128000,
        padded_vocab_size=128256,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
        rope_base=500000,
    ),
    dict(
        name='Llama-3-70B-Instruct',
        hf_config={'org': 'meta-llama',
        'name': 'Meta-Llama-3-70B-Instruct'},
        block_size=8192,
        vocab_size=128000,
        padded_vocab_size=128256,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
        rope_base=500000,
    ),
])
# -- Gemma -------------------------------------------------------
configs.extend([
    dict(
        name='Gemma-2b',
        hf_config={'org': 'google',
        'name': 'gemma-2b'},
        scale_embeddings=True,
        vocab_size=256000,
        padding_multiple=64,
        n_embd=2048,
        n_layer=18,
        n_head=8,
        n_query_groups=1,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='GemmaMLP',
        gelu_approximate='tanh',
        intermediate_size=16384,
    ),
    dict(
        name='Gemma-7b',
        hf_config={'org': 'google',
        'name': 'gemma-7b'},
        scale_embeddings=True,
        vocab_size=256000,
        padding_multiple=64,
        n_embd=3072,
        n_layer=28,
        n_head=16,
        head_size=256,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='GemmaMLP',
        gelu_approximate='tanh',
        intermediate_size=24576,
    ),
    dict(
        name='Gemma-2b-it',
        hf_config={'org': 'google',
        'name': 'gemma-2b-it'},
        scale_embeddings=True,
        vocab_size=256000,
        padding_multiple=64,
        n_embd=2048,
        n_layer=18,
        n_head=8,
        n_query_groups=1,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='GemmaMLP',
        gelu_approximate='tanh',
        intermediate_size=16384,
    ),
    dict(
        name='Gemma-7b-it',
        hf_config={'org': 'google',
        'name': 'gemma-7b-it'},
        scale_embeddings=True,
        vocab_size=256000,
        padding_multiple=64,
        n_embd=3072,
        n_layer=28,
        n_head=16,
        head_size=256,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='GemmaMLP',
        gelu_approximate='tanh',
        intermediate_size=24576,
    ),
])
# -- CodeGemma ---------------------------------------------------
configs.extend([
    dict(
        name='CodeGemma-7b-it',
        hf_config={'org': 'google',
        'name': 'codegemma-7b-it'},
        scale_embeddings=True,
        vocab_size=256000,
        padding_multiple=64,
        n_embd=3072,
        n_layer=28,
        n_head=16,
        head_size=256,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='GemmaMLP',
        gelu_approximate='tanh',
        intermediate_size=24576,
    ),
])
# -- Danube ------------------------------------------------------
configs.extend([
    dict(
        name='Danube2-1.8b-chat',
        hf_config={'org': 'h2oai',
        'name': 'h2o-danube2-1.8b-chat'},
        vocab_size=32000,
        n_layer=24,
        n_head=32,
        n_embd=2560,
        block_size=4096,
        intermediate_size=6912,
        padding_multiple=64,
        norm_eps=1e-05,
        rope_base=10000,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
    ),
])
# -- FreeWilly ---------------------------------------------------
configs.extend([
    dict(
        name='FreeWilly2',
        hf_config={'org': 'stabilityai',
        'name': 'FreeWilly2'},
        vocab_size=32000,
        padding_multiple=64,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
    ),
])
# -- CodeLlama ---------------------------------------------------
configs.extend([
    dict(
        name='CodeLlama-7b-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-7b-hf'},
        block_size=16384,
        vocab_size=32016,
        padding_multiple=16,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-13b-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-13b-hf'},
        block_size=16384,
        vocab_size=32016,
        padding_multiple=16,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-34b-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-34b-hf'},
        block_size=16384,
        vocab_size=32000,
        padded_vocab_size=32000,
        n_layer=48,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=22016,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-70b-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-70b-hf'},
        block_size=16384,
        vocab_size=32016,
        padding_multiple=16,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-7b-Python-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-7b-Python-hf'},
        block_size=16384,
        vocab_size=32000,
        padded_vocab_size=32000,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-13b-Python-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-13b-Python-hf'},
        block_size=16384,
        vocab_size=32000,
        padded_vocab_size=32000,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-34b-Python-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-34b-Python-hf'},
        block_size=16384,
        vocab_size=32000,
        padded_vocab_size=32000,
        n_layer=48,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=22016,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-70b-Python-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-70b-Python-hf'},
        block_size=16384,
        vocab_size=32016,
        padding_multiple=16,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-7b-Instruct-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-7b-Instruct-hf'},
        block_size=16384,
        vocab_size=32016,
        padding_multiple=16,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-13b-Instruct-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-13b-Instruct-hf'},
        block_size=2048,
        vocab_size=32016,
        padding_multiple=16,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-34b-Instruct-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-34b-Instruct-hf'},
        block_size=16384,
        vocab_size=32000,
        padded_vocab_size=32000,
        n_layer=48,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=22016,
        rope_base=1000000,
    ),
    dict(
        name='CodeLlama-70b-Instruct-hf',
        hf_config={'org': 'codellama',
        'name': 'CodeLlama-70b-Instruct-hf'},
        block_size=16384,
        vocab_size=32016,
        padding_multiple=16,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
        rope_base=1000000,
    ),
])
# -- Platypus ----------------------------------------------------
configs.extend([
    dict(
        name='Platypus-30B',
        hf_config={'org': 'garage-bAInd',
        'name': 'Platypus-30B'},
        block_size=2048,
        padded_vocab_size=32000,
        n_layer=60,
        n_head=52,
        n_embd=6656,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-06,
        mlp_class_name='LLaMAMLP',
        intermediate_size=17920,
    ),
    dict(
        name='Platypus2-7B',
        hf_config={'org': 'garage-bAInd',
        'name': 'Platypus2-7B'},
        padded_vocab_size=32000,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
    ),
    dict(
        name='Platypus2-13B',
        hf_config={'org': 'garage-bAInd',
        'name': 'Platypus2-13B'},
        padded_vocab_size=32000,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
    ),
    dict(
        name='Platypus2-70B',
        hf_config={'org': 'garage-bAInd',
        'name': 'Platypus2-70B'},
        padded_vocab_size=32000,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
    ),
])
# -- Camel -------------------------------------------------------
configs.extend([
    dict(
        name='Camel-Platypus2-13B',
        hf_config={'org': 'garage-bAInd',
        'name': 'Camel-Platypus2-13B'},
        padded_vocab_size=32000,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
    ),
    dict(
        name='Camel-Platypus2-70B',
        hf_config={'org': 'garage-bAInd',
        'name': 'Camel-Platypus2-70B'},
        padded_vocab_size=32000,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
    ),
])
# -- Stable- -----------------------------------------------------
configs.extend([
    dict(
        name='Stable-Platypus2-13B',
        hf_config={'org': 'garage-bAInd',
        'name': 'Stable-Platypus2-13B'},
        padded_vocab_size=32000,
        n_layer=40,
        n_head=40,
        n_embd=5120,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=13824,
    ),
])
# -- Platypus ----------------------------------------------------
configs.extend([
    dict(
        name='Platypus2-70B-instruct',
        hf_config={'org': 'garage-bAInd',
        'name': 'Platypus2-70B-instruct'},
        padded_vocab_size=32000,
        n_layer=80,
        n_head=64,
        n_embd=8192,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=28672,
    ),
])
# -- LLaMA -------------------------------------------------------
configs.extend([
    dict(
        name='LLaMA-2-7B-32K',
        hf_config={'org': 'togethercomputer',
        'name': 'LLaMA-2-7B-32K'},
        vocab_size=32000,
        padding_multiple=64,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
        rope_condense_ratio=8,
    ),
])
# -- phi ---------------------------------------------------------
configs.extend([
    dict(
        name='phi-1_5',
        hf_config={'org': 'microsoft',
        'name': 'phi-1_5'},
        vocab_size=50257,
        padded_vocab_size=51200,
        block_size=2048,
        n_embd=2048,
        n_layer=24,
        rotary_percentage=0.5,
        shared_attention_norm=True,
        lm_head_bias=True,
        gelu_approximate='tanh',
    ),
])
# -- Mistral -----------------------------------------------------
configs.extend([
    dict(
        name='Mistral-7B-Instruct-v0.1',
        hf_config={'org': 'mistralai',
        'name': 'Mistral-7B-Instruct-v0.1'},
        padded_vocab_size=32000,
        block_size=4096,
        n_layer=32,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=14336,
    ),
])
# -- Mixtral -----------------------------------------------------
configs.extend([
    dict(
        name='Mixtral-8x7B-Instruct-v0.1',
        hf_config={'org': 'mistralai',
        'name': 'Mixtral-8x7B-Instruct-v0.1'},
        padded_vocab_size=32000,
        block_size=32768,
        n_layer=32,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMoE',
        intermediate_size=14336,
        rope_base=1000000,
        n_expert=8,
        n_expert_per_token=2,
    ),
])
# -- Mistral -----------------------------------------------------
configs.extend([
    dict(
        name='Mistral-7B-v0.2',
        hf_config={'org': 'unsloth',
        'name': 'Mistral-7B-v0.2'},
        padded_vocab_size=32000,
        block_size=32768,
        n_layer=32,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=14336,
    ),
    dict(
        name='Mistral-7B-v0.3',
        hf_config={'org': 'mistralai',
        'name': 'Mistral-7B-v0.3'},
        padded_vocab_size=32768,
        block_size=32768,
        n_layer=32,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=14336,
    ),
    dict(
        name='Mistral-7B-Instruct-v0.3',
        hf_config={'org': 'mistralai',
        'name': 'Mistral-7B-Instruct-v0.3'},
        padded_vocab_size=32768,
        block_size=32768,
        n_layer=32,
        n_query_groups=8,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=14336,
    ),
])
# -- tiny-llama --------------------------------------------------
configs.extend([
    dict(
        name='tiny-llama-1.1b-chat',
        hf_config={'org': 'TinyLlama',
        'name': 'TinyLlama-1.1B-Chat-v1.0'},
        block_size=2048,
        vocab_size=32000,
        padding_multiple=64,
        n_layer=22,
        n_head=32,
        n_embd=2048,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        norm_eps=1e-05,
        mlp_class_name='LLaMAMLP',
        intermediate_size=5632,
        n_query_groups=4,
    ),
])
# -- Llama-2 -----------------------------------------------------
configs.extend([
    dict(
        name='Llama-2-7b-chat-hf-function-calling-v2',
        hf_config={'org': 'Trelis',
        'name': 'Llama-2-7b-chat-hf-function-calling-v2'},
        padding_multiple=64,
        n_layer=32,
        rotary_percentage=1.0,
        parallel_residual=False,
        bias=False,
        norm_class_name='RMSNorm',
        mlp_class_name='LLaMAMLP',
        intermediate_size=11008,
        norm_eps=1e-06,
        block_size=4096,
        vocab_size=32000,
        n_head=32,
        n_embd=4096,
        rope_base=10000,
    ),
])


name_to_config: dict[str, dict] = {c["name"]: c for c in configs}
