"""Decoder-only transformer for mdi_llm_amd — PyTorch execution path.

This module is the framework's *reference* execution path: it defines the
model, the KV-cache pool, and the prefill/decode forward used on CPU, for
training, and as the fp32 numerics oracle the hand-written HIP kernels are
tested against.  The MI355X decode engine (``mdi_llm_amd.ops``) consumes the
same parameters and the same cache pool.

Capability parity with the reference model (/root/reference/src/sub/model.py:
``GPT`` 276, ``Block`` 576, ``CausalSelfAttention`` 632, MLPs 782-853,
``build_rope_cache`` 856, ``apply_rope`` 881, ``KVCache`` 894, ``RMSNorm``
950) — re-designed, not translated:

* KV caches live in a single per-process **pool tensor** indexed by sample
  slot (``KVCachePool``), not one tensor pair per sample swapped into the
  modules per message (reference ``gptserver.py:975-978``).  A pooled cache
  is what lets the HIP decode path replay one hipGraph for every in-flight
  sample and keeps GQA caches at ``n_query_groups`` width (the reference
  stores them expanded to ``n_head`` — SURVEY §5.7).
* Attention is computed with explicit matmul + softmax (library GEMMs) on
  the torch path — no Triton, no flash shims; the fused flash-decode lives
  in the HIP extension.
* State-dict key names follow the litGPT layout so the reference's chunked
  checkpoint format loads unchanged (``transformer.h.<i>.attn.attn.weight``
  interleaved-QKV etc., see /root/reference/src/sub/utils/utils.py:241-438).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn
from torch.nn import functional as F

from ..config import ModelConfig

__all__ = [
    "GPT",
    "Block",
    "CausalSelfAttention",
    "GptNeoxMLP",
    "LLaMAMLP",
    "GemmaMLP",
    "LLaMAMoE",
    "RMSNorm",
    "KVCachePool",
    "build_rope_cache",
    "apply_rope",
]


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------

def build_rope_cache(
    seq_len: int,
    n_elem: int,
    device: Optional[torch.device] = None,
    base: int = 10000,
    condense_ratio: int = 1,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Precomputed cos/sin tables, shape ``(seq_len, n_elem)`` fp32.

    Rotate-half convention (theta duplicated across the two halves), matching
    the reference (/root/reference/src/sub/model.py:856-878).
    """
    theta = 1.0 / (base ** (torch.arange(0, n_elem, 2, device=device).float() / n_elem))
    seq_idx = torch.arange(seq_len, device=device) / condense_ratio
    idx_theta = torch.outer(seq_idx, theta)
    idx_theta = torch.cat([idx_theta, idx_theta], dim=1)
    return torch.cos(idx_theta), torch.sin(idx_theta)


def apply_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """``x``: (..., T, rope_n_elem); cos/sin: (T, rope_n_elem)."""
    head_size = x.size(-1)
    xf = x.float()
    x1 = xf[..., : head_size // 2]
    x2 = xf[..., head_size // 2 :]
    rotated = torch.cat((-x2, x1), dim=-1)
    return (xf * cos.float() + rotated * sin.float()).to(dtype=x.dtype)


# ---------------------------------------------------------------------------
# Norms
# ---------------------------------------------------------------------------


class RMSNorm(nn.Module):
    """fp32-accumulated RMSNorm (reference semantics, model.py:950-980)."""

    def __init__(self, size: int, dim: int = -1, eps: float = 1e-5) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.ones(size))
        self.eps = eps
        self.dim = dim

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        dtype = x.dtype
        x = x.float()
        norm_x = torch.mean(x * x, dim=self.dim, keepdim=True)
        x_normed = x * torch.rsqrt(norm_x + self.eps)
        return (x_normed * self.weight.float()).to(dtype=dtype)

    def reset_parameters(self) -> None:
        nn.init.ones_(self.weight)


# ---------------------------------------------------------------------------
# KV cache pool
# ---------------------------------------------------------------------------


class KVCachePool:
    """One contiguous KV arena per stage, indexed by sample slot.

    Layout ``[n_slots, n_local_layers, n_kv_heads, max_seq, head_size]`` for K
    and V separately: the decode-attention kernel streams K/V rows of one
    (slot, layer, kv-head) contiguously along the sequence axis, and one
    hipGraph can serve every slot because the slot index is data, not a
    pointer.
    """

    def __init__(
        self,
        n_slots: int,
        n_layers: int,
        n_kv_heads: int,
        max_seq: int,
        head_size: int,
        device: torch.device,
        dtype: torch.dtype,
    ) -> None:
        shape = (n_slots, n_layers, n_kv_heads, max_seq, head_size)
        self.k = torch.zeros(shape, device=device, dtype=dtype)
        self.v = torch.zeros(shape, device=device, dtype=dtype)
        # tokens currently stored per slot
        self.seq_len = torch.zeros(n_slots, dtype=torch.int64)
        self.n_slots = n_slots
        self.max_seq = max_seq
        self.fp8 = False
        self.kscale = None
        self.vscale = None

    def to_fp8(self) -> None:
        """Convert the arena to an fp8 (OCP e4m3) cache: uint8 payload +
        one fp32 scale per cached row.  Only the HIP kernels read/write
        this layout; the torch ``append`` path refuses (the engine gates
        fp8-KV on the HIP prefill path)."""
        if self.fp8:
            return
        shape = self.k.shape
        dev = self.k.device
        self.k = torch.zeros(shape, device=dev, dtype=torch.uint8)
        self.v = torch.zeros(shape, device=dev, dtype=torch.uint8)
        self.kscale = torch.zeros(shape[:-1], device=dev,
                                  dtype=torch.float32)
        self.vscale = torch.zeros_like(self.kscale)
        self.fp8 = True

    def reset(self, slot: Optional[int] = None) -> None:
        if slot is None:
            self.seq_len.zero_()
        else:
            self.seq_len[slot] = 0

    def append(
        self, slot: int, layer: int, k: torch.Tensor, v: torch.Tensor, pos: int
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Write k,v of shape (n_kv_heads, T, head_size) at ``pos``; return
        views of the full cache up to ``pos+T``."""
        if self.fp8:
            raise RuntimeError(
                "fp8 KV cache is written only by the HIP kernels "
                "(rope_prefill_append / fused attention append); the torch "
                "path cannot append to it"
            )
        T = k.size(-2)
        self.k[slot, layer, :, pos : pos + T] = k
        self.v[slot, layer, :, pos : pos + T] = v
        return (
            self.k[slot, layer, :, : pos + T],
            self.v[slot, layer, :, : pos + T],
        )

    def memory_bytes(self) -> int:
        return self.k.numel() * self.k.element_size() * 2


# ---------------------------------------------------------------------------
# Attention
# ---------------------------------------------------------------------------


class CausalSelfAttention(nn.Module):
    """GQA/MQA attention with fused interleaved QKV projection.

    Weight layout (litGPT-compatible): ``attn.weight`` rows grouped per query
    group as [q_0..q_{q_per_kv-1}, k, v] × n_query_groups
    (reference model.py:644-646, 686-718).
    """

    def __init__(self, config: ModelConfig, block_idx: int) -> None:
        super().__init__()
        self.config = config
        self.block_idx = block_idx
        shape = (config.n_head + 2 * config.n_query_groups) * config.head_size
        self.attn = nn.Linear(config.n_embd, shape, bias=config.bias)
        self.proj = nn.Linear(
            config.head_size * config.n_head, config.n_embd, bias=config.bias
        )

    def split_qkv(
        self, qkv: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """(B, T, qkv_dim) -> q (B,nh,T,hs), k/v (B,ng,T,hs)."""
        B, T, _ = qkv.shape
        cfg = self.config
        q_per_kv = cfg.q_per_kv
        qkv = qkv.view(B, T, cfg.n_query_groups, q_per_kv + 2, cfg.head_size)
        q, k, v = qkv.split((q_per_kv, 1, 1), dim=-2)
        q = q.reshape(B, T, cfg.n_head, cfg.head_size).transpose(1, 2)
        k = k.reshape(B, T, cfg.n_query_groups, cfg.head_size).transpose(1, 2)
        v = v.reshape(B, T, cfg.n_query_groups, cfg.head_size).transpose(1, 2)
        return q, k, v

    def forward(
        self,
        x: torch.Tensor,
        cos: torch.Tensor,
        sin: torch.Tensor,
        kv_pool: Optional[KVCachePool] = None,
        slot: int = 0,
        local_layer: int = 0,
        input_pos: int = 0,
        mask: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        B, T, _ = x.shape
        cfg = self.config
        q, k, v = self.split_qkv(self.attn(x))

        n_elem = cfg.rope_n_elem
        if n_elem > 0:
            q = torch.cat(
                [apply_rope(q[..., :n_elem], cos, sin), q[..., n_elem:]], dim=-1
            )
            k = torch.cat(
                [apply_rope(k[..., :n_elem], cos, sin), k[..., n_elem:]], dim=-1
            )

        if kv_pool is not None:
            assert B == 1, "cached decode path is per-sample (B=1)"
            k_all, v_all = kv_pool.append(slot, local_layer, k[0], v[0], input_pos)
            k, v = k_all.unsqueeze(0), v_all.unsqueeze(0)

        y = self._attend(q, k, v, mask)
        y = y.transpose(1, 2).reshape(B, T, cfg.head_size * cfg.n_head)
        return self.proj(y)

    def _attend(
        self,
        q: torch.Tensor,
        k: torch.Tensor,
        v: torch.Tensor,
        mask: Optional[torch.Tensor],
    ) -> torch.Tensor:
        cfg = self.config
        if cfg.n_query_groups != cfg.n_head:
            reps = cfg.q_per_kv
            k = k.repeat_interleave(reps, dim=1)
            v = v.repeat_interleave(reps, dim=1)
        scale = 1.0 / math.sqrt(cfg.head_size)
        att = (q.float() @ k.float().transpose(-2, -1)) * scale
        T, S = att.shape[-2], att.shape[-1]
        if mask is not None:
            att = att.masked_fill(~mask, float("-inf"))
        elif T > 1:
            causal = torch.ones(T, S, dtype=torch.bool, device=att.device).tril(
                diagonal=S - T
            )
            att = att.masked_fill(~causal, float("-inf"))
        att = F.softmax(att, dim=-1)
        return (att @ v.float()).to(q.dtype)


# ---------------------------------------------------------------------------
# MLPs (reference model.py:782-853)
# ---------------------------------------------------------------------------


class GptNeoxMLP(nn.Module):
    def __init__(self, config: ModelConfig) -> None:
        super().__init__()
        self.fc = nn.Linear(config.n_embd, config.intermediate_size, bias=config.bias)
        self.proj = nn.Linear(config.intermediate_size, config.n_embd, bias=config.bias)
        self.config = config

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.fc(x)
        x = F.gelu(x, approximate=self.config.gelu_approximate)
        return self.proj(x)


class LLaMAMLP(nn.Module):
    def __init__(self, config: ModelConfig) -> None:
        super().__init__()
        self.fc_1 = nn.Linear(config.n_embd, config.intermediate_size, bias=config.bias)
        self.fc_2 = nn.Linear(config.n_embd, config.intermediate_size, bias=config.bias)
        self.proj = nn.Linear(config.intermediate_size, config.n_embd, bias=config.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.proj(F.silu(self.fc_1(x)) * self.fc_2(x))


class GemmaMLP(LLaMAMLP):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.proj(
            F.gelu(self.fc_1(x), approximate="tanh") * self.fc_2(x)
        )


class LLaMAMoE(nn.Module):
    """Local (unsharded) mixture-of-experts, reference model.py:823-853."""

    def __init__(self, config: ModelConfig) -> None:
        super().__init__()
        self.gate = nn.Linear(config.n_embd, config.n_expert, bias=False)
        self.experts = nn.ModuleList(
            LLaMAMLP(config) for _ in range(config.n_expert)
        )
        self.config = config

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, T, C = x.shape
        x = x.view(-1, C)
        router = self.gate(x)
        probs, indices = torch.topk(router, self.config.n_expert_per_token)
        probs = probs.softmax(dim=1, dtype=torch.float).to(dtype=x.dtype)
        masks = indices.unsqueeze(-1) == torch.arange(
            self.config.n_expert, device=x.device
        )
        masks = masks.permute(2, 0, 1)  # (n_expert, B*T, top_k)
        y = torch.zeros_like(x)
        for mask, expert in zip(masks, self.experts):
            token_idx, expert_idx = torch.where(mask)
            if token_idx.numel() == 0:
                continue
            y[token_idx] += probs[token_idx, expert_idx, None] * expert(
                x[token_idx]
            )
        return y.view(B, T, C)


def mlp_class(config: ModelConfig) -> type:
    return {
        "GptNeoxMLP": GptNeoxMLP,
        "LLaMAMLP": LLaMAMLP,
        "GemmaMLP": GemmaMLP,
        "LLaMAMoE": LLaMAMoE,
    }[config.mlp_class_name]


def norm_class(config: ModelConfig):
    if config.norm_class_name == "RMSNorm":
        return lambda size: RMSNorm(size, eps=config.norm_eps)
    return lambda size: nn.LayerNorm(size, eps=config.norm_eps)


# ---------------------------------------------------------------------------
# Block
# ---------------------------------------------------------------------------


class Block(nn.Module):
    def __init__(self, config: ModelConfig, block_idx: int) -> None:
        super().__init__()
        self.config = config
        self.norm_1 = norm_class(config)(config.n_embd)
        self.attn = CausalSelfAttention(config, block_idx)
        self.norm_2 = (
            None
            if config.shared_attention_norm
            else norm_class(config)(config.n_embd)
        )
        self.mlp = mlp_class(config)(config)

    def forward(
        self,
        x: torch.Tensor,
        cos: torch.Tensor,
        sin: torch.Tensor,
        kv_pool: Optional[KVCachePool] = None,
        slot: int = 0,
        local_layer: int = 0,
        input_pos: int = 0,
        mask: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        n1 = self.norm_1(x)
        h = self.attn(
            n1, cos, sin, kv_pool, slot, local_layer, input_pos, mask
        )
        if self.config.parallel_residual:
            n2 = n1 if self.config.shared_attention_norm else self.norm_2(x)
            return self.mlp(n2) + h + x
        x = h + x
        return self.mlp(self.norm_2(x)) + x


# ---------------------------------------------------------------------------
# Full model
# ---------------------------------------------------------------------------


class GPT(nn.Module):
    """Full decoder model (standalone generation + training)."""

    def __init__(self, config: ModelConfig) -> None:
        super().__init__()
        assert config.padded_vocab_size is not None
        self.config = config
        self.lm_head = nn.Linear(
            config.n_embd, config.padded_vocab_size, bias=config.lm_head_bias
        )
        modules = dict(
            wte=nn.Embedding(config.padded_vocab_size, config.n_embd),
            h=nn.ModuleList(Block(config, i) for i in range(config.n_layer)),
            ln_f=norm_class(config)(config.n_embd),
        )
        if config.pos_embedding == "learned":
            modules["wpe"] = nn.Embedding(config.block_size, config.n_embd)
        self.transformer = nn.ModuleDict(modules)
        self.max_seq_length = config.block_size
        self.kv_pool: Optional[KVCachePool] = None

    # -- rope / seq-length machinery -------------------------------------
    @property
    def max_seq_length(self) -> int:
        return self._max_seq_length

    @max_seq_length.setter
    def max_seq_length(self, value: int) -> None:
        if value > self.config.block_size:
            raise ValueError(
                f"seq length {value} > block_size {self.config.block_size}"
            )
        self._max_seq_length = value
        # rebuild on whatever device the model currently lives on
        dev = self.cos.device if hasattr(self, "cos") else None
        cos, sin = build_rope_cache(
            value,
            self.config.rope_n_elem,
            device=dev,
            base=self.config.rope_base,
            condense_ratio=self.config.rope_condense_ratio,
        )
        # buffers so .to(device) moves them; not persisted
        self.register_buffer("cos", cos, persistent=False)
        self.register_buffer("sin", sin, persistent=False)

    def _apply(self, fn, recurse=True):
        # keep the RoPE tables fp32 whatever dtype the module is cast to
        # (the HIP engine and the torch path must read identical tables)
        ret = super()._apply(fn, recurse)
        if hasattr(self, "cos") and self.cos.dtype != torch.float32:
            cos, sin = build_rope_cache(
                self._max_seq_length,
                self.config.rope_n_elem,
                device=self.cos.device,
                base=self.config.rope_base,
                condense_ratio=self.config.rope_condense_ratio,
            )
            self.cos, self.sin = cos, sin
        return ret

    def set_kv_cache(
        self,
        batch_size: int = 1,
        device: Optional[torch.device] = None,
        dtype: Optional[torch.dtype] = None,
    ) -> None:
        """Allocate the pooled KV cache with ``batch_size`` sample slots."""
        if device is None:
            device = next(self.parameters()).device
        if dtype is None:
            dtype = next(self.parameters()).dtype
        self.kv_pool = KVCachePool(
            batch_size,
            self.config.n_layer,
            self.config.n_query_groups,
            self.max_seq_length,
            self.config.head_size,
            device,
            dtype,
        )

    def clear_kv_cache(self) -> None:
        self.kv_pool = None

    # -- forward ----------------------------------------------------------
    def forward(
        self,
        idx: torch.Tensor,
        input_pos: Optional[int] = None,
        slot: int = 0,
    ) -> torch.Tensor:
        """``idx``: (B, T) token ids.

        Training / no-cache: ``input_pos=None`` — full causal attention.
        Cached decode/prefill: ``input_pos`` is the int position of idx[0]
        in the sequence; KV pool slot ``slot`` is updated.
        """
        B, T = idx.shape
        x = self.transformer.wte(idx)
        if self.config.scale_embeddings:
            x = x * (self.config.n_embd ** 0.5)
        if self.config.pos_embedding == "learned":
            pos0 = input_pos or 0
            positions = torch.arange(pos0, pos0 + T, device=idx.device)
            x = x + self.transformer.wpe(positions)

        if input_pos is not None:
            cos = self.cos[input_pos : input_pos + T]
            sin = self.sin[input_pos : input_pos + T]
            assert self.kv_pool is not None, "call set_kv_cache() first"
            for i, block in enumerate(self.transformer.h):
                x = block(
                    x, cos, sin, self.kv_pool, slot, i, input_pos, None
                )
        else:
            cos = self.cos[:T]
            sin = self.sin[:T]
            for block in self.transformer.h:
                x = block(x, cos, sin)
        x = self.transformer.ln_f(x)
        return self.lm_head(x)

    # -- generation (standalone path, reference model.py:461-525) ---------
    @torch.inference_mode()
    def generate(
        self,
        prompt: torch.Tensor,
        max_new_tokens: int,
        *,
        temperature: float = 0.8,
        top_k: Optional[int] = 200,
        top_p: float = 1.0,
        slot: int = 0,
        stop_tokens: tuple = (),
        generator: Optional[torch.Generator] = None,
        token_callback=None,
    ) -> torch.Tensor:
        """Autoregressive generation with the pooled KV cache; returns the
        full sequence (prompt + generated)."""
        from .sampling import sample

        device = prompt.device
        T = prompt.size(0)
        assert self.kv_pool is not None, "call set_kv_cache() first"
        self.kv_pool.reset(slot)
        tokens = [prompt]
        logits = self.forward(prompt.view(1, -1), input_pos=0, slot=slot)
        input_pos = T
        generated: list = []
        for _ in range(max_new_tokens):
            nxt = sample(
                logits[0, -1],
                temperature=temperature,
                top_k=top_k,
                top_p=top_p,
                generator=generator,
            ).to(device)
            generated.append(nxt)
            if token_callback is not None:
                token_callback(int(nxt))
            if stop_tokens and _ends_with_stop(generated, stop_tokens):
                break
            if input_pos >= self.max_seq_length:
                break
            logits = self.forward(
                nxt.view(1, 1), input_pos=input_pos, slot=slot
            )
            input_pos += 1
        if generated:
            tokens.append(torch.stack(generated))
        return torch.cat(tokens)

    # -- training utilities -----------------------------------------------
    def estimate_mfu(self, fwdbwd_per_iter: float, dt: float, peak_flops: float) -> float:
        """Model-FLOPs-utilisation, reference model.py:348-368."""
        cfg = self.config
        L, H, Q, T = cfg.n_layer, cfg.n_head, cfg.head_size, self.max_seq_length
        N = sum(p.numel() for p in self.parameters())
        flops_per_token = 6 * N + 12 * L * H * Q * T
        flops_per_iter = flops_per_token * T * fwdbwd_per_iter
        return flops_per_iter * (1.0 / dt) / peak_flops

    def _init_weights(self, module: nn.Module) -> None:
        if isinstance(module, nn.Linear):
            torch.nn.init.normal_(module.weight, mean=0.0, std=0.02)
            if module.bias is not None:
                torch.nn.init.zeros_(module.bias)
        elif isinstance(module, nn.Embedding):
            torch.nn.init.normal_(module.weight, mean=0.0, std=0.02)

    def apply_init(self) -> None:
        self.apply(self._init_weights)


def _ends_with_stop(generated: list, stop_tokens: tuple) -> bool:
    for seq in stop_tokens:
        n = len(seq)
        if n and len(generated) >= n:
            if all(int(generated[-n + i]) == seq[i] for i in range(n)):
                return True
    return False
