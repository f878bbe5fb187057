"""Pipeline-stage modules: StarterStage (head+tail of the net) and
SecondaryStage (middle blocks).

Capability parity with the reference submodels
(/root/reference/src/sub/submodels.py: ``NodePrototype`` 34, ``StarterNode``
132 with its dual-phase ``forward(first_pass=…)`` 170-220, ``SecondaryNode``
223-282) — re-designed for the MI355X runtime:

* the starter's two roles are explicit methods ``forward_head`` /
  ``forward_tail`` instead of a boolean flag, so the scheduler can overlap
  them on different HIP streams;
* KV caches are a pooled arena (see ``model.KVCachePool``) indexed by
  sample slot, never swapped module attributes;
* parameter names match the reference chunk files, so
  ``load_state_dict(torch.load("model_starter.pth"))`` works unchanged.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from ..config import ModelConfig
from .model import Block, KVCachePool, build_rope_cache, norm_class

__all__ = ["StageBase", "StarterStage", "SecondaryStage", "build_stage"]


class StageBase(nn.Module):
    """Shared RoPE / KV-pool machinery for pipeline stages."""

    def __init__(self, config: ModelConfig, n_local_layers: int) -> None:
        super().__init__()
        self.config = config
        self.n_local_layers = n_local_layers
        self.kv_pool: Optional[KVCachePool] = None
        self._max_seq_length = config.block_size
        self._build_rope()

    # -- rope -------------------------------------------------------------
    def _build_rope(self) -> None:
        cos, sin = build_rope_cache(
            self._max_seq_length,
            self.config.rope_n_elem,
            base=self.config.rope_base,
            condense_ratio=self.config.rope_condense_ratio,
        )
        self.register_buffer("cos", cos, persistent=False)
        self.register_buffer("sin", sin, persistent=False)

    def _apply(self, fn, recurse=True):
        # rope tables stay fp32 across .to(dtype=...) — see model.GPT._apply
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
        self._build_rope()
        # buffers were re-registered on CPU; move next to the params
        dev = next(self.parameters()).device
        self.cos = self.cos.to(dev)
        self.sin = self.sin.to(dev)

    # -- kv ---------------------------------------------------------------
    def set_kv_cache(
        self,
        n_slots: int,
        device: Optional[torch.device] = None,
        dtype: Optional[torch.dtype] = None,
    ) -> None:
        if device is None:
            device = next(self.parameters()).device
        if dtype is None:
            dtype = next(self.parameters()).dtype
        self.kv_pool = KVCachePool(
            n_slots,
            self.n_local_layers,
            self.config.n_query_groups,
            self._max_seq_length,
            self.config.head_size,
            device,
            dtype,
        )

    def clear_kv_cache(self) -> None:
        self.kv_pool = None

    def _run_blocks(
        self, blocks, x: torch.Tensor, slot: int, input_pos: int
    ) -> torch.Tensor:
        T = x.size(1)
        cos = self.cos[input_pos : input_pos + T]
        sin = self.sin[input_pos : input_pos + T]
        assert self.kv_pool is not None, "call set_kv_cache() first"
        for i, block in enumerate(blocks):
            x = block(x, cos, sin, self.kv_pool, slot, i, input_pos, None)
        return x


class StarterStage(StageBase):
    """wte + first k blocks + ln_f + lm_head (chunk ``model_starter.pth``)."""

    def __init__(self, config: ModelConfig, n_local_layers: int) -> None:
        super().__init__(config, n_local_layers)
        modules = dict(
            wte=nn.Embedding(config.padded_vocab_size, config.n_embd),
            h=nn.ModuleList(Block(config, i) for i in range(n_local_layers)),
            ln_f=norm_class(config)(config.n_embd),
        )
        if config.pos_embedding == "learned":
            modules["wpe"] = nn.Embedding(config.block_size, config.n_embd)
        self.transformer = nn.ModuleDict(modules)
        self.lm_head = nn.Linear(
            config.n_embd, config.padded_vocab_size, bias=config.lm_head_bias
        )

    def forward_head(
        self, idx: torch.Tensor, slot: int, input_pos: int
    ) -> torch.Tensor:
        """Embed tokens and run the local blocks; returns activations
        (1, T, n_embd) to be sent down the ring."""
        if idx.dim() == 1:
            idx = idx.view(1, -1)
        x = self.transformer.wte(idx)
        if self.config.scale_embeddings:
            x = x * (self.config.n_embd ** 0.5)
        if self.config.pos_embedding == "learned":
            T = idx.size(1)
            positions = torch.arange(
                input_pos, input_pos + T, device=idx.device
            )
            x = x + self.transformer.wpe(positions)
        return self._run_blocks(self.transformer.h, x, slot, input_pos)

    def forward_tail(self, x: torch.Tensor) -> torch.Tensor:
        """Final norm + lm head on activations arriving from the ring tail
        (the reference's ``first_pass=False`` call, submodels.py:170-220).
        Only the last token's logits are produced."""
        x = self.transformer.ln_f(x[:, -1:, :])
        return self.lm_head(x)


class SecondaryStage(StageBase):
    """A run of middle blocks (chunk ``model_secondary<i>.pth``)."""

    def __init__(self, config: ModelConfig, n_local_layers: int) -> None:
        super().__init__(config, n_local_layers)
        self.transformer = nn.ModuleDict(
            dict(
                h=nn.ModuleList(
                    Block(config, i) for i in range(n_local_layers)
                )
            )
        )

    def forward(
        self, x: torch.Tensor, slot: int, input_pos: int
    ) -> torch.Tensor:
        return self._run_blocks(self.transformer.h, x, slot, input_pos)


def build_stage(
    config: ModelConfig, stage: int, n_local_layers: int
) -> StageBase:
    if stage == 0:
        return StarterStage(config, n_local_layers)
    return SecondaryStage(config, n_local_layers)
