"""Stage runners: a uniform decode/prefill interface over either the
PyTorch stage modules (CPU, fallback) or the HIP DecodeEngine (MI355X).

The runtime talks to a StageRunner and never cares which backend computes.
Prefill (T>1) always goes through the torch stage forward (writes the same
pooled KV cache the HIP kernels read); decode (T=1) goes through the
hand-written kernels when available.
"""

from __future__ import annotations

import torch

from ..config import ModelConfig
from ..models.stages import StarterStage

__all__ = ["TorchRunner", "HipRunner", "make_runner"]


class TorchRunner:
    """Pure-PyTorch execution of a stage (CPU hosts, unsupported models)."""

    backend = "torch"

    def __init__(self, stage, n_slots: int) -> None:
        self.stage = stage
        self.config: ModelConfig = stage.config
        self.is_starter = isinstance(stage, StarterStage)
        if stage.kv_pool is None:
            stage.set_kv_cache(n_slots)
        self.pos = [0] * n_slots

    def reset(self) -> None:
        self.stage.kv_pool.reset()
        self.pos = [0] * len(self.pos)

    @torch.inference_mode()
    def prefill_head(self, tokens: torch.Tensor, slot: int) -> torch.Tensor:
        x = self.stage.forward_head(tokens.view(1, -1), slot=slot, input_pos=0)
        self.pos[slot] = tokens.numel()
        return x[0]  # (T, n_embd)

    @torch.inference_mode()
    def prefill_mid(self, x: torch.Tensor, slot: int) -> torch.Tensor:
        out = self.stage(x.view(1, -1, self.config.n_embd), slot=slot,
                         input_pos=0)
        self.pos[slot] = x.view(-1, self.config.n_embd).size(0)
        return out[0]

    @torch.inference_mode()
    def decode_head(self, token: torch.Tensor, slot: int) -> torch.Tensor:
        p = self.pos[slot]
        x = self.stage.forward_head(token.long().view(1, 1), slot=slot,
                                    input_pos=p)
        self.pos[slot] = p + 1
        return x.view(-1)  # (n_embd,)

    @torch.inference_mode()
    def decode_mid(self, x: torch.Tensor, slot: int) -> torch.Tensor:
        p = self.pos[slot]
        out = self.stage(x.view(1, 1, -1), slot=slot, input_pos=p)
        self.pos[slot] = p + 1
        return out.view(-1)

    @torch.inference_mode()
    def tail(self, x_last: torch.Tensor) -> torch.Tensor:
        logits = self.stage.forward_tail(x_last.view(1, 1, -1))
        return logits.view(-1)


class HipRunner(TorchRunner):
    """HIP DecodeEngine for T=1 steps; torch stage forward for prefill."""

    backend = "hip"

    def __init__(self, stage, n_slots: int, n_chunks: int = 16,
                 use_graphs: bool = True,
                 expected_s: "int | None" = None) -> None:
        super().__init__(stage, n_slots)
        from ..ops.engine import DecodeEngine

        self.engine = DecodeEngine(stage, stage.kv_pool, n_chunks=n_chunks,
                                   use_graphs=use_graphs,
                                   expected_s=expected_s)
        if use_graphs:
            self.engine.capture_graphs()
            # graph warm-up scribbled on the cache pool; start clean
            self.reset()

    def reset(self) -> None:
        super().reset()
        self.engine.pos_table.zero_()
        self.engine.clear_scratch()

    @torch.inference_mode()
    def prefill_head(self, tokens: torch.Tensor, slot: int) -> torch.Tensor:
        if self.engine.supports_hip_prefill:
            out = self.engine.prefill_prompt(tokens, slot, 0)
            self.pos[slot] = tokens.numel()
        else:
            out = super().prefill_head(tokens, slot)
        self.engine.set_slot_pos(slot, self.pos[slot])
        return out

    @torch.inference_mode()
    def prefill_mid(self, x: torch.Tensor, slot: int) -> torch.Tensor:
        if self.engine.supports_hip_prefill:
            out = self.engine.prefill_hidden(
                x.view(-1, self.config.n_embd).to(torch.bfloat16), slot, 0
            )
            self.pos[slot] = x.view(-1, self.config.n_embd).size(0)
        else:
            out = super().prefill_mid(x, slot)
        self.engine.set_slot_pos(slot, self.pos[slot])
        return out

    @torch.inference_mode()
    def decode_head(self, token: torch.Tensor, slot: int) -> torch.Tensor:
        self.pos[slot] += 1
        return self.engine.decode_step_head(token, slot)

    @torch.inference_mode()
    def decode_mid(self, x: torch.Tensor, slot: int) -> torch.Tensor:
        self.pos[slot] += 1
        return self.engine.decode_step_mid(x, slot)

    @torch.inference_mode()
    def tail(self, x_last: torch.Tensor) -> torch.Tensor:
        return self.engine.tail(x_last)


def make_runner(stage, n_slots: int, device: torch.device,
                n_chunks: int = 16, use_graphs: bool = True,
                force_torch: bool = False,
                expected_s: "int | None" = None):
    """Pick the HIP runner on GPU when the config is supported; fail loudly
    if a GPU is present but the extension is missing (no silent eager
    fallback for supported configs)."""
    if force_torch or device.type != "cuda":
        return TorchRunner(stage, n_slots)
    from ..ops.engine import engine_supported

    if engine_supported(stage.config):
        return HipRunner(stage, n_slots, n_chunks=n_chunks,
                         use_graphs=use_graphs, expected_s=expected_s)
    import sys

    print(
        f"[mdi_llm_amd] WARNING: config {stage.config.name!r} has no HIP "
        "decode-engine instantiation (unsupported attention geometry or "
        "MoE shape) — running the much slower torch fallback on this GPU",
        file=sys.stderr, flush=True,
    )
    return TorchRunner(stage, n_slots)
