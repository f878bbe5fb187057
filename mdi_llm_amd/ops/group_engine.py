"""Grouped (batched) decode engine: B in-flight samples advance one token
per stage pass.

Beyond-parity optimization over the reference's strict one-sample-per-
message schedule: with B samples batched, each stage streams
its weights from HBM once per B tokens instead of once per token — decode
is bandwidth-bound, so stage throughput scales nearly linearly with B until
compute/latency limits.  The recurrent-pipeline principle is preserved at
group granularity: n_stages groups of B samples keep every stage busy.

Implementation: the skinny [B,K]x[K,M] projections go through hipBLASLt
(``F.linear`` — a plain library GEMM), while attention (batched split-S
flash-decode with fused RoPE + KV-append, per-sample positions), sampling
(radix top-k + gumbel) and embedding use the hand-written CDNA4 kernels.
The whole group step is captured as one hipGraph per role.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from ..config import ModelConfig
from ..models.model import KVCachePool
from . import require_hip_ops
from .engine import _BlockWeights, engine_supported

__all__ = ["GroupDecodeEngine", "group_engine_supported"]


def _maybe_enable_tunableop(B: int) -> None:
    """Load the committed MI355X hipBLASLt/rocBLAS algorithm selections for
    the grouped-decode GEMM shapes (PyTorch TunableOp, tuning disabled).

    Measured on MI355X: +14% rotation throughput at B=128 (16.4k -> 18.7k
    tok/s), but slightly negative at B=64 — so it is enabled only for
    B >= 96 by default.  MDI_TUNABLEOP=1/0 forces it on/off."""
    import os
    from pathlib import Path

    env = os.environ.get("MDI_TUNABLEOP", "")
    if env == "0" or (env != "1" and B < 96):
        return
    try:
        import torch.cuda.tunable as tunable

        fn = Path(__file__).resolve().parents[1] / "data" / \
            "tunableop_mi355x.csv"
        tunable.enable(True)
        tunable.tuning_enable(False)
        if fn.is_file():
            tunable.read_file(str(fn))
    except Exception as e:  # noqa: BLE001 - best-effort acceleration
        import warnings

        warnings.warn(f"TunableOp setup failed ({e}); using default GEMMs")


def group_engine_supported(config: ModelConfig) -> bool:
    return (
        engine_supported(config)
        and config.norm_class_name == "RMSNorm"
        and config.mlp_class_name in ("LLaMAMLP", "GemmaMLP")
    )


class GroupDecodeEngine:
    def __init__(
        self,
        stage,
        kv_pool: KVCachePool,
        group_size: int,
        n_chunks: int = 16,
        use_graphs: bool = True,
    ) -> None:
        import os

        self.ops = require_hip_ops()
        cfg: ModelConfig = stage.config
        _maybe_enable_tunableop(group_size)
        if not group_engine_supported(cfg):
            raise ValueError(f"{cfg.name!r} unsupported by GroupDecodeEngine")
        # fp8 grouped GEMMs measured SLOWER than bf16 hipBLASLt on ROCm 7
        # (skinny-M _scaled_mm + dynamic-quant overhead): keep groups bf16.
        # fp8 weights remain available for B=1 decode (DecodeEngine).
        if os.environ.get("MDI_WEIGHT_DTYPE", "bf16") == "fp8":
            import warnings

            warnings.warn("grouped decode ignores MDI_WEIGHT_DTYPE=fp8 "
                          "(bf16 is faster for batched GEMMs; see ROADMAP)")
        self.fp8 = False
        self.config = cfg
        self.stage = stage
        self.kv_pool = kv_pool
        self.B = group_size
        self.n_chunks = n_chunks
        self.device = next(stage.parameters()).device
        self.is_starter = hasattr(stage, "lm_head")
        self.use_graphs = use_graphs and self.device.type == "cuda"

        dev = self.device
        B = self.B
        self.blocks = [_BlockWeights(b, cfg) for b in stage.transformer.h]
        if self.fp8:
            from .engine import quantize_fp8_rowwise

            for w in self.blocks:
                w.quantize_fp8(cfg)
        if self.is_starter:
            self.wte = stage.transformer.wte.weight.detach().contiguous()
            self.lnf_w = stage.transformer.ln_f.weight.detach().contiguous()
            self.head_w = stage.lm_head.weight.detach().contiguous()
            if self.fp8:
                from .engine import quantize_fp8_rowwise

                self.head_w8, self.head_s = quantize_fp8_rowwise(self.head_w)
        self.cos = stage.cos.detach().to(torch.float32).contiguous()
        self.sin = stage.sin.detach().to(torch.float32).contiguous()

        bf = dict(device=dev, dtype=torch.bfloat16)
        E, I = cfg.n_embd, cfg.intermediate_size
        n_head, hs = cfg.n_head, cfg.head_size
        self.X = torch.zeros(B, E, **bf)
        self.XN = torch.zeros(B, E, **bf)
        self.HN = torch.zeros(B, E, **bf)
        self.QKV = torch.zeros(B, cfg.qkv_dim, **bf)
        self.Y = torch.zeros(B, n_head * hs, **bf)
        # hand-written M-tile MFMA GEMM (ops/hip mtile_gemm_kernel).
        # Measured per-shape vs hipBLASLt (profiles/mtile_gemm_r02.md):
        # standalone it WINS the skinny-M proj shape (1.49x at B=32) and
        # loses the wide-M shapes (~0.3-0.7x; the library streams those
        # at ~5 TB/s), and IN-ROTATION even the proj win evaporates
        # (addmm fuses the residual and picks a better algorithm:
        # B=32 +0.2% noise, B=64 -2%).  Default OFF — a documented
        # measured refutation; MDI_MTILE=proj|all opts in.
        mt_env = os.environ.get("MDI_MTILE", "0")
        shapes_ok = (B in (16, 32, 64, 128)
                     and all(self._mtile_k_ok(B, k)
                             for k in (E, n_head * hs, I)))
        self.use_mtile = mt_env == "all" and shapes_ok
        self._mtile_proj = (
            mt_env in ("all", "proj", "1") and B <= 64
            and B in (16, 32, 64)
            and self._mtile_k_ok(B, n_head * hs)
            and E % 16 == 0
        )
        if self.use_mtile or self._mtile_proj:
            self.A2 = torch.zeros(B, E, **bf)
        if self.use_mtile:
            self.G = torch.zeros(B, I, **bf)
            self.U2 = torch.zeros(B, I, **bf)
        self.part_o = torch.zeros(B * n_head * n_chunks * hs, device=dev,
                                  dtype=torch.float32)
        self.part_ml = torch.zeros(B * n_head * n_chunks * 2, device=dev,
                                   dtype=torch.float32)
        if self.is_starter:
            self.LOGITS = torch.zeros(B, cfg.padded_vocab_size, **bf)
            self.sample_scratch = torch.zeros(520 * B, device=dev,
                                              dtype=torch.int32)
            self.tokens = torch.zeros(B, device=dev, dtype=torch.int32)
            self.token_table = torch.zeros(kv_pool.n_slots, device=dev,
                                           dtype=torch.int32)
        # per-call slot set + derived positions
        self.slots = torch.zeros(B, device=dev, dtype=torch.int32)
        self.slots_long = torch.zeros(B, device=dev, dtype=torch.int64)
        self.pos = torch.zeros(B, device=dev, dtype=torch.int32)
        self.pos_table = torch.zeros(kv_pool.n_slots, device=dev,
                                     dtype=torch.int32)
        self._ones = torch.ones(B, device=dev, dtype=torch.int32)

        self._g_standalone: Optional[torch.cuda.CUDAGraph] = None
        self._g_starter: Optional[torch.cuda.CUDAGraph] = None
        self._g_tail: Optional[torch.cuda.CUDAGraph] = None
        self._g_mid: Optional[torch.cuda.CUDAGraph] = None
        self._fused_params = None

    @staticmethod
    def _mtile_k_ok(B: int, K: int) -> bool:
        kc = 512 if B <= 32 else (256 if B <= 64 else 128)
        return K % kc == 0 and K >= 2 * kc

    # ------------------------------------------------------------------
    def set_slot_pos(self, slot: int, pos: int) -> None:
        self.pos_table[slot] = pos

    def _stage_idx(self) -> None:
        self.slots_long.copy_(self.slots)
        torch.index_select(self.pos_table, 0, self.slots_long, out=self.pos)

    def _rms(self, X: torch.Tensor, w: torch.Tensor,
             out: torch.Tensor) -> torch.Tensor:
        # one batched HIP kernel (replaces a ~6-kernel torch composition)
        self.ops.rmsnorm(out, X, w, self.config.norm_eps)
        return out

    def _mm(self, x: torch.Tensor, w8: torch.Tensor, ws: torch.Tensor,
            wbf: torch.Tensor):
        """fp8 GEMM via hipBLASLt _scaled_mm with dynamic per-token
        activation scales; falls back to bf16 F.linear."""
        if not self.fp8:
            return F.linear(x, wbf)
        ax = (x.float().abs().amax(dim=1, keepdim=True)
              .clamp(min=1e-12) / 448.0)
        xq = (x.float() / ax).to(torch.float8_e4m3fn)
        return torch._scaled_mm(xq, w8.t(), scale_a=ax,
                                scale_b=ws[None, :],
                                out_dtype=torch.bfloat16)

    def _run_blocks_mtile(self) -> None:
        """Block stack on the hand-written M-tile MFMA GEMMs (bias and
        residual adds fused into the GEMM epilogue)."""
        cfg = self.config
        scale = 1.0 / (cfg.head_size ** 0.5)
        gelu_gate = cfg.mlp_class_name == "GemmaMLP"
        ops = self.ops
        for li, w in enumerate(self.blocks):
            xn = self._rms(self.X, w.norm1_w, self.XN)
            ops.mtile_gemm(self.QKV, w.attn_w, xn, w.attn_b, None)
            ops.attn_decode(
                self.Y, self.part_o, self.part_ml, self.QKV, self.kv_pool.k,
                self.kv_pool.v, self.cos, self.sin, self.pos, self.slots,
                li, self.n_chunks, scale, self.B,
                kscale=self.kv_pool.kscale, vscale=self.kv_pool.vscale,
            )
            ops.mtile_gemm(self.A2, w.proj_w, self.Y, w.proj_b, self.X)
            hn = self._rms(self.A2, w.norm2_w, self.HN)
            ops.mtile_gemm(self.G, w.fc1_w, hn, None, None)
            ops.mtile_gemm(self.U2, w.fc2_w, hn, None, None)
            ops.swiglu_mul(self.U2, self.G, self.U2, gelu_gate)
            ops.mtile_gemm(self.X, w.mlp_proj_w, self.U2, w.mlp_proj_b,
                           self.A2)

    def _run_blocks(self) -> None:
        if self.use_mtile:
            self._run_blocks_mtile()
            return
        cfg = self.config
        scale = 1.0 / (cfg.head_size ** 0.5)
        gelu_gate = cfg.mlp_class_name == "GemmaMLP"
        X = self.X
        fp8 = self.fp8
        for li, w in enumerate(self.blocks):
            xn = self._rms(X, w.norm1_w, self.XN)
            if fp8:
                qkv = self._mm(xn, w.attn_w8, w.attn_s, w.attn_w)
                if w.attn_b is not None:
                    qkv = qkv + w.attn_b
                qkv = qkv.contiguous()
            else:
                qkv = F.linear(xn, w.attn_w, w.attn_b)
            # contiguous capture-pool temp: stable address under graph replay
            self.ops.attn_decode(
                self.Y, self.part_o, self.part_ml, qkv, self.kv_pool.k,
                self.kv_pool.v, self.cos, self.sin, self.pos, self.slots, li,
                self.n_chunks, scale, self.B,
                kscale=self.kv_pool.kscale, vscale=self.kv_pool.vscale,
            )
            # residual adds fused into the GEMMs (addmm beta=1) on bf16;
            # the proj shape goes to the hand-written M-tile MFMA kernel
            # where it measures faster than the library (see __init__)
            if self._mtile_proj and not fp8:
                self.ops.mtile_gemm(self.A2, w.proj_w, self.Y, w.proj_b, X)
                a = self.A2
            elif fp8:
                a = X + self._mm(self.Y, w.proj_w8, w.proj_s, w.proj_w)
            elif w.proj_b is None:
                a = torch.addmm(X, self.Y, w.proj_w.t())
            else:
                a = X + F.linear(self.Y, w.proj_w, w.proj_b)
            hn = self._rms(a, w.norm2_w, self.HN)
            if fp8:
                gate = self._mm(hn, w.fc1_w8, w.fc1_s, w.fc1_w)
                up = self._mm(hn, w.fc2_w8, w.fc2_s, w.fc2_w)
            else:
                gate = F.linear(hn, w.fc1_w)
                up = F.linear(hn, w.fc2_w)
            # one fused HIP launch instead of the silu/gelu + mul pair
            self.ops.swiglu_mul(up, gate, up, gelu_gate)
            act = up
            if fp8:
                X = a + self._mm(act, w.mlp_proj_w8, w.mlp_proj_s,
                                 w.mlp_proj_w)
            elif w.mlp_proj_b is None:
                X = torch.addmm(a, act, w.mlp_proj_w.t())
            else:
                X = a + F.linear(act, w.mlp_proj_w, w.mlp_proj_b)
        self.X.copy_(X)

    def _tail_seq(self) -> None:
        xn = self._rms(self.X, self.lnf_w, self.XN)
        if self.use_mtile and self.head_w.size(0) % 16 == 0:
            self.ops.mtile_gemm(self.LOGITS, self.head_w, xn, None, None)
        elif self.fp8:
            self.LOGITS.copy_(
                self._mm(xn, self.head_w8, self.head_s, self.head_w))
        else:
            torch.matmul(xn, self.head_w.t(), out=self.LOGITS)

    def _sample_seq(self, temperature, top_k, seed, top_p=1.0) -> None:
        self.ops.sample(
            self.tokens, self.LOGITS, self.sample_scratch,
            float(temperature), int(top_k or 0), temperature > 0.0,
            int(seed) & 0x7FFFFFFF, self.pos, self.slots, self.B,
            float(top_p),
        )

    def _embed_seq(self) -> None:
        torch.index_select(self.token_table, 0, self.slots_long,
                           out=self.tokens)
        emb = self.wte[self.tokens.long()]
        if self.config.scale_embeddings:
            emb = emb * (self.config.n_embd ** 0.5)
        self.X.copy_(emb)

    def _advance(self) -> None:
        self.pos_table.index_add_(0, self.slots_long, self._ones)

    def _write_tokens(self) -> None:
        self.token_table.index_copy_(0, self.slots_long, self.tokens)

    # ------------------------------------------------------------------
    # per-lane private workspaces (everything per-group-step; the
    # pos/token tables and KV pool are per-slot and stay shared)
    _LANE_ATTRS = ("X", "XN", "HN", "QKV", "Y", "part_o", "part_ml",
                   "LOGITS", "sample_scratch", "tokens", "slots",
                   "slots_long", "pos")

    def _set_lane(self, lane: int) -> None:
        for a, t_ in self._lane_bufs[lane].items():
            setattr(self, a, t_)

    @property
    def n_lanes(self) -> int:
        return len(getattr(self, "_lane_graphs", None) or [1])

    def ensure_graphs(self, temperature, top_k, seed,
                      n_lanes: int = 1) -> None:
        """n_lanes > 1 (standalone only): capture the group step over N
        private workspace sets replayed on N HIP streams, so consecutive
        GROUPS overlap — one group's GEMM weight streams fill the other
        group's attention/sampler phases.  Tokens are unchanged (groups
        own disjoint slots; sampling is (seed, slot, pos)-keyed)."""
        params = (float(temperature), int(top_k or 0), int(seed),
                  int(n_lanes))
        if not self.use_graphs:
            return
        if self._fused_params == params:
            return
        t, k, sd, n_lanes = params
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._stage_idx()
                if self.is_starter:
                    self._embed_seq()
                self._run_blocks()
                if self.is_starter:
                    self._tail_seq()
                    self._sample_seq(t, k, sd)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self.kv_pool.k.zero_()
        self.kv_pool.v.zero_()
        self.pos_table.zero_()
        if self.is_starter:
            self.token_table.zero_()

        if self.is_starter:
            base = {a: getattr(self, a) for a in self._LANE_ATTRS
                    if getattr(self, a, None) is not None}
            self._lane_bufs = [dict(base)]
            for _l in range(1, n_lanes):
                self._lane_bufs.append(
                    {a: torch.zeros_like(t_) for a, t_ in base.items()})
            self._lane_graphs = []
            for l in range(n_lanes):
                self._set_lane(l)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._stage_idx()
                    self._embed_seq()
                    self._run_blocks()
                    self._tail_seq()
                    self._sample_seq(t, k, sd)
                    self._write_tokens()
                    self._advance()
                self._lane_graphs.append(g)
            self._set_lane(0)
            self._lane_streams = [torch.cuda.Stream()
                                  for _ in range(n_lanes)]
            self._g_standalone = self._lane_graphs[0]
            g2 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g2):
                self._stage_idx()
                self._tail_seq()
                self._sample_seq(t, k, sd)
                self._write_tokens()
                self._embed_seq()
                self._run_blocks()
                self._advance()
            self._g_starter = g2
            g3 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g3):
                self._stage_idx()
                self._tail_seq()
                self._sample_seq(t, k, sd)
                self._write_tokens()
            self._g_tail = g3
        else:
            g4 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g4):
                self._stage_idx()
                self._run_blocks()
                self._advance()
            self._g_mid = g4
        self._fused_params = params
        torch.cuda.synchronize()

    # ------------------------------------------------------------------
    # step API (slots must be set before each call via set_group)
    # ------------------------------------------------------------------
    def set_group(self, slot_tensor: torch.Tensor) -> None:
        self.slots.copy_(slot_tensor, non_blocking=True)

    def standalone_step(self) -> None:
        self._g_standalone.replay()

    def lanes_begin(self) -> None:
        cur = torch.cuda.current_stream()
        for st in self._lane_streams:
            st.wait_stream(cur)

    def lanes_join(self) -> None:
        cur = torch.cuda.current_stream()
        for st in self._lane_streams:
            cur.wait_stream(st)

    def standalone_lane_step(self, lane: int,
                             slot_tensor: torch.Tensor) -> None:
        """set_group + standalone_step on lane `lane`'s stream."""
        st = self._lane_streams[lane]
        with torch.cuda.stream(st):
            self._lane_bufs[lane]["slots"].copy_(slot_tensor,
                                                 non_blocking=True)
            self._lane_graphs[lane].replay()

    def starter_step(self, X_in: torch.Tensor) -> torch.Tensor:
        if X_in.data_ptr() != self.X.data_ptr():
            self.X.copy_(X_in.view(self.B, -1), non_blocking=True)
        self._g_starter.replay()
        return self.X

    def tail_step(self, X_in: torch.Tensor) -> None:
        if X_in.data_ptr() != self.X.data_ptr():
            self.X.copy_(X_in.view(self.B, -1), non_blocking=True)
        self._g_tail.replay()

    def head_step(self) -> torch.Tensor:
        """embed current tokens + blocks (group seeding, eager)."""
        self._stage_idx()
        self._embed_seq()
        self._run_blocks()
        self._advance()
        return self.X

    def mid_step(self, X_in: torch.Tensor) -> torch.Tensor:
        if X_in.data_ptr() != self.X.data_ptr():
            self.X.copy_(X_in.view(self.B, -1), non_blocking=True)
        if self._g_mid is not None:
            self._g_mid.replay()
        else:
            self._stage_idx()
            self._run_blocks()
            self._advance()
        return self.X
