"""MDIRuntime — the distributed-inference orchestrator.

Capability parity with the reference's ``GPTDistributed``
(/root/reference/src/sub/model_dist.py:124-573): reads the node-topology
JSON, resolves or creates model chunks, bootstraps every node over the HTTP
control plane (optionally pushing chunk weights in the /init message),
brings up the torch.distributed ring (RCCL on GPUs / gloo on CPU), runs the
recurrent-pipeline generation, and tears everything down with PUT /stop.
"""

from __future__ import annotations

import os
import sys
import time
from datetime import timedelta
from pathlib import Path
from typing import Optional

import torch
import torch.distributed as dist

from ..config import ModelConfig
from ..models.stages import build_stage
from ..prompts import (
    get_user_prompt,
    has_prompt_style,
    load_prompt_style,
    model_name_to_prompt_style,
)
from ..tokenizer import Tokenizer
from ..utils.checkpoint import load_from_pt
from ..utils.partition import (
    chunk_dir,
    chunk_file_name,
    layer_split,
    split_and_store,
)
from .control import ControlClient, ControlServer, NodeTopology
from .ring import RingComm
from .runner import make_runner
from .runtime import GenerationResult, PipelineRuntime, SamplingParams

__all__ = ["MDIRuntime", "default_dtype"]


def default_dtype(name: Optional[str] = None) -> torch.dtype:
    """bf16 if supported, else fp32 (reference config.py:104-115)."""
    if name:
        return {
            "float32": torch.float32,
            "float16": torch.float16,
            "bfloat16": torch.bfloat16,
        }[name]
    if torch.cuda.is_available() and torch.cuda.is_bf16_supported():
        return torch.bfloat16
    return torch.float32


class MDIRuntime:
    """One node of the MDI deployment (rank 0 = starter)."""

    def __init__(
        self,
        node_type: str,
        config_file: Path,
        ckpt_dir: Optional[Path] = None,
        chunk_path: Optional[Path] = None,
        device: Optional[str] = None,
        dtype: Optional[str] = None,
        model_seq_length: Optional[int] = None,
        verb: bool = False,
    ) -> None:
        self.topology = NodeTopology.from_file(config_file)
        self.verb = verb
        if node_type == "starter":
            self.rank = 0
        elif node_type.startswith("secondary"):
            self.rank = int(node_type.split(":")[1]) + 1
        else:
            raise ValueError(f"bad node_type {node_type!r}")
        self.world = self.topology.n_nodes
        self.device = torch.device(
            self.topology.device_for(self.rank, device)
        )
        self.dtype = default_dtype(dtype)
        self.seq_length = model_seq_length
        self.ckpt_dir = Path(ckpt_dir) if ckpt_dir else None
        self.chunk_path = Path(chunk_path) if chunk_path else None

        self.stage = None
        self.runner = None
        self.comm = None
        self.runtime: Optional[PipelineRuntime] = None
        self.tokenizer = None
        self.prompt_style = None
        self.control_server: Optional[ControlServer] = None

    # ------------------------------------------------------------------
    def _log(self, msg: str) -> None:
        from ..utils.console import get_logger

        get_logger().debug("[node %s] %s", self.rank, msg)
        if self.verb:
            print(f"[node {self.rank}] {msg}", file=sys.stderr, flush=True)

    # ------------------------------------------------------------------
    # starter path
    # ------------------------------------------------------------------
    def _resolve_chunks(self, n_samples: int):
        """Find or create per-stage chunk files; returns (config, list of
        chunk paths or None when weights must be pushed)
        (reference model_dist.py:229-247)."""
        assert self.ckpt_dir is not None, "starter needs --ckpt"
        config = ModelConfig.from_checkpoint(self.ckpt_dir)
        if self.world == 1:
            return config, [None], True
        cdir = chunk_dir(self.ckpt_dir, self.world)
        paths = [cdir / chunk_file_name(i) for i in range(self.world)]
        pre_existing = all(p.is_file() for p in paths)
        if not pre_existing:
            self._log(f"chunks for {self.world} nodes missing -> splitting")
            _, sd = load_from_pt(self.ckpt_dir, config)
            split_and_store(sd, self.world, self.ckpt_dir, config.n_layer)
        return config, paths, pre_existing

    def start(
        self,
        n_samples: int = 3,
        tokens_per_sample: int = 300,
        prompt: str = "Who are you?",
        temperature: float = 0.8,
        top_k: Optional[int] = 200,
        top_p: float = 1.0,
        seed: Optional[int] = None,
        push_weights: bool = False,
    ) -> Optional[GenerationResult]:
        if self.rank == 0:
            return self._run_starter(
                n_samples, tokens_per_sample, prompt, temperature, top_k,
                top_p, seed, push_weights,
            )
        self._run_secondary()
        return None

    def _run_starter(self, n_samples, tokens_per_sample, prompt, temperature,
                     top_k, top_p, seed, push_weights) -> GenerationResult:
        config, chunk_paths, pre_existing = self._resolve_chunks(n_samples)
        split = layer_split(config.n_layer, self.world)
        seq_len = min(self.seq_length or config.block_size, config.block_size)

        # 1. configure secondaries over HTTP (reference model_dist.py:402-484)
        client = ControlClient()
        # envelope (pipelined) serve: device-side header routing on GPU
        # secondaries, no per-hop host syncs; MDI_ENV_SERVE=0/1 overrides
        env_serve = self.device.type == "cuda"
        if os.environ.get("MDI_ENV_SERVE"):
            env_serve = os.environ["MDI_ENV_SERVE"] not in ("0", "false")
        for r in range(1, self.world):
            addr, port = self.topology.http_endpoint(r)
            msg = {
                "role": f"secondary:{r-1}",
                "rank": r,
                "world": self.world,
                "model_config": config.to_dict(),
                "n_nodes": self.world,
                "n_local_layers": split[r],
                "n_samples": n_samples,
                "max_seq_length": seq_len,
                "master_addr": self.topology.master_addr,
                "master_port": self.topology.master_port,
                "env_serve": env_serve,
                # upper bound on decoded S (prompt allowance + new tokens):
                # lets the engine pick block-local vs split-S attention
                "expected_s": min(seq_len, tokens_per_sample + 256),
            }
            # path hint works on shared/local filesystems; params are pushed
            # when the chunks were split just now (the secondary host may not
            # see this filesystem) or on request (model_dist.py:454-456)
            msg["chunk_path"] = str(chunk_paths[r])
            if push_weights or not pre_existing:
                _, sd = load_from_pt(self.ckpt_dir, config)
                from ..utils.partition import split_parameters

                msg["params"] = split_parameters(sd, self.world,
                                                 config.n_layer)[r]
            self._log(f"POST /init -> node {r} ({addr}:{port})")
            client.init_node(addr, port, msg)

        # 2. bring up the ring
        if self.world > 1:
            os.environ.setdefault("MASTER_ADDR", self.topology.master_addr)
            os.environ.setdefault("MASTER_PORT", str(self.topology.master_port))
            # MDI_DIST_BACKEND overrides (e.g. gloo to run several ranks
            # on ONE GPU — RCCL refuses duplicate devices; the ring then
            # stages through host mirrors automatically)
            backend = "nccl" if self.device.type == "cuda" else "gloo"
            backend = os.environ.get("MDI_DIST_BACKEND", backend)
            # bounded timeout: a dead peer surfaces as an error instead of a
            # wedged ring (reference detects peer death via zero-byte recv,
            # connections.py:174-182)
            dist.init_process_group(backend, rank=0, world_size=self.world,
                                    timeout=timedelta(seconds=600))

        # 3. build the local stage
        self.stage = build_stage(config, 0, split[0])
        if self.world == 1:
            _, sd = load_from_pt(self.ckpt_dir, config)
        else:
            sd = torch.load(chunk_paths[0], weights_only=True) \
                if chunk_paths[0] and chunk_paths[0].is_file() else None
            if sd is None:
                _, full = load_from_pt(self.ckpt_dir, config)
                from ..utils.partition import split_parameters

                sd = split_parameters(full, self.world, config.n_layer)[0]
        self.stage.load_state_dict(sd)
        self.stage = self.stage.to(device=self.device, dtype=self.dtype)
        self.stage.max_seq_length = seq_len
        self.stage.eval()

        self.runner = make_runner(
            self.stage, n_samples, self.device,
            expected_s=min(seq_len, tokens_per_sample + 256),
        )
        if self.world > 1:
            self.comm = RingComm(config.n_embd, seq_len, self.device,
                                 n_samples, dtype=self.dtype)
        self.runtime = PipelineRuntime(self.runner, 0, self.world, self.comm,
                                       self.device)

        # tokenizer + prompt style + stop tokens (gptserver.py:716-749)
        self.tokenizer = Tokenizer(self.ckpt_dir)
        self.prompt_style = (
            load_prompt_style(self.ckpt_dir)
            if has_prompt_style(self.ckpt_dir)
            else model_name_to_prompt_style(config.name)
        )
        stop_tokens = self.prompt_style.stop_tokens(self.tokenizer)

        prompts_text = get_user_prompt(prompt, n_samples)
        prompts = [
            self.tokenizer.encode(self.prompt_style.apply(p),
                                  device=self.device)
            for p in prompts_text
        ]

        self._log(f"generating {n_samples} samples x {tokens_per_sample} tok")
        res = self.runtime.generate(
            prompts,
            tokens_per_sample,
            SamplingParams(temperature, top_k, top_p, seed),
            stop_tokens=stop_tokens,
            env=env_serve and self.world > 1,
        )

        # 4. teardown (PUT /stop; reference model_dist.py:486-497)
        if self.world > 1:
            dist.barrier()
        for r in range(1, self.world):
            addr, port = self.topology.http_endpoint(r)
            client.stop_node(addr, port)
        if self.world > 1:
            dist.destroy_process_group()
        return res

    # ------------------------------------------------------------------
    # secondary path
    # ------------------------------------------------------------------
    def _run_secondary(self) -> None:
        addr, port = self.topology.http_endpoint(self.rank)
        bind = "0.0.0.0" if addr not in ("127.0.0.1", "localhost") else addr
        self.control_server = ControlServer(bind, port)
        self._log(f"secondary {self.rank}: waiting for /init on :{port}")
        msg = self.control_server.wait_for_init()

        config = ModelConfig.from_dict(msg["model_config"])
        n_local = msg["n_local_layers"]
        n_samples = msg["n_samples"]
        seq_len = msg["max_seq_length"]

        os.environ.setdefault("MASTER_ADDR", msg["master_addr"])
        os.environ.setdefault("MASTER_PORT", str(msg["master_port"]))
        backend = "nccl" if self.device.type == "cuda" else "gloo"
        backend = os.environ.get("MDI_DIST_BACKEND", backend)
        dist.init_process_group(backend, rank=msg["rank"],
                                world_size=msg["world"],
                                timeout=timedelta(seconds=600))

        self.stage = build_stage(config, self.rank, n_local)
        hint = Path(msg["chunk_path"]) if msg.get("chunk_path") else None
        if self.chunk_path and self.chunk_path.is_file():
            self.stage.load_state_dict(
                torch.load(self.chunk_path, weights_only=True)
            )
        elif "params" in msg:
            self.stage.load_state_dict(msg["params"])
        elif hint and hint.is_file():
            self.stage.load_state_dict(torch.load(hint, weights_only=True))
        else:
            raise FileNotFoundError(
                "no chunk available: pass --chunk or let the starter push "
                "weights"
            )
        self.stage = self.stage.to(device=self.device, dtype=self.dtype)
        self.stage.max_seq_length = seq_len
        self.stage.eval()

        env_serve = bool(msg.get("env_serve"))
        # +1 KV slot: the envelope serve routes stop/flush envelopes to a
        # scratch slot on device
        self.runner = make_runner(self.stage,
                                  n_samples + (1 if env_serve else 0),
                                  self.device,
                                  expected_s=msg.get("expected_s"))
        self.comm = RingComm(config.n_embd, seq_len, self.device, n_samples,
                             dtype=self.dtype)
        self.runtime = PipelineRuntime(self.runner, msg["rank"],
                                       msg["world"], self.comm, self.device)
        self._log(f"serving (env={env_serve})")
        self.runtime.serve(env=env_serve, n_samples=n_samples)
        dist.barrier()
        dist.destroy_process_group()
        # wait for the control-plane stop before exiting
        self.control_server.stop_event.wait(timeout=120)
        self.control_server.shutdown()
        self._log("stopped")
