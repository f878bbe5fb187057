"""Recurrent pipeline runtime: the scheduler that keeps every stage busy.

Capability parity with the reference's node loops
(/root/reference/src/sub/gptserver.py: ``_starter_loop`` 788-1019,
``_secondary_loop`` 1021-1110): >= n_stages samples in flight, each with its
own KV-cache slot, single-token activation messages between stages, in-band
per-sample stop envelopes that travel the whole ring back to the starter,
and the tok/time instrumentation the reference writes to CSV.

Re-designed for one 8xMI355X node: one process per GPU over RCCL/xGMI
(``RingComm``), the decode step on each stage either replayed as a hipGraph
(``HipRunner``) or run by the torch stage module (CPU tests use gloo with
the same code).  Standalone (1 node) short-circuits the ring exactly like
the reference's queue aliasing (gptserver.py:276-278).
"""

from __future__ import annotations

import os
import time
from dataclasses import dataclass, field
from typing import List, Optional, Sequence, Tuple

import torch

from ..models.sampling import sample as sample_token
from .ring import (ENV_DATA, ENV_FLUSH, ENV_FLUSHES, ENV_STOP, ENV_WINDOW,
                   RingComm)

__all__ = ["PipelineRuntime", "GenerationResult", "SamplingParams"]


@dataclass
class SamplingParams:
    temperature: float = 0.8
    top_k: Optional[int] = 200
    top_p: float = 1.0
    seed: Optional[int] = None


@dataclass
class GenerationResult:
    sequences: List[torch.Tensor] = field(default_factory=list)
    tok_time: List[Tuple[int, float]] = field(default_factory=list)
    gen_time: float = 0.0
    total_new_tokens: int = 0

    @property
    def tokens_per_second(self) -> float:
        return self.total_new_tokens / self.gen_time if self.gen_time else 0.0


class PipelineRuntime:
    """One rank of the ring (or the single standalone node)."""

    def __init__(
        self,
        runner,
        rank: int = 0,
        world: int = 1,
        comm: Optional[RingComm] = None,
        device: Optional[torch.device] = None,
    ) -> None:
        self.runner = runner
        self.rank = rank
        self.world = world
        self.comm = comm
        self.is_starter = rank == 0
        self.device = device or torch.device("cpu")
        if world > 1:
            assert comm is not None

    # ------------------------------------------------------------------
    # starter
    # ------------------------------------------------------------------
    def generate(
        self,
        prompts: Sequence[torch.Tensor],
        max_new_tokens: int,
        sampling: SamplingParams = SamplingParams(),
        stop_tokens: Sequence[Sequence[int]] = (),
        token_callback=None,
        env: bool = False,
    ) -> GenerationResult:
        """Starter entry: run the recurrent pipeline until every sample
        produced ``max_new_tokens`` tokens (or hit a stop sequence).

        env=True selects the envelope protocol for decode messages (fixed
        hdr+payload pairs, stop/flush envelopes carry dummy payloads):
        secondaries then serve with serve(env=True, ...) — device-side
        header routing, no per-hop host syncs."""
        assert self.is_starter, "generate() runs on the starter"
        env = env and self.world > 1
        n_samples = len(prompts)
        runner = self.runner
        if (self.world == 1 and getattr(runner, "backend", "") == "hip"
                and runner.engine.use_graphs and sampling.top_p >= 1.0):
            return self._generate_standalone_fused(
                prompts, max_new_tokens, sampling, stop_tokens,
                token_callback,
            )
        gens = self._generators(sampling, n_samples, self.device)
        # pipelined fused path: tail+sample+next-head in one graph replay;
        # graphs must be captured BEFORE prefill (capture wipes caches)
        fused = (
            self.world > 1
            and getattr(runner, "backend", "") == "hip"
            and runner.engine.use_graphs
            and sampling.top_p >= 1.0
        )
        if fused:
            runner.engine.ensure_fused_graphs(
                sampling.temperature, sampling.top_k, sampling.seed or 0
            )
            runner.reset()

        res = GenerationResult()
        seqs: List[List[int]] = [list(map(int, p.tolist())) for p in prompts]
        new_counts = [0] * n_samples
        active = [True] * n_samples
        pending_x = {}

        t_start = time.perf_counter()
        # ---- prefill: seed every sample into the ring --------------------
        for s, prompt in enumerate(prompts):
            ptoks = prompt.to(self.device)
            x = runner.prefill_head(ptoks, s)
            if self.world > 1:
                self.comm.send(s, x, stop=False)
            else:
                pending_x[s] = x
        order = list(range(n_samples))

        # ---- main loop ---------------------------------------------------
        n_active = n_samples
        total_new = 0
        while n_active > 0:
            if self.world > 1:
                s, x, stop = self.comm.recv()
                if stop:
                    n_active -= 1
                    continue
            else:
                s = order.pop(0)
                x = pending_x.pop(s)
            # tail: logits for the last position of this sample
            x2 = x.view(-1, x.size(-1))
            if fused:
                # one graph: tail + on-GPU sample + embed + blocks
                runner.engine.starter_step(x2[-1], s)
                runner.pos[s] += 1
                itok = int(runner.engine.token_table[s])
                self._fused_x = runner.engine.x
            else:
                logits = runner.tail(x2[-1])
                tok = sample_token(
                    logits,
                    temperature=sampling.temperature,
                    top_k=sampling.top_k,
                    top_p=sampling.top_p,
                    generator=gens[s],
                )
                itok = int(tok)
            seqs[s].append(itok)
            new_counts[s] += 1
            total_new += 1
            res.tok_time.append((total_new, time.perf_counter() - t_start))
            if token_callback is not None:
                token_callback(s, itok)

            done = (
                new_counts[s] >= max_new_tokens
                or self._hit_stop(seqs[s], new_counts[s], stop_tokens)
                or runner.pos[s] + 1 >= self.runner.stage.max_seq_length
            )
            if done:
                active[s] = False
                if self.world > 1:
                    # stop envelope travels the whole ring back here
                    if env:
                        self.comm.send_env(s, None, ENV_STOP)
                    else:
                        self.comm.send(s, None, stop=True)
                else:
                    n_active -= 1
            elif fused:
                # the graph already embedded the sampled token and ran the
                # local blocks; just forward the activations
                if env:
                    self.comm.send_env(s, self._fused_x, ENV_DATA)
                else:
                    self.comm.send(s, self._fused_x, stop=False)
            else:
                x = runner.decode_head(tok.view(1).to(self.device), s)
                if self.world > 1:
                    if env:
                        self.comm.send_env(s, x, ENV_DATA)
                    else:
                        self.comm.send(s, x, stop=False)
                else:
                    order.append(s)
                    pending_x[s] = x.clone() if self._needs_clone() else x

        if self.world > 1:
            if env:
                # flush padding: fills every pre-posted secondary recv so
                # the ring drains with no unmatched work (see ring.py)
                for _ in range(ENV_FLUSHES):
                    self.comm.send_env(-1, None, ENV_FLUSH)
                for _ in range(ENV_FLUSHES):
                    self.comm.recv()
            self.comm.drain()
        res.gen_time = time.perf_counter() - t_start
        res.total_new_tokens = total_new
        res.sequences = [torch.tensor(s, dtype=torch.int64) for s in seqs]
        return res

    def _generate_standalone_fused(
        self, prompts, max_new_tokens, sampling: SamplingParams,
        stop_tokens, token_callback,
    ) -> GenerationResult:
        """Standalone generation on the fused hipGraph step (one graph
        replay + one 4-byte readback per token).  Sampling runs on-GPU
        (radix top-k + gumbel) with the run's seed.

        Determinism note: draws come from a (seed, slot, position) hash,
        so each sample's token stream is reproducible and independent of
        the scheduling order across samples — one sample stopping early
        does not shift any other sample's draws (matching the torch
        path's per-sample generator streams in spirit)."""
        runner = self.runner
        eng = runner.engine
        n_samples = len(prompts)
        # graphs must be captured before prefill (capture scribbles caches)
        L = self._lane_count(n_samples)
        eng.ensure_fused_graphs(sampling.temperature, sampling.top_k,
                                sampling.seed or 0, n_lanes=L)
        runner.reset()

        res = GenerationResult()
        seqs = [list(map(int, p.tolist())) for p in prompts]
        t_start = time.perf_counter()
        for s, prompt in enumerate(prompts):
            x = runner.prefill_head(prompt.to(self.device), s)
            eng.slot.fill_(s)
            eng.tail_sample_step(x.view(-1, x.size(-1))[-1], s)

        new_counts = [0] * n_samples
        active = set(range(n_samples))
        total_new = 0
        while active:
            # consume the current token of every active sample (one D2H
            # sync point per round), then launch the next steps lane-
            # parallel: samples on different lanes overlap on the GPU
            stepping = []
            for s in sorted(active):
                tok = int(eng.token_table[s])
                seqs[s].append(tok)
                new_counts[s] += 1
                total_new += 1
                res.tok_time.append(
                    (total_new, time.perf_counter() - t_start))
                if token_callback is not None:
                    token_callback(s, tok)
                if (new_counts[s] >= max_new_tokens
                        or self._hit_stop(seqs[s], new_counts[s],
                                          stop_tokens)
                        or runner.pos[s] + 1 >= runner.stage.max_seq_length):
                    active.discard(s)
                    continue
                stepping.append(s)
            if eng.n_lanes == 1:
                for s in stepping:
                    eng.standalone_step(s)
                    runner.pos[s] += 1
            else:
                eng.lanes_begin()
                for s in stepping:
                    eng.standalone_lane_step(s % eng.n_lanes, s)
                    runner.pos[s] += 1
                eng.lanes_join()
        res.gen_time = time.perf_counter() - t_start
        res.total_new_tokens = total_new
        res.sequences = [torch.tensor(q, dtype=torch.int64) for q in seqs]
        return res

    def _needs_clone(self) -> bool:
        # HIP runner returns a view of the engine's x buffer; standalone
        # loopback must snapshot it (the ring path copies into send bufs).
        return getattr(self.runner, "backend", "") == "hip"

    @staticmethod
    def _generators(sampling: SamplingParams, n: int, device):
        if sampling.seed is None:
            return [None] * n
        gens = []
        for s in range(n):
            g = torch.Generator(device=device)
            g.manual_seed(sampling.seed + s)
            gens.append(g)
        return gens

    @staticmethod
    def _hit_stop(seq: List[int], n_new: int, stop_tokens) -> bool:
        for st in stop_tokens:
            n = len(st)
            if 0 < n <= n_new and seq[-n:] == list(st):
                return True
        return False

    # ------------------------------------------------------------------
    # bench phases (deterministic schedule, headerless ring, no host syncs
    # in the decode loop; each phase leaves the ring drained so callers can
    # barrier/synchronize between phases)
    # ------------------------------------------------------------------
    @staticmethod
    def _lane_count(n_samples: int) -> int:
        """Concurrent HIP-stream lanes for standalone multi-sample decode
        (MDI_LANES; 0/unset = auto: up to 4, capped by the sample count).
        Small models are launch-floor-bound, so overlapping samples on
        separate streams raises aggregate throughput without changing
        any token (sampling is (seed, slot, pos)-keyed)."""
        v = int(os.environ.get("MDI_LANES", "0") or 0)
        if v <= 0:
            v = 4
        return max(1, min(v, n_samples))

    def prepare_bench(self, sampling: SamplingParams,
                      n_samples: int = 1) -> None:
        """Capture the fused step graphs BEFORE prefill (capture warm-up
        scribbles on the KV pool, so it must precede cache filling)."""
        r = self.runner
        if (self.is_starter and getattr(r, "backend", "") == "hip"
                and sampling.top_p >= 1.0 and r.engine.use_graphs):
            lanes = self._lane_count(n_samples) if self.world == 1 else 1
            r.engine.ensure_fused_graphs(
                sampling.temperature, sampling.top_k, sampling.seed or 0,
                n_lanes=lanes,
            )
            r.reset()

    def bench_prefill(self, prompts: Sequence[torch.Tensor]) -> list:
        """Starter: prefill all samples; returns per-sample device tokens
        (argmax of the first logits — value irrelevant for timing)."""
        runner = self.runner
        n = len(prompts)
        xs = []
        for s, p in enumerate(prompts):
            x = runner.prefill_head(p.to(self.device), s)
            if self.world > 1:
                self.comm.send(s, x, stop=False)
            else:
                xs.append(x)
        toks = []
        for s in range(n):
            if self.world > 1:
                _, x, _ = self.comm.recv()
            else:
                x = xs[s]
            logits = runner.tail(x.view(-1, x.size(-1))[-1])
            toks.append(logits.float().argmax().view(1).to(torch.int32))
        return toks

    def bench_serve_prefill(self, n_samples: int) -> None:
        for _ in range(n_samples):
            s, x, _ = self.comm.recv()
            out = self.runner.prefill_mid(x, s)
            self.comm.send(s, out, stop=False)
        self.comm.drain()

    def bench_decode_rounds(self, toks: list, n_rounds: int,
                            sampling: SamplingParams, gens=None) -> list:
        """Starter: n_rounds full rotations; every sample advances one token
        per rotation.  The final rotation does not re-seed, so the ring is
        empty on return."""
        runner = self.runner
        n = len(toks)
        if gens is None:
            gens = self._generators(sampling, n, self.device)
        fused = (
            getattr(runner, "backend", "") == "hip"
            and sampling.top_p >= 1.0
            and runner.engine.use_graphs
        )
        eng = runner.engine if fused else None
        if fused:
            eng.ensure_fused_graphs(
                sampling.temperature, sampling.top_k, sampling.seed or 0,
                n_lanes=self._lane_count(n) if self.world == 1 else 1)
            for s in range(n):
                eng.token_table[s] = toks[s].view(())

        if self.world == 1:
            if fused:
                L = eng.n_lanes
                if L == 1:
                    # single lane: stay on the current stream (no stream
                    # round-trip per replay)
                    for _ in range(n_rounds):
                        for s in range(n):
                            eng.standalone_step(s)
                            runner.pos[s] += 1
                    return [eng.token_table[s: s + 1] for s in range(n)]
                eng.lanes_begin()
                for _ in range(n_rounds):
                    for s in range(n):
                        eng.standalone_lane_step(s % L, s)
                        runner.pos[s] += 1
                eng.lanes_join()
                return [eng.token_table[s: s + 1] for s in range(n)]
            for _ in range(n_rounds):
                for s in range(n):
                    x = runner.decode_head(toks[s], s)
                    logits = runner.tail(x)
                    toks[s] = self._draw(logits, sampling, gens[s])
            return toks

        # ---- pipeline: seed one in-flight message per sample -------------
        from .ring import RecvRing

        for s in range(n):
            if fused:
                x = runner.decode_head(eng.token_table[s: s + 1], s)
            else:
                x = runner.decode_head(toks[s], s)
            self.comm.send_sched(s, x)
        # pre-posted recv window: message i+W is on the wire while message
        # i's compute runs (arrival overlaps compute)
        ring = RecvRing(self.comm, (1, self.comm.n_embd), n * n_rounds,
                        dtype=self.comm.dtype)
        for r in range(n_rounds):
            last = r == n_rounds - 1
            for s in range(n):
                x = ring.take()[0]
                if fused:
                    if last:
                        eng.tail_sample_step(x, s)
                    else:
                        out = eng.starter_step(x, s)
                        runner.pos[s] += 1
                        self.comm.send_sched(s, out)
                    ring.repost()
                    continue
                logits = runner.tail(x)
                toks[s] = self._draw(logits, sampling, gens[s])
                if not last:
                    x = runner.decode_head(toks[s], s)
                    self.comm.send_sched(s, x)
                ring.repost()
        self.comm.drain()
        if fused:
            return [eng.token_table[s: s + 1] for s in range(n)]
        return toks

    # ------------------------------------------------------------------
    # grouped (batched) bench phases — GroupDecodeEngine drives B samples
    # per pass; a rotation advances all n_groups*B samples by one token
    # ------------------------------------------------------------------
    def bench_group_rounds(self, geng, group_slots, n_rounds: int) -> None:
        """Starter: n_rounds rotations over all groups (graphs captured by
        geng.ensure_graphs beforehand)."""
        G = len(group_slots)
        if self.world == 1:
            L = geng.n_lanes
            if L == 1:
                for _ in range(n_rounds):
                    for g in range(G):
                        geng.set_group(group_slots[g])
                        geng.standalone_step()
                return
            geng.lanes_begin()
            for _ in range(n_rounds):
                for g in range(G):
                    geng.standalone_lane_step(g % L, group_slots[g])
            geng.lanes_join()
            return
        from .ring import RecvRing

        for g in range(G):
            geng.set_group(group_slots[g])
            X = geng.head_step()
            self.comm.send_group(g, X)
        ring = RecvRing(self.comm, self.comm.grecv.shape, G * n_rounds,
                        dtype=self.comm.grecv.dtype)
        for r in range(n_rounds):
            last = r == n_rounds - 1
            for g in range(G):
                X = ring.take()
                geng.set_group(group_slots[g])
                if last:
                    geng.tail_step(X)
                else:
                    out = geng.starter_step(X)
                    self.comm.send_group(g, out)
                ring.repost()
        self.comm.drain()

    def bench_group_serve(self, geng, group_slots, n_rounds: int) -> None:
        from .ring import RecvRing

        G = len(group_slots)
        ring = RecvRing(self.comm, self.comm.grecv.shape, G * n_rounds,
                        dtype=self.comm.grecv.dtype)
        for r in range(n_rounds):
            for g in range(G):
                X = ring.take()
                geng.set_group(group_slots[g])
                out = geng.mid_step(X)
                self.comm.send_group(g, out)
                ring.repost()
        self.comm.drain()

    def bench_serve_rounds(self, n_samples: int, n_rounds: int) -> None:
        """Secondary: the matching deterministic message count, with a
        pre-posted recv window so the next activation rides RCCL while
        the current one computes."""
        from .ring import RecvRing

        ring = RecvRing(self.comm, (1, self.comm.n_embd),
                        n_samples * n_rounds, dtype=self.comm.dtype)
        for r in range(n_rounds):
            for s in range(n_samples):
                x = ring.take()[0]
                out = self.runner.decode_mid(x, s)
                self.comm.send_sched(s, out)
                ring.repost()
        self.comm.drain()

    def _draw(self, logits, sampling: SamplingParams, gen):
        r = self.runner
        if getattr(r, "backend", "") == "hip":
            # fused on-GPU sampler (radix top-k + mass-radix top-p +
            # gumbel), no host sync.  Clone: sample_out is a single device
            # scalar reused by every draw — callers keep per-sample tokens
            # across rounds, so each must own its value.
            return r.engine.sample_into_token(
                sampling.temperature, sampling.top_k, sampling.seed or 0,
                sampling.top_p,
            ).clone()
        tok = sample_token(
            logits,
            temperature=sampling.temperature,
            top_k=sampling.top_k,
            top_p=sampling.top_p,
            generator=gen,
        )
        return tok.view(1).to(torch.int32)

    # ------------------------------------------------------------------
    # secondary
    # ------------------------------------------------------------------
    def serve(self, env: bool = False,
              n_samples: Optional[int] = None) -> int:
        """Secondary entry: process messages until every sample stopped.
        Returns the number of activation messages processed.  Exceptions /
        Ctrl-C shut the loop down cleanly (reference
        utils/context_managers.py:16-56 semantics).

        env=True (requires n_samples): pipelined envelope serve — W
        pre-posted recv pairs, device-side header routing on the HIP
        engine, lagged stop detection; the host never blocks the GPU on a
        header read (round-1 VERDICT item 2)."""
        from ..utils.context_managers import catch_loop_errors

        assert not self.is_starter
        runner = self.runner
        with catch_loop_errors(label=f"secondary-{self.rank}"):
            if env:
                assert n_samples is not None, "env serve needs n_samples"
                return self._serve_env(n_samples)
            return self._serve_loop(runner, set(), set(), 0)
        return 0

    def _serve_env(self, n_samples: int) -> int:
        """Envelope serve: classic prefill phase, then the pipelined
        decode loop."""
        runner = self.runner
        hip = getattr(runner, "backend", "") == "hip"
        if hip:
            # graph capture scribbles the KV pool: do it BEFORE prefill
            runner.engine.ensure_env_graph()
        # ---- phase 1: prefill (variable-T, classic headered protocol) ---
        for _ in range(n_samples):
            s, x, _stop = self.comm.recv()
            out = runner.prefill_mid(x, s)
            self.comm.send(s, out, stop=False)
        # ---- phase 2: pipelined decode ----------------------------------
        W = ENV_WINDOW
        ring = self.comm.env_ring(W)
        self.comm.env_alloc_send(W)
        use_lag = self.device.type == "cuda" and not self.comm.staged
        if use_lag:
            lag_pin = torch.zeros(W, 4, dtype=torch.int32).pin_memory()
            lag_ev = [torch.cuda.Event() for _ in range(W)]
        kinds: dict = {}
        processed = 0
        i = 0
        while True:
            if i >= W:
                j = i - W
                if use_lag:
                    lag_ev[j % W].synchronize()
                    kj = int(lag_pin[j % W][2])
                else:
                    kj = kinds.pop(j)
                if kj == ENV_FLUSH:
                    break
            w, hdr, buf, khost = ring.take()
            if hip:
                out = runner.engine.env_step(hdr, buf)
                self.comm.send_env_fwd(w, hdr, out)
                if use_lag:
                    lag_pin[w].copy_(hdr, non_blocking=True)
                    lag_ev[w].record()
                else:
                    kinds[i] = khost
            else:
                # host-routed fallback (torch backend): header is a CPU
                # tensor (or staged mirror) — free to read
                s = int(hdr[0])
                if khost == ENV_DATA:
                    out = runner.decode_mid(buf[0], s)
                else:
                    out = buf[0]
                self.comm.send_env_fwd(w, hdr, out)
                kinds[i] = khost
            processed += 1
            ring.repost()
            i += 1
        # drain: the flush padding guarantees every posted recv is matched
        while ring.outstanding() > 0:
            w, hdr, buf, _ = ring.take()
            self.comm.send_env_fwd(w, hdr, buf[0])
        self.comm.drain()
        return processed

    def _serve_loop(self, runner, seen, stopped, processed) -> int:
        while True:
            s, x, stop = self.comm.recv()
            if stop:
                stopped.add(s)
                self.comm.send(s, None, stop=True)  # forward along the ring
                if seen and stopped >= seen:
                    break
                if not seen:
                    break
                continue
            seen.add(s)
            if x.size(0) > 1:
                out = runner.prefill_mid(x, s)
            else:
                out = runner.decode_mid(x[0], s)
            self.comm.send(s, out, stop=False)
            processed += 1
        self.comm.drain()
        return processed
