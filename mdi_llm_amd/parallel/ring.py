"""Activation ring over torch.distributed P2P (RCCL on GPU, gloo on CPU).

MI355X-native replacement for the reference's TCP-socket data plane
(/root/reference/src/sub/connections.py: 16-byte ASCII header + pickled
dict per hop).  Here a hop is one (or two) ``isend`` of fixed-layout
tensors on the torch stream: an int32 header [sample, ntok, stop] and a
bf16 activation payload.  On ROCm the ``nccl`` backend IS RCCL and adjacent
ranks map to single xGMI links; on CPU test hosts the same code runs over
``gloo``.

A headerless "scheduled" mode skips the header entirely when the message
sequence is deterministic (the bench decode loop): payloads are fixed-size
(1, n_embd) and arrival order is the seeding order, so no host readback of
headers is needed on the critical path.

A "staged" mode bridges backends that cannot move device tensors (gloo):
device payloads are staged through pinned host mirrors around each
send/recv.  This is how multiple pipeline ranks can SHARE one physical
GPU (RCCL refuses two ranks on one device), so the full HIP pipeline —
graph replays interleaved with ring traffic, per-slot send-buffer reuse —
is testable on a single-GPU box.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist

__all__ = ["RingComm"]

HDR_LEN = 4  # sample_id, ntok, stop, reserved


class RingComm:
    def __init__(
        self,
        n_embd: int,
        max_seq: int,
        device: torch.device,
        n_slots: int,
        dtype: torch.dtype = torch.bfloat16,
        group: Optional[dist.ProcessGroup] = None,
        staged: Optional[bool] = None,
    ) -> None:
        assert dist.is_initialized(), "torch.distributed must be initialized"
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        self.next_rank = (self.rank + 1) % self.world
        self.prev_rank = (self.rank - 1) % self.world
        self.device = device
        self.n_embd = n_embd
        self.max_seq = max_seq
        if staged is None:
            # gloo cannot carry CUDA tensors point-to-point: stage via host
            backend = str(dist.get_backend(group))
            staged = device.type == "cuda" and "gloo" in backend
        self.staged = bool(staged)

        dev = device
        # per-slot single-token send buffers (decode) + one shared prefill
        # buffer, so isend never races buffer reuse
        self.send_hdr = torch.zeros(n_slots + 1, HDR_LEN, dtype=torch.int32,
                                    device=dev)
        self.send_buf = torch.zeros(n_slots + 1, 1, n_embd,
                                    dtype=dtype, device=dev)
        self.prefill_buf = torch.zeros(max_seq, n_embd,
                                       dtype=dtype, device=dev)
        self.recv_hdr = torch.zeros(HDR_LEN, dtype=torch.int32, device=dev)
        self.recv_buf = torch.zeros(max_seq, n_embd, dtype=dtype,
                                    device=dev)
        self._pending: dict = {}
        self._PF = "prefill"
        self._mirrors: dict = {}

    # -- staged-mode helpers ----------------------------------------------
    def _mirror(self, key, t: torch.Tensor) -> torch.Tensor:
        m = self._mirrors.get(key)
        if m is None or m.shape != t.shape:
            m = torch.empty_like(t, device="cpu").pin_memory()
            self._mirrors[key] = m
        return m

    def _isend(self, t: torch.Tensor, key) -> "dist.Work":
        """isend of a device tensor; staged mode bounces through a pinned
        per-key host mirror (blocking D2H copy orders after the producing
        stream work, then the wire send is async)."""
        if not self.staged:
            return dist.isend(t, self.next_rank, group=self.group)
        m = self._mirror(("s", key), t)
        m.copy_(t)  # syncs the current stream for this copy only
        return dist.isend(m, self.next_rank, group=self.group)

    def _recv_into(self, t: torch.Tensor, key) -> None:
        if not self.staged:
            dist.recv(t, self.prev_rank, group=self.group)
            return
        m = self._mirror(("r", key), t)
        dist.recv(m, self.prev_rank, group=self.group)
        # blocking H2D: the mirror is reused by the next recv, so the copy
        # must complete before this returns (staged mode is a correctness
        # bridge, not the RCCL fast path)
        t.copy_(m)

    # -- headered path (general generate loop) ----------------------------
    def send(self, sample: int, x: Optional[torch.Tensor],
             stop: bool = False) -> None:
        """Send one activation message (or a stop envelope) to next rank."""
        slot = sample if sample >= 0 else self.send_hdr.size(0) - 1
        ntok = 0 if x is None else x.view(-1, self.n_embd).size(0)
        key = self._PF if ntok > 1 else slot
        for k in (key, slot):
            if k in self._pending:
                for w in self._pending.pop(k):
                    w.wait()
        payload = None
        if ntok > 0:
            x2 = x.view(-1, self.n_embd)
            if ntok > 1:
                payload = self.prefill_buf[:ntok]
            else:
                payload = self.send_buf[slot, :1]
            payload.copy_(x2)
        hdr = self.send_hdr[slot]
        hdr[0], hdr[1], hdr[2], hdr[3] = sample, ntok, int(stop), 0
        works = [self._isend(hdr, ("h", slot))]
        if payload is not None:
            works.append(self._isend(payload, key))
        self._pending[key] = works

    def recv(self) -> Tuple[int, Optional[torch.Tensor], bool]:
        """Blocking receive of one message from prev rank.

        Returns (sample, activations (ntok, n_embd) view or None, stop).
        """
        if self.staged:
            hdr = self._mirror(("r", "hdr"), self.recv_hdr)
            dist.recv(hdr, self.prev_rank, group=self.group)
        else:
            dist.recv(self.recv_hdr, self.prev_rank, group=self.group)
            hdr = self.recv_hdr.cpu()  # host sync: 16 bytes
        sample, ntok, stop = int(hdr[0]), int(hdr[1]), bool(hdr[2])
        x = None
        if ntok > 0:
            x = self.recv_buf[:ntok]
            self._recv_into(x, "payload")
        return sample, x, stop

    # -- scheduled (headerless) path: fixed T=1 payloads -------------------
    def send_sched(self, slot: int, x: torch.Tensor) -> None:
        if slot in self._pending:
            for w in self._pending.pop(slot):
                w.wait()
        self.send_buf[slot, 0].copy_(x.view(-1))
        self._pending[slot] = [
            self._isend(self.send_buf[slot, :1], slot)
        ]

    def recv_sched(self) -> torch.Tensor:
        self._recv_into(self.recv_buf[:1], "sched")
        return self.recv_buf[0]

    # -- grouped (batched) scheduled path: fixed (B, n_embd) payloads ------
    def alloc_groups(self, n_groups: int, group_size: int,
                     dtype: torch.dtype = torch.bfloat16) -> None:
        self.gsend = torch.zeros(n_groups, group_size, self.n_embd,
                                 dtype=dtype, device=self.device)
        self.grecv = torch.zeros(group_size, self.n_embd, dtype=dtype,
                                 device=self.device)

    def send_group(self, gi: int, X: torch.Tensor) -> None:
        key = ("g", gi)
        if key in self._pending:
            for w in self._pending.pop(key):
                w.wait()
        buf = self.gsend[gi]
        buf.copy_(X.view(buf.shape))
        self._pending[key] = [
            self._isend(buf, key)
        ]

    def recv_group(self) -> torch.Tensor:
        self._recv_into(self.grecv, "group")
        return self.grecv

    def drain(self) -> None:
        for works in self._pending.values():
            for w in works:
                w.wait()
        self._pending.clear()
