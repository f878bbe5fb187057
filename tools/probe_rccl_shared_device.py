#!/usr/bin/env python3
"""Probe: does RCCL accept two communicator ranks on ONE GPU?

NCCL/RCCL documents one rank per device; this probes the actual behaviour
on the MI355X image so the single-GPU pipeline validation path (gloo +
staged ring) is chosen on evidence, not folklore.  Run under `timeout`:
a duplicate-device init may hang instead of erroring.

Usage: python tools/probe_rccl_shared_device.py
"""

import os
import sys

import torch
import torch.multiprocessing as mp


def _worker(rank, world, port):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=rank, world_size=world)
    t = torch.ones(8, device="cuda:0") * (rank + 1)
    if rank == 0:
        dist.send(t, 1)
        dist.recv(t, 1)
        print(f"[probe] rank0 got {t[0].item()} (expect 2.0)", flush=True)
    else:
        r = torch.zeros(8, device="cuda:0")
        dist.recv(r, 0)
        dist.send(t, 0)
    dist.barrier()
    dist.destroy_process_group()


def main():
    try:
        mp.spawn(_worker, args=(2, 29799), nprocs=2, join=True)
        print("[probe] RCCL accepted 2 ranks on one device: send/recv OK")
    except Exception as e:  # noqa: BLE001
        print(f"[probe] RCCL refused/failed 2 ranks on one device: "
              f"{type(e).__name__}: {str(e)[:500]}")
        sys.exit(1)


if __name__ == "__main__":
    main()
