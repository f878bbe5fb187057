"""Direct RingComm unit tests on gloo (world_size 2): headered envelopes
(sample id + stop flag round-trip), headerless scheduled path, and the
grouped [B, E] payloads."""

import os

import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, out_file):
    import torch.distributed as dist

    from mdi_llm_amd.parallel.ring import RingComm

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank,
        world_size=world,
    )
    E, max_seq, n_slots = 16, 32, 3
    comm = RingComm(E, max_seq, torch.device("cpu"), n_slots,
                    dtype=torch.float32)
    torch.manual_seed(10 + rank)

    if rank == 0:
        # headered: payload + sample id survive the hop; stop envelope too
        x = torch.randn(1, 1, E)
        comm.send(2, x, stop=False)
        comm.send(1, None, stop=True)
        # scheduled (headerless)
        y = torch.randn(1, 1, E)
        comm.send_sched(0, y)
        # grouped
        comm.alloc_groups(2, 4, torch.float32)
        G0 = torch.randn(4, E)
        comm.send_group(0, G0)
        # receive the echoes
        s, xe, stop = comm.recv()
        assert s == 2 and not stop and torch.equal(xe.view(-1), x.view(-1))
        s, _, stop = comm.recv()
        assert s == 1 and stop
        ye = comm.recv_sched()
        assert torch.equal(ye.view(-1), y.view(-1))
        Ge = comm.recv_group()
        assert torch.equal(Ge.view(4, E), G0)
        comm.drain()
        torch.save(True, out_file)
    else:
        comm.alloc_groups(2, 4, torch.float32)
        # echo everything back
        s, x, stop = comm.recv()
        comm.send(s, x, stop=stop)
        s, x, stop = comm.recv()
        comm.send(s, x, stop=stop)
        y = comm.recv_sched()
        comm.send_sched(0, y)
        G = comm.recv_group()
        comm.send_group(0, G)
        comm.drain()
    dist.barrier()
    dist.destroy_process_group()


def test_ring_envelopes_roundtrip(tmp_path):
    out_file = os.path.join(str(tmp_path), "ok.pt")
    mp.spawn(_worker, args=(2, 29650, out_file), nprocs=2, join=True)
    assert torch.load(out_file, weights_only=True)
