"""The bench's deterministic scheduled phases (headerless ring) must run
deadlock-free on world_size 2 (gloo) and produce the right message counts —
this is the exact code path the driver runs on 8 GPUs."""

import os

import torch
import torch.multiprocessing as mp

from mdi_llm_amd import GPT, ModelConfig

N_SAMPLES = 2
ROUNDS_W = 2
ROUNDS_K = 3


def _worker(rank, world, tmp, port, out_file):
    import torch.distributed as dist

    from mdi_llm_amd.models.stages import build_stage
    from mdi_llm_amd.parallel.ring import RingComm
    from mdi_llm_amd.parallel.runner import TorchRunner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams
    from mdi_llm_amd.utils import layer_split, split_parameters

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank,
        world_size=world,
    )
    cfg = ModelConfig.from_name("nano-test")
    sd = torch.load(os.path.join(tmp, "model.pt"), weights_only=True)
    split = layer_split(cfg.n_layer, world)
    chunks = split_parameters(sd, world)
    stage = build_stage(cfg, rank, split[rank])
    stage.load_state_dict(chunks[rank])
    stage.eval()
    runner = TorchRunner(stage, N_SAMPLES)
    comm = RingComm(cfg.n_embd, stage.max_seq_length, torch.device("cpu"),
                    N_SAMPLES, dtype=torch.float32)
    rt = PipelineRuntime(runner, rank=rank, world=world, comm=comm)
    sampling = SamplingParams(temperature=0.8, top_k=20, seed=3)

    torch.manual_seed(5)
    prompts = [torch.randint(0, 255, (6,)) for _ in range(N_SAMPLES)]
    if rank == 0:
        toks = rt.bench_prefill(prompts)
        dist.barrier()
        toks = rt.bench_decode_rounds(toks, ROUNDS_W, sampling)
        dist.barrier()
        toks = rt.bench_decode_rounds(toks, ROUNDS_K, sampling)
        dist.barrier()
        # every sample advanced prompt + W + K positions
        assert runner.pos[0] == 6 + ROUNDS_W + ROUNDS_K, runner.pos
        torch.save([int(t) for t in toks], out_file)
    else:
        rt.bench_serve_prefill(N_SAMPLES)
        dist.barrier()
        rt.bench_serve_rounds(N_SAMPLES, ROUNDS_W)
        dist.barrier()
        rt.bench_serve_rounds(N_SAMPLES, ROUNDS_K)
        dist.barrier()
        assert runner.pos[0] == 6 + ROUNDS_W + ROUNDS_K, runner.pos
    dist.destroy_process_group()


def test_bench_phases_world2(tmp_path):
    tmp = str(tmp_path)
    torch.manual_seed(0)
    cfg = ModelConfig.from_name("nano-test")
    m = GPT(cfg)
    m.apply_init()
    torch.save(m.state_dict(), os.path.join(tmp, "model.pt"))
    out_file = os.path.join(tmp, "toks.pt")
    mp.spawn(_worker, args=(2, tmp, 29755, out_file), nprocs=2, join=True)
    toks = torch.load(out_file, weights_only=True)
    assert len(toks) == N_SAMPLES
