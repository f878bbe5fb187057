"""Multi-rank pipeline on real silicon: several pipeline stages SHARING one
MI355X (gloo ring with device staging — RCCL refuses two ranks on one
device), each running the HIP DecodeEngine with hipGraph replays.

Proves on-GPU what tests/test_pipeline_cpu.py proves on CPU: the pipelined
generation (fused starter graph + ring hops + in-band stops) produces
exactly the tokens of standalone generation, and the scheduled
(headerless) bench path matches too.  VERDICT round-1 item 1: the
interaction of graph replays with ring recv ordering and per-slot
send-buffer reuse, exercised on the GPU.

Reference behaviour matched: /root/reference/src/sub/gptserver.py:788-1110
(starter + secondary loops, per-sample caches, stop propagation).
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

DEV = "cuda:0"
N_SAMPLES = 3
MAX_NEW = 12
MODEL = "nano-gpu"


def _build_and_save(tmp, model=MODEL):
    from mdi_llm_amd import GPT, ModelConfig

    torch.manual_seed(0)
    cfg = ModelConfig.from_name(model)
    m = GPT(cfg)
    m.apply_init()
    m = m.to(dtype=torch.bfloat16)
    m.eval()
    torch.save(m.state_dict(), os.path.join(tmp, "model.pt"))
    torch.manual_seed(1)
    prompts = [torch.randint(0, cfg.vocab_size - 1, (n,)) for n in (5, 8, 3)]
    torch.save(prompts, os.path.join(tmp, "prompts.pt"))
    return cfg, m, prompts


def _standalone_hip(tmp, model=MODEL, top_p=1.0):
    """Standalone (1-node) generation on the HIP engine (fused graphs)."""
    from mdi_llm_amd import ModelConfig
    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.parallel.runner import make_runner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams

    cfg = ModelConfig.from_name(model)
    sd = torch.load(os.path.join(tmp, "model.pt"), weights_only=True)
    prompts = torch.load(os.path.join(tmp, "prompts.pt"), weights_only=True)
    stage = StarterStage(cfg, cfg.n_layer).to(DEV, dtype=torch.bfloat16)
    stage.load_state_dict(sd)
    stage.eval()
    runner = make_runner(stage, N_SAMPLES, torch.device(DEV))
    assert runner.backend == "hip"
    rt = PipelineRuntime(runner, device=torch.device(DEV))
    res = rt.generate(
        [p.to(DEV) for p in prompts], MAX_NEW,
        SamplingParams(temperature=0.8, top_k=50, top_p=top_p, seed=42),
    )
    return [s.tolist() for s in res.sequences]


def _worker(rank, world, tmp, port, out_file, top_p, model=MODEL,
            env=False):
    import torch.distributed as dist

    from mdi_llm_amd import ModelConfig
    from mdi_llm_amd.models.stages import build_stage
    from mdi_llm_amd.parallel.ring import RingComm
    from mdi_llm_amd.parallel.runner import make_runner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams
    from mdi_llm_amd.utils import layer_split, split_parameters

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank,
        world_size=world,
    )
    device = torch.device(DEV)
    torch.cuda.set_device(device)
    cfg = ModelConfig.from_name(model)
    sd = torch.load(os.path.join(tmp, "model.pt"), weights_only=True)
    split = layer_split(cfg.n_layer, world)
    chunks = split_parameters(sd, world)
    stage = build_stage(cfg, rank, split[rank]).to(device,
                                                   dtype=torch.bfloat16)
    stage.load_state_dict(chunks[rank])
    stage.eval()

    n_slots = N_SAMPLES + (1 if (env and rank > 0) else 0)
    runner = make_runner(stage, n_slots, device)
    assert runner.backend == "hip", "HIP engine must drive the GPU pipeline"
    comm = RingComm(cfg.n_embd, stage.max_seq_length, device, N_SAMPLES,
                    dtype=torch.bfloat16)
    assert comm.staged, "gloo + cuda must select the staged ring"
    rt = PipelineRuntime(runner, rank=rank, world=world, comm=comm,
                         device=device)

    if rank == 0:
        prompts = torch.load(os.path.join(tmp, "prompts.pt"),
                             weights_only=True)
        res = rt.generate(
            [p.to(device) for p in prompts], MAX_NEW,
            SamplingParams(temperature=0.8, top_k=50, top_p=top_p, seed=42),
            env=env,
        )
        torch.save([s.tolist() for s in res.sequences], out_file)
    else:
        rt.serve(env=env, n_samples=N_SAMPLES)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 3])
def test_gpu_pipeline_matches_standalone(world, tmp_path):
    """Fused-graph pipeline (starter_step) vs fused standalone: token-exact."""
    tmp = str(tmp_path)
    _build_and_save(tmp)
    ref = _standalone_hip(tmp)
    out_file = os.path.join(tmp, "out.pt")
    port = 29711 + world
    mp.spawn(_worker, args=(world, tmp, port, out_file, 1.0), nprocs=world,
             join=True)
    got = torch.load(out_file, weights_only=True)
    assert got == ref


@pytest.mark.parametrize("world", [2, 3])
def test_gpu_pipeline_env_matches_standalone(world, tmp_path):
    """Envelope serve on the HIP engine: device-side header routing
    (route_env graph), pre-posted recv pairs, stop/flush protocol —
    token-exact vs standalone on the GPU."""
    tmp = str(tmp_path)
    _build_and_save(tmp)
    ref = _standalone_hip(tmp)
    out_file = os.path.join(tmp, "out.pt")
    port = 29721 + world
    mp.spawn(_worker, args=(world, tmp, port, out_file, 1.0, MODEL, True),
             nprocs=world, join=True)
    got = torch.load(out_file, weights_only=True)
    assert got == ref


def test_gpu_pipeline_topp_matches_standalone(tmp_path):
    """top_p < 1 path (HIP tail + torch sampler w/ per-sample generators):
    pipeline and standalone draw identical tokens."""
    tmp = str(tmp_path)
    _build_and_save(tmp)
    ref = _standalone_hip(tmp, top_p=0.9)
    out_file = os.path.join(tmp, "out.pt")
    mp.spawn(_worker, args=(2, tmp, 29741, out_file, 0.9), nprocs=2,
             join=True)
    got = torch.load(out_file, weights_only=True)
    assert got == ref


# ---------------------------------------------------------------------------
# scheduled (headerless) bench path — what the driver's SCALE run executes
# ---------------------------------------------------------------------------
def _bench_worker(rank, world, tmp, port, out_file, model=MODEL):
    import torch.distributed as dist

    from mdi_llm_amd import ModelConfig
    from mdi_llm_amd.models.stages import build_stage
    from mdi_llm_amd.parallel.ring import RingComm
    from mdi_llm_amd.parallel.runner import make_runner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams
    from mdi_llm_amd.utils import layer_split, split_parameters

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank,
        world_size=world,
    )
    device = torch.device(DEV)
    torch.cuda.set_device(device)
    cfg = ModelConfig.from_name(model)
    sd = torch.load(os.path.join(tmp, "model.pt"), weights_only=True)
    split = layer_split(cfg.n_layer, world)
    chunks = split_parameters(sd, world)
    stage = build_stage(cfg, rank, split[rank]).to(device,
                                                   dtype=torch.bfloat16)
    stage.load_state_dict(chunks[rank])
    stage.eval()

    runner = make_runner(stage, N_SAMPLES, device)
    assert runner.backend == "hip"
    comm = RingComm(cfg.n_embd, stage.max_seq_length, device, N_SAMPLES,
                    dtype=torch.bfloat16)
    rt = PipelineRuntime(runner, rank=rank, world=world, comm=comm,
                         device=device)
    sampling = SamplingParams(temperature=0.8, top_k=50, seed=42)
    rounds = 8

    if rank == 0:
        prompts = torch.load(os.path.join(tmp, "prompts.pt"),
                             weights_only=True)
        rt.prepare_bench(sampling, N_SAMPLES)
        toks = rt.bench_prefill([p.to(device) for p in prompts])
        toks = rt.bench_decode_rounds(toks, rounds, sampling)
        torch.save([int(t) for t in toks], out_file)
    else:
        rt.bench_serve_prefill(N_SAMPLES)
        rt.bench_serve_rounds(N_SAMPLES, rounds)
    dist.barrier()
    dist.destroy_process_group()


def _bench_standalone(tmp, model=MODEL):
    from mdi_llm_amd import ModelConfig
    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.parallel.runner import make_runner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams

    cfg = ModelConfig.from_name(model)
    sd = torch.load(os.path.join(tmp, "model.pt"), weights_only=True)
    prompts = torch.load(os.path.join(tmp, "prompts.pt"), weights_only=True)
    stage = StarterStage(cfg, cfg.n_layer).to(DEV, dtype=torch.bfloat16)
    stage.load_state_dict(sd)
    stage.eval()
    runner = make_runner(stage, N_SAMPLES, torch.device(DEV))
    rt = PipelineRuntime(runner, device=torch.device(DEV))
    sampling = SamplingParams(temperature=0.8, top_k=50, seed=42)
    rt.prepare_bench(sampling, N_SAMPLES)
    toks = rt.bench_prefill([p.to(DEV) for p in prompts])
    toks = rt.bench_decode_rounds(toks, 8, sampling)
    return [int(t) for t in toks]


def test_gpu_sched_bench_matches_standalone(tmp_path):
    """bench.py's decode-rotation phase at world 2 lands on the same final
    tokens as world 1 (deterministic (seed, slot, pos)-keyed sampling)."""
    tmp = str(tmp_path)
    _build_and_save(tmp)
    ref = _bench_standalone(tmp)
    out_file = os.path.join(tmp, "out.pt")
    mp.spawn(_bench_worker, args=(2, tmp, 29751, out_file), nprocs=2,
             join=True)
    got = torch.load(out_file, weights_only=True)
    assert got == ref
