"""Multi-process pipeline correctness on CPU (gloo, world_size 2 and 3):
pipelined generation must produce exactly the tokens of standalone
generation (same model, same seeds)."""

import os
import tempfile

import pytest
import torch
import torch.multiprocessing as mp

from mdi_llm_amd import GPT, ModelConfig

N_SAMPLES = 3
MAX_NEW = 10


def _build_and_save(tmp, model="nano-test"):
    torch.manual_seed(0)
    cfg = ModelConfig.from_name(model)
    m = GPT(cfg)
    m.apply_init()
    m.eval()
    torch.save(m.state_dict(), os.path.join(tmp, "model.pt"))
    torch.manual_seed(1)
    prompts = [torch.randint(0, 255, (n,)) for n in (5, 8, 3)]
    torch.save(prompts, os.path.join(tmp, "prompts.pt"))
    return cfg, m, prompts


def _standalone_reference(tmp, model="nano-test"):
    """Standalone (1-node) generation via the same runtime."""
    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.parallel.runner import TorchRunner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams

    cfg = ModelConfig.from_name(model)
    sd = torch.load(os.path.join(tmp, "model.pt"), weights_only=True)
    prompts = torch.load(os.path.join(tmp, "prompts.pt"), weights_only=True)
    stage = StarterStage(cfg, cfg.n_layer)
    stage.load_state_dict(sd)
    stage.eval()
    runner = TorchRunner(stage, N_SAMPLES)
    rt = PipelineRuntime(runner)
    res = rt.generate(prompts, MAX_NEW,
                      SamplingParams(temperature=0.8, top_k=50, seed=42))
    return [s.tolist() for s in res.sequences]


def _worker(rank, world, tmp, port, out_file, model="nano-test",
            env=False):
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
    cfg = ModelConfig.from_name(model)
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

    if rank == 0:
        prompts = torch.load(os.path.join(tmp, "prompts.pt"),
                             weights_only=True)
        res = rt.generate(prompts, MAX_NEW,
                          SamplingParams(temperature=0.8, top_k=50, seed=42),
                          env=env)
        torch.save([s.tolist() for s in res.sequences], out_file)
    else:
        rt.serve(env=env, n_samples=N_SAMPLES)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 3])
def test_pipeline_matches_standalone(world, tmp_path):
    tmp = str(tmp_path)
    _build_and_save(tmp)
    ref = _standalone_reference(tmp)

    port = 29611 + world
    out_file = os.path.join(tmp, "out.pt")
    ctx = mp.spawn(
        _worker, args=(world, tmp, port, out_file), nprocs=world, join=True
    )
    got = torch.load(out_file, weights_only=True)
    assert got == ref


@pytest.mark.parametrize("world", [2, 3])
def test_pipeline_env_mode_matches_standalone(world, tmp_path):
    """Envelope (pipelined-serve) protocol: fixed hdr+payload pairs,
    stop/flush envelopes, pre-posted recv window — same tokens as
    standalone (host-routed fallback on CPU)."""
    tmp = str(tmp_path)
    _build_and_save(tmp)
    ref = _standalone_reference(tmp)
    port = 29651 + world
    out_file = os.path.join(tmp, "out.pt")
    mp.spawn(_worker, args=(world, tmp, port, out_file, "nano-test", True),
             nprocs=world, join=True)
    got = torch.load(out_file, weights_only=True)
    assert got == ref


def test_pipeline_moe_matches_standalone(tmp_path):
    """A mixture-of-experts model through the 2-stage pipeline must
    reproduce its standalone token streams exactly (MoE routing is
    per-token local state — the ring must not perturb it)."""
    tmp = str(tmp_path)
    _build_and_save(tmp, model="nano-test-moe")
    ref = _standalone_reference(tmp, model="nano-test-moe")
    out_file = os.path.join(tmp, "out.pt")
    mp.spawn(_worker, args=(2, tmp, 29640, out_file, "nano-test-moe"),
             nprocs=2, join=True)
    got = torch.load(out_file, weights_only=True)
    assert got == ref


def test_standalone_with_stop_tokens(tmp_path):
    """Stop sequences truncate a sample early; others continue."""
    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.parallel.runner import TorchRunner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams

    tmp = str(tmp_path)
    cfg, m, prompts = _build_and_save(tmp)
    stage = StarterStage(cfg, cfg.n_layer)
    stage.load_state_dict(m.state_dict())
    stage.eval()
    runner = TorchRunner(stage, N_SAMPLES)
    rt = PipelineRuntime(runner)
    base = rt.generate(prompts, MAX_NEW, SamplingParams(seed=7))
    # use the first generated token of sample 0 as its stop token
    stop = (int(base.sequences[0][prompts[0].numel()]),)
    runner.reset()
    res = rt.generate(prompts, MAX_NEW, SamplingParams(seed=7),
                      stop_tokens=[stop])
    assert res.sequences[0].numel() == prompts[0].numel() + 1
    assert res.sequences[1].tolist() == base.sequences[1].tolist()


def _worker_stops(rank, world, tmp, port, out_file, env):
    """Generation with a stop token that truncates sample 0 early."""
    import torch.distributed as dist

    from mdi_llm_amd.models.stages import build_stage
    from mdi_llm_amd.parallel.ring import RingComm
    from mdi_llm_amd.parallel.runner import TorchRunner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams
    from mdi_llm_amd.utils import layer_split, split_parameters

    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{port}",
                            rank=rank, world_size=world)
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
    stop = torch.load(os.path.join(tmp, "stop.pt"), weights_only=True)
    if rank == 0:
        prompts = torch.load(os.path.join(tmp, "prompts.pt"),
                             weights_only=True)
        res = rt.generate(prompts, MAX_NEW, SamplingParams(seed=7),
                          stop_tokens=[stop], env=env)
        torch.save([s.tolist() for s in res.sequences], out_file)
    else:
        rt.serve(env=env, n_samples=N_SAMPLES)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("env", [False, True])
def test_pipeline_stop_tokens_match_standalone(env, tmp_path):
    """Early per-sample stops through the ring (classic and envelope
    protocols): same truncation as standalone."""
    from mdi_llm_amd.models.stages import StarterStage
    from mdi_llm_amd.parallel.runner import TorchRunner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams

    tmp = str(tmp_path)
    cfg, m, prompts = _build_and_save(tmp)
    stage = StarterStage(cfg, cfg.n_layer)
    stage.load_state_dict(m.state_dict())
    stage.eval()
    runner = TorchRunner(stage, N_SAMPLES)
    rt = PipelineRuntime(runner)
    base = rt.generate(prompts, MAX_NEW, SamplingParams(seed=7))
    stop = (int(base.sequences[0][prompts[0].numel()]),)
    runner.reset()
    ref = rt.generate(prompts, MAX_NEW, SamplingParams(seed=7),
                      stop_tokens=[stop])
    ref = [s.tolist() for s in ref.sequences]
    torch.save(list(stop), os.path.join(tmp, "stop.pt"))

    out_file = os.path.join(tmp, "out.pt")
    port = 29681 + int(env)
    mp.spawn(_worker_stops, args=(3, tmp, port, out_file, env), nprocs=3,
             join=True)
    got = torch.load(out_file, weights_only=True)
    assert got == ref


def _worker_tiny(rank, world, tmp, port, out_file):
    """Envelope protocol at the degenerate size: 1 sample, 1 new token."""
    import torch.distributed as dist

    from mdi_llm_amd.models.stages import build_stage
    from mdi_llm_amd.parallel.ring import RingComm
    from mdi_llm_amd.parallel.runner import TorchRunner
    from mdi_llm_amd.parallel.runtime import PipelineRuntime, SamplingParams
    from mdi_llm_amd.utils import layer_split, split_parameters

    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{port}",
                            rank=rank, world_size=world)
    cfg = ModelConfig.from_name("nano-test")
    sd = torch.load(os.path.join(tmp, "model.pt"), weights_only=True)
    split = layer_split(cfg.n_layer, world)
    chunks = split_parameters(sd, world)
    stage = build_stage(cfg, rank, split[rank])
    stage.load_state_dict(chunks[rank])
    stage.eval()
    runner = TorchRunner(stage, 1)
    comm = RingComm(cfg.n_embd, stage.max_seq_length, torch.device("cpu"),
                    1, dtype=torch.float32)
    rt = PipelineRuntime(runner, rank=rank, world=world, comm=comm)
    if rank == 0:
        prompts = [torch.randint(0, 255, (4,))]
        res = rt.generate(prompts, 1, SamplingParams(seed=3), env=True)
        torch.save([s.tolist() for s in res.sequences], out_file)
    else:
        rt.serve(env=True, n_samples=1)
    dist.barrier()
    dist.destroy_process_group()


def test_pipeline_env_one_sample_one_token(tmp_path):
    tmp = str(tmp_path)
    _build_and_save(tmp)
    out_file = os.path.join(tmp, "out.pt")
    mp.spawn(_worker_tiny, args=(2, tmp, 29691, out_file), nprocs=2,
             join=True)
    got = torch.load(out_file, weights_only=True)
    assert len(got) == 1 and len(got[0]) == 5  # 4 prompt + 1 new
