"""CLI end-to-end smokes (CPU): sample.py generation, prepare_data + train.py
(single and DDP), prepare_model chunking, scripts/inspect_lit."""

import subprocess
import sys
from pathlib import Path

import pytest

from helpers import make_toy_checkpoint

ROOT = Path(__file__).resolve().parent.parent


def run(args, timeout=240):
    return subprocess.run([sys.executable] + args, cwd=ROOT,
                          capture_output=True, text=True, timeout=timeout)


@pytest.fixture(scope="module")
def ckpt(tmp_path_factory):
    d = tmp_path_factory.mktemp("ck") / "NanoTest"
    make_toy_checkpoint(d, "nano-test")
    return d


def test_sample_cli(ckpt):
    r = run(["sample.py", "--ckpt", str(ckpt), "--n-samples", "2",
             "--n-tokens", "6", "--device", "cpu", "--dtype", "float32"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert "sample 1" in r.stdout
    assert "tok/s" in r.stdout


def test_prepare_model_chunks(ckpt):
    r = run(["prepare_model.py", "--ckpt", str(ckpt), "--n-nodes", "3"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert (ckpt / "chunks" / "3nodes" / "model_secondary1.pth").is_file()


def test_inspect_lit(ckpt):
    r = run(["scripts/inspect_lit.py", str(ckpt)])
    assert r.returncode == 0, r.stderr[-1000:]
    assert "n_layer" in r.stdout


def test_train_and_ddp(ckpt, tmp_path):
    data = tmp_path / "data"
    text_file = tmp_path / "input.txt"
    text_file.write_text("the quick brown fox jumps over the lazy dog. " * 400)
    r = run(["prepare_data.py", "--input", str(text_file),
             "--tokenizer-dir", str(ckpt), "--out-dir", str(data)])
    assert r.returncode == 0, r.stderr[-1000:]

    r = run(["train.py", "--ckpt", str(ckpt), "--data-dir", str(data),
             "--max-iters", "4", "--batch-size", "2", "--block-size", "32",
             "--grad-accum", "2", "--dtype", "float32",
             "--eval-interval", "100", "--log-interval", "2"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert "loss" in r.stdout

    r = run(["-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", "29671", "train.py", "--ckpt", str(ckpt),
             "--data-dir", str(data), "--max-iters", "3",
             "--batch-size", "2", "--block-size", "32", "--grad-accum", "2",
             "--dtype", "float32", "--eval-interval", "100",
             "--log-interval", "1"])
    assert r.returncode == 0, r.stderr[-2000:]


def test_bench_cpu_mode(tmp_path):
    r = run(["-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", "29672", "bench.py", "--gpus", "2",
             "--steps", "3", "--warmup", "1", "--model", "nano-test",
             "--prompt-len", "8", "--seq-len", "64"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert '"n_gpus": 2' in r.stdout


@pytest.mark.parametrize("world", [4, 8])
def test_bench_cpu_mode_wide(world):
    """The exact flow the driver runs on the 8-GPU node (torchrun, one
    rank per device, balanced split) on an 8-layer model, CPU/gloo."""
    r = run(["-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
             "--master-port", str(29680 + world), "bench.py",
             "--gpus", str(world), "--steps", "3", "--warmup", "1",
             "--model", "nano-test-deep", "--prompt-len", "8",
             "--seq-len", "64"], timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    assert f'"n_gpus": {world}' in r.stdout
    assert '"value"' in r.stdout


def test_chat_cli_two_turns(ckpt, tmp_path):
    """chat.py streaming loop: two piped turns, clean EOF exit."""
    import subprocess
    import sys as _sys

    r = subprocess.run(
        [_sys.executable, "chat.py", "--ckpt", str(ckpt),
         "--max-new-tokens", "8", "--device", "cpu"],
        input="who are you?\nand what can you do?\n",
        capture_output=True, text=True, timeout=300, cwd=ROOT,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert r.stdout.count(">>") >= 2
