"""Full-system CPU test: starter + secondary processes bootstrapped over
the HTTP control plane (chunk auto-split, /init with RCCL rendezvous info,
generation through the gloo ring, /stop teardown) — the reference's
starter.py/secondary.py flow end to end."""

import json
import multiprocessing as mp
import os

import pytest
import torch

from helpers import make_toy_checkpoint


def _topo(tmp, http0, http1, dist_port):
    return {
        "nodes": {
            "starter": {
                "addr": "127.0.0.1",
                "communication": {"port": http0},
                "inference": {"port_in": dist_port, "port_out": dist_port + 1},
                "device": "cpu",
            },
            "secondary": [
                {
                    "addr": "127.0.0.1",
                    "communication": {"starter_addr": "127.0.0.1",
                                      "port": http1},
                    "inference": {"port_in": dist_port + 4,
                                  "port_out": dist_port + 5},
                    "device": "cpu",
                }
            ],
        }
    }


def _secondary_proc(cfg_path, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    from mdi_llm_amd.parallel.orchestrator import MDIRuntime

    rt = MDIRuntime("secondary:0", cfg_path, dtype="float32", verb=True)
    rt.start()
    q.put("secondary-done")


def _starter_proc(cfg_path, ckpt, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    from mdi_llm_amd.parallel.orchestrator import MDIRuntime

    rt = MDIRuntime("starter", cfg_path, ckpt_dir=ckpt, dtype="float32",
                    verb=True)
    res = rt.start(n_samples=2, tokens_per_sample=6, prompt="hello world",
                   seed=3)
    q.put([s.tolist() for s in res.sequences])


@pytest.mark.timeout(180)
def test_two_node_http_bootstrap(tmp_path):
    ckpt = make_toy_checkpoint(tmp_path / "NanoTest", name="nano-test")
    cfg_path = tmp_path / "topo.json"
    cfg_path.write_text(json.dumps(_topo(tmp_path, 18791, 18793, 29821)))

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    sec = ctx.Process(target=_secondary_proc, args=(str(cfg_path), q))
    st = ctx.Process(target=_starter_proc, args=(str(cfg_path), str(ckpt), q))
    sec.start()
    st.start()
    st.join(timeout=150)
    sec.join(timeout=30)
    if sec.is_alive():
        sec.terminate()
    assert st.exitcode == 0, "starter failed"
    results = [q.get(timeout=5), q.get(timeout=5)]
    seqs = next(r for r in results if isinstance(r, list))
    assert len(seqs) == 2
    assert all(len(s) > 2 for s in seqs)
    # chunks were auto-split to the reference on-disk layout
    assert (ckpt / "chunks" / "2nodes" / "model_starter.pth").is_file()
    assert (ckpt / "chunks" / "2nodes" / "model_secondary0.pth").is_file()
