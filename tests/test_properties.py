"""Property-based tests (hypothesis) for geometry-sensitive utilities:
the qkv weave, the byte-balanced partitioner, and config round-trips."""

import torch
from hypothesis import given, settings, strategies as st

from mdi_llm_amd.config import ModelConfig


def _geom():
    return st.tuples(
        st.sampled_from([1, 2, 4, 8]),        # n_query_groups
        st.sampled_from([1, 2, 4, 8, 16]),    # q_per_kv
        st.sampled_from([16, 32, 64, 128]),   # head_size
    )


@settings(max_examples=25, deadline=None)
@given(_geom())
def test_weave_unweave_roundtrip(geom):
    ng, qpk, hs = geom
    from mdi_llm_amd.utils.convert_hf import unweave_qkv, weave_qkv

    cfg = ModelConfig.from_name(
        "nano-test", n_head=ng * qpk, n_query_groups=ng,
        n_embd=ng * qpk * hs, head_size=hs, intermediate_size=64,
    )
    E = 32
    q = torch.randn(ng * qpk * hs, E)
    k = torch.randn(ng * hs, E)
    v = torch.randn(ng * hs, E)
    qkv = weave_qkv(q, k, v, cfg)
    assert qkv.shape == ((qpk + 2) * ng * hs, E)
    q2, k2, v2 = unweave_qkv(qkv, cfg)
    assert torch.equal(q, q2) and torch.equal(k, k2) and torch.equal(v, v2)


@settings(max_examples=30, deadline=None)
@given(st.integers(2, 120), st.integers(1, 8))
def test_balanced_split_invariants(n_layer, n_nodes):
    from mdi_llm_amd.utils.partition import balanced_split

    if n_nodes > n_layer:
        return
    cfg = ModelConfig.from_name("nano-test", n_layer=n_layer)
    s = balanced_split(cfg, n_nodes)
    assert len(s) == n_nodes and sum(s) == n_layer
    assert all(c >= 0 for c in s)
    if n_nodes > 1:
        assert all(c >= 1 for c in s[1:])  # secondaries never empty


@settings(max_examples=20, deadline=None)
@given(st.sampled_from(["nano-test", "nano-test-moe", "nano-test-falcon",
                        "Meta-Llama-3-8B-Instruct", "gpt2-xl",
                        "falcon-40b", "Mixtral-8x7B-v0.1"]))
def test_config_dict_roundtrip(name):
    cfg = ModelConfig.from_name(name)
    cfg2 = ModelConfig.from_dict(cfg.to_dict())
    assert cfg == cfg2


@settings(max_examples=20, deadline=None)
@given(st.integers(1, 1000), st.integers(1, 7))
def test_layer_split_table_consistency(n_layer, n_nodes):
    from mdi_llm_amd.utils.partition import layer_split

    if n_nodes > n_layer:
        return
    s = layer_split(n_layer, n_nodes)
    assert sum(s) == n_layer and len(s) == n_nodes
    assert max(s) - min(s) <= max(2, n_layer)  # sane spread
