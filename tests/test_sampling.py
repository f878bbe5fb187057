import torch

from mdi_llm_amd.models.sampling import logits_to_probs, sample, sample_top_p


def test_greedy():
    logits = torch.tensor([0.1, 2.0, -1.0, 0.5])
    assert int(sample(logits, temperature=0.0)) == 1


def test_top_k_restricts_support():
    torch.manual_seed(0)
    logits = torch.randn(100)
    probs = logits_to_probs(logits, temperature=1.0, top_k=5)
    assert (probs > 0).sum() == 5
    topk_idx = torch.topk(logits, 5).indices
    assert set(probs.nonzero().flatten().tolist()) == set(topk_idx.tolist())


def test_top_p_keeps_nucleus():
    logits = torch.log(torch.tensor([0.5, 0.3, 0.15, 0.05]))
    filtered = sample_top_p(logits, 0.8)
    assert torch.isfinite(filtered[0]) and torch.isfinite(filtered[1])
    assert filtered[3] == float("-inf")


def test_top_p_always_keeps_argmax():
    logits = torch.log(torch.tensor([0.97, 0.01, 0.01, 0.01]))
    filtered = sample_top_p(logits, 0.5)
    assert torch.isfinite(filtered[0])


def test_temperature_sharpening():
    logits = torch.tensor([1.0, 0.0])
    sharp = logits_to_probs(logits, temperature=0.1)
    soft = logits_to_probs(logits, temperature=10.0)
    assert sharp[0] > soft[0]


def test_generator_reproducible():
    logits = torch.randn(1000)
    g1 = torch.Generator().manual_seed(7)
    g2 = torch.Generator().manual_seed(7)
    a = [int(sample(logits, generator=g1)) for _ in range(20)]
    b = [int(sample(logits, generator=g2)) for _ in range(20)]
    assert a == b


def test_distribution_roughly_matches():
    torch.manual_seed(0)
    probs_target = torch.tensor([0.7, 0.2, 0.1])
    logits = torch.log(probs_target)
    g = torch.Generator().manual_seed(0)
    draws = torch.tensor(
        [int(sample(logits, temperature=1.0, top_k=None, generator=g)) for _ in range(3000)]
    )
    freq = torch.bincount(draws, minlength=3).float() / 3000
    assert torch.allclose(freq, probs_target, atol=0.05)
