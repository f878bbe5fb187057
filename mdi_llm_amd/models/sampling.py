"""Token sampling: temperature, top-k, top-p (nucleus), multinomial.

Capability parity with the reference sampler
(/root/reference/src/sub/model.py:34-90: ``sample``, ``sample_top_p``,
``multinomial_num_samples_1``).  Fresh implementation; the HIP decode engine
has a fused on-GPU version of the same semantics, validated against this.
"""

from __future__ import annotations

from typing import Optional

import torch

__all__ = ["sample", "sample_top_p", "logits_to_probs"]


def sample_top_p(logits: torch.Tensor, top_p: float) -> torch.Tensor:
    """Mask logits outside the nucleus of cumulative probability ``top_p``."""
    sorted_logits, sorted_indices = torch.sort(logits, descending=False)
    cum_probs = sorted_logits.softmax(dim=-1).cumsum(dim=-1)
    # drop tokens whose cumulative prob (from the low end) stays below 1-p
    drop = cum_probs <= (1.0 - top_p)
    # always keep the most likely token
    drop[..., -1] = False
    indices_to_drop = sorted_indices[drop]
    logits = logits.clone()
    logits[indices_to_drop] = float("-inf")
    return logits


def logits_to_probs(
    logits: torch.Tensor,
    temperature: float = 1.0,
    top_k: Optional[int] = None,
    top_p: float = 1.0,
) -> torch.Tensor:
    """Apply temperature/top-k/top-p and return a probability vector."""
    logits = logits.float()
    if temperature > 0.0:
        logits = logits / max(temperature, 1e-5)
    if top_k is not None and top_k > 0 and top_k < logits.size(-1):
        kth = torch.topk(logits, top_k).values[..., -1, None]
        logits = torch.where(
            logits < kth, torch.full_like(logits, float("-inf")), logits
        )
    if 0.0 < top_p < 1.0:
        logits = sample_top_p(logits, top_p)
    return torch.softmax(logits, dim=-1)


def sample(
    logits: torch.Tensor,
    temperature: float = 0.8,
    top_k: Optional[int] = 200,
    top_p: float = 1.0,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """Draw the next token id from ``logits`` (1-D, vocab-sized).

    ``temperature == 0`` is greedy argmax (reference model.py:90).
    """
    if logits.dim() > 1:
        logits = logits.reshape(-1)
    if temperature == 0.0:
        return torch.argmax(logits, dim=-1, keepdim=False)
    probs = logits_to_probs(logits, temperature, top_k, top_p)
    # exponential-race trick == multinomial with 1 draw but generator-stable
    # on every device (reference model.py:34-39 uses the same construction).
    q = torch.empty_like(probs).exponential_(1, generator=generator)
    return torch.argmax(probs / q, dim=-1, keepdim=False)
