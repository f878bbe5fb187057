"""Generation text/token utilities.

Parity with the reference helpers (utils/utils.py:28-59, 185-239):
``get_obj_size``, ``find_eot``, ``detect_stop_tokens``.
"""

from __future__ import annotations

import sys
from typing import Iterable, List, Sequence

import torch

__all__ = ["get_obj_size", "find_eot", "detect_stop_tokens"]


def get_obj_size(obj) -> int:
    """Approximate in-memory size in bytes (tensors counted by storage)."""
    if torch.is_tensor(obj):
        return obj.numel() * obj.element_size()
    if isinstance(obj, dict):
        return sum(get_obj_size(v) for v in obj.values())
    if isinstance(obj, (list, tuple, set)):
        return sum(get_obj_size(v) for v in obj)
    return sys.getsizeof(obj)


def detect_stop_tokens(tokens: Sequence[int],
                       stop_sequences: Iterable[Sequence[int]]) -> bool:
    """True when ``tokens`` ends with any of the stop sequences."""
    toks: List[int] = list(tokens)
    for seq in stop_sequences:
        n = len(seq)
        if 0 < n <= len(toks) and toks[-n:] == list(seq):
            return True
    return False


def find_eot(tokens: torch.Tensor,
             stop_sequences: Iterable[Sequence[int]],
             prompt_len: int = 0) -> int:
    """Index right before the first stop sequence after ``prompt_len``
    (for truncating decoded output); len(tokens) when none found."""
    toks = tokens.tolist() if torch.is_tensor(tokens) else list(tokens)
    for i in range(prompt_len, len(toks)):
        for seq in stop_sequences:
            n = len(seq)
            if n and toks[i:i + n] == list(seq):
                return i
    return len(toks)
