"""Training-data utilities: tokenize text into .bin memmaps + batch loader.

Capability parity with the reference's data pipeline
(/root/reference/src/sub/utils/data_loader.py:14-126 ``load_dataset`` /
``split_dataset`` / ``get_batch`` and the prepare CLIs).
"""

from __future__ import annotations

from pathlib import Path
from typing import Tuple, Union

import numpy as np
import torch

__all__ = ["prepare_bin", "split_dataset", "load_bin", "get_batch"]

PathLike = Union[str, Path]


def split_dataset(tokens: np.ndarray, train_frac: float = 0.9):
    n = int(len(tokens) * train_frac)
    return tokens[:n], tokens[n:]


def prepare_bin(
    text: str,
    tokenizer,
    out_dir: PathLike,
    train_frac: float = 0.9,
) -> Tuple[Path, Path]:
    """Tokenize raw text and write train.bin / val.bin (uint16/uint32)."""
    out_dir = Path(out_dir)
    out_dir.mkdir(parents=True, exist_ok=True)
    ids = tokenizer.encode(text, bos=False).numpy()
    dtype = np.uint16 if ids.max() < 2 ** 16 else np.uint32
    train, val = split_dataset(ids.astype(dtype), train_frac)
    train_p, val_p = out_dir / "train.bin", out_dir / "val.bin"
    train.tofile(train_p)
    val.tofile(val_p)
    return train_p, val_p


def load_bin(path: PathLike, dtype=np.uint16) -> np.ndarray:
    return np.memmap(path, dtype=dtype, mode="r")


def get_batch(
    data: np.ndarray,
    batch_size: int,
    block_size: int,
    device: torch.device = torch.device("cpu"),
    generator: torch.Generator = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Random (x, y) LM batch from a memmapped token array
    (reference data_loader.py:70-126)."""
    ix = torch.randint(len(data) - block_size - 1, (batch_size,),
                       generator=generator)
    x = torch.stack(
        [torch.from_numpy(data[i: i + block_size].astype(np.int64))
         for i in ix]
    )
    y = torch.stack(
        [torch.from_numpy(data[i + 1: i + 1 + block_size].astype(np.int64))
         for i in ix]
    )
    if device.type == "cuda":
        x = x.pin_memory().to(device, non_blocking=True)
        y = y.pin_memory().to(device, non_blocking=True)
    else:
        x, y = x.to(device), y.to(device)
    return x, y
