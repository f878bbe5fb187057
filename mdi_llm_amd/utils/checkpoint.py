"""litGPT-format checkpoint IO.

Capability parity with the reference's checkpoint handling
(/root/reference/src/sub/utils/utils.py:495-641 ``load_sd``/``load_from_pt``/
``save_config``/``init_from_state_dict`` and the lazy-loading machinery in
utils/litgpt_utils.py).  On-disk layout is kept compatible:
``<ckpt_dir>/model_config.yaml`` + ``lit_model.pth`` (+ tokenizer files).

Memory-lean loading uses ``torch.load(mmap=True)`` — the modern equivalent
of the reference's hand-rolled ``NotYetLoadedTensor`` unpickler (tensors
page in lazily from disk instead of materializing the whole file in RAM).
"""

from __future__ import annotations

from pathlib import Path
from typing import Optional, Tuple, Union

import torch

from ..config import ModelConfig

__all__ = [
    "load_from_pt",
    "load_state_dict_lazy",
    "save_checkpoint",
    "init_from_state_dict",
    "get_checkpoint_files",
]

PathLike = Union[str, Path]


def get_checkpoint_files(ckpt_dir: PathLike) -> Tuple[Path, Path]:
    ckpt_dir = Path(ckpt_dir)
    return ckpt_dir / "model_config.yaml", ckpt_dir / "lit_model.pth"


def load_state_dict_lazy(path: PathLike) -> dict:
    """mmap-load a .pth state dict (tensors lazily paged from disk)."""
    try:
        return torch.load(path, map_location="cpu", mmap=True,
                          weights_only=True)
    except (RuntimeError, ValueError):
        # older zip-less files can't mmap
        return torch.load(path, map_location="cpu", weights_only=True)


def load_from_pt(
    ckpt_dir: PathLike,
    config: Optional[ModelConfig] = None,
) -> Tuple[ModelConfig, dict]:
    """Load a litGPT checkpoint dir -> (config, state_dict).

    Mirrors the reference's ``load_from_pt`` (utils/utils.py:527-562).
    """
    cfg_file, model_file = get_checkpoint_files(ckpt_dir)
    if config is None:
        config = ModelConfig.from_checkpoint(ckpt_dir)
    if not model_file.is_file():
        raise FileNotFoundError(f"{model_file} not found")
    sd = load_state_dict_lazy(model_file)
    if "model" in sd and isinstance(sd["model"], dict):
        sd = sd["model"]
    return config, sd


def save_checkpoint(ckpt_dir: PathLike, config: ModelConfig,
                    state_dict: dict) -> None:
    ckpt_dir = Path(ckpt_dir)
    ckpt_dir.mkdir(parents=True, exist_ok=True)
    cfg_file, model_file = get_checkpoint_files(ckpt_dir)
    config.save(cfg_file)
    torch.save(state_dict, model_file)


def init_from_state_dict(module: torch.nn.Module, state_dict: dict,
                         strict: bool = True) -> None:
    """Fill a (possibly meta-device) module from a state dict
    (reference utils/utils.py:614-641)."""
    module.load_state_dict(state_dict, strict=strict, assign=True)
