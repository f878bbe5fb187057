"""HF Hub weight download (network-gated).

Capability parity with the reference downloader
(/root/reference/src/sub/utils/download.py:15-181: snapshot download of
tokenizer + weights, safetensors preference, gated-repo error surfacing).
"""

from __future__ import annotations

from pathlib import Path
from typing import Optional, Union

__all__ = ["download_from_hub"]


def download_from_hub(
    repo_id: str,
    checkpoints_root: Union[str, Path] = "checkpoints",
    access_token: Optional[str] = None,
    tokenizer_only: bool = False,
    convert: bool = True,
) -> Path:
    """Download ``org/model`` into ``checkpoints/org/model`` and (optionally)
    convert to the litGPT layout."""
    try:
        from huggingface_hub import snapshot_download
        from huggingface_hub.utils import GatedRepoError, RepositoryNotFoundError
    except ImportError as e:  # pragma: no cover
        raise RuntimeError("huggingface_hub is required for downloads") from e

    target = Path(checkpoints_root) / repo_id
    patterns = (
        ["tokenizer*", "*.json", "*.model"]
        if tokenizer_only
        else ["tokenizer*", "*.json", "*.model", "*.safetensors", "*.bin"]
    )
    try:
        snapshot_download(
            repo_id,
            local_dir=target,
            allow_patterns=patterns,
            token=access_token,
        )
    except GatedRepoError as e:
        raise RuntimeError(
            f"{repo_id} is a gated repo: pass --access-token / set HF_TOKEN "
            "(reference behavior: download.py:146-181)"
        ) from e
    except RepositoryNotFoundError as e:
        raise RuntimeError(f"repository {repo_id!r} not found") from e

    if convert and not tokenizer_only:
        from .convert_hf import convert_hf_checkpoint

        convert_hf_checkpoint(target, model_name=Path(repo_id).name)
    return target
