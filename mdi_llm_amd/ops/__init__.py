"""Loader for the hand-written CDNA4 (gfx950) kernel extension.

The extension is built in-tree (``python setup.py build_ext --inplace`` →
``mdi_llm_amd/_hip_ops*.so``) so it travels with the repo snapshot to GPU
boxes.  On a machine with a GPU the HIP path is mandatory: a missing
extension raises instead of silently falling back to eager PyTorch
(the driver checks which .so the GPU processes actually load).
"""

from __future__ import annotations

import torch

__all__ = ["hip_ops", "have_hip_ops", "require_hip_ops"]

try:
    from mdi_llm_amd import _hip_ops as hip_ops  # type: ignore

    have_hip_ops = True
except ImportError as _e:  # pragma: no cover
    hip_ops = None
    have_hip_ops = False
    _import_error = _e


def require_hip_ops():
    """Return the extension module, failing loudly if it is missing on a
    GPU host."""
    if hip_ops is None:
        if torch.cuda.is_available():
            raise RuntimeError(
                "mdi_llm_amd._hip_ops is not built but a GPU is present. "
                "Build it with `python setup.py build_ext --inplace` "
                f"(import error: {_import_error})"
            )
        raise RuntimeError(
            "mdi_llm_amd._hip_ops is not built (CPU-only host): the HIP "
            "decode engine is unavailable here; use the torch path."
        )
    return hip_ops
