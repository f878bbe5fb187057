"""Bounded-RAM checkpoint writing: stream tensors into a
``torch.save``-compatible zip one at a time.

The reference converts an 8B model on an 8 GB device via lazy tensors +
an incremental pickler (/root/reference/src/sub/utils/litgpt_utils.py:
14-343).  This is the MI355X-framework equivalent with a different
mechanism: tensors are appended straight into the PyTorch zip container
(``torch._C.PyTorchFileWriter`` — the same writer ``torch.save`` uses)
as they are produced and immediately freed; the pickle that indexes them
is synthesized at ``close()`` from recorded metadata, so peak RSS is one
tensor, not the model.  The output loads with plain ``torch.load``
(including ``mmap=True``), byte-layout-compatible with the litGPT
``lit_model.pth`` files the chunker consumes.
"""

from __future__ import annotations

import io
import pickle
from pathlib import Path
from typing import Union

import torch

__all__ = ["IncrementalSaver"]


class _StorageStub:
    """Placeholder pickled in place of a tensor's storage: persistent_id
    resolves it to the zip record written earlier."""

    def __init__(self, key: str, dtype: torch.dtype, numel: int) -> None:
        self.key = key
        self.dtype = dtype
        self.numel = numel


class _TensorStub:
    """Reduces exactly like a saved torch.Tensor, but references a
    _StorageStub instead of holding data."""

    def __init__(self, storage: _StorageStub, shape, stride) -> None:
        self.storage = storage
        self.shape = tuple(shape)
        self.stride = tuple(stride)

    def __reduce_ex__(self, protocol):
        from collections import OrderedDict

        return (
            torch._utils._rebuild_tensor_v2,
            (self.storage, 0, self.shape, self.stride, False, OrderedDict()),
        )


class _Pickler(pickle.Pickler):
    def persistent_id(self, obj):
        if isinstance(obj, _StorageStub):
            stype = getattr(
                torch, torch.storage._dtype_to_storage_type_map()[obj.dtype]
            )
            return ("storage", stype, obj.key, "cpu", obj.numel)
        return None


class IncrementalSaver:
    """``with IncrementalSaver(path) as s: s.add(name, tensor)`` — each
    tensor's bytes go to disk immediately; nothing accumulates."""

    def __init__(self, path: Union[str, Path]) -> None:
        self.path = Path(path)
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self._zip = torch._C.PyTorchFileWriter(str(self.path))
        self._stubs: dict = {}
        self._next = 0

    def add(self, name: str, tensor: torch.Tensor) -> None:
        t = tensor.detach().cpu().contiguous()
        storage = t.untyped_storage()
        key = str(self._next)
        self._next += 1
        self._zip.write_record(f"data/{key}", storage, storage.nbytes())
        self._stubs[name] = _TensorStub(
            _StorageStub(key, t.dtype, t.numel()), t.shape, t.stride()
        )
        del t, storage  # the bytes are in the zip; free now

    def close(self) -> None:
        buf = io.BytesIO()
        _Pickler(buf, protocol=2).dump(self._stubs)
        data = buf.getvalue()
        self._zip.write_record("data.pkl", data, len(data))
        self._zip.write_end_of_file()

    def __enter__(self) -> "IncrementalSaver":
        return self

    def __exit__(self, *exc) -> None:
        if exc[0] is None:
            self.close()
