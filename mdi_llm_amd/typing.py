"""Shared type aliases (reference sub/typing.py parity)."""

from pathlib import Path
from typing import Union

FileType = Union[str, Path]
