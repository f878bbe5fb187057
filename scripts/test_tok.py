#!/usr/bin/env python3
"""Tokenizer quick-check (parity: /root/reference/src/scripts/test_tok.py)."""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mdi_llm_amd.tokenizer import Tokenizer  # noqa: E402

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("ckpt", type=Path)
    p.add_argument("--text", default="Hello, world!")
    args = p.parse_args()
    tok = Tokenizer(args.ckpt)
    print(f"backend: {tok.backend}  vocab: {tok.vocab_size}")
    print(f"bos: {tok.bos_id}  eos: {tok.eos_id}  use_bos: {tok.use_bos}")
    ids = tok.encode(args.text)
    print(f"encode({args.text!r}) = {ids.tolist()}")
    print(f"decode = {tok.decode(ids)!r}")
