#!/usr/bin/env python3
"""Training-data preparation CLI: tokenize a text file into train/val .bin.

Capability parity with /root/reference/src/prepare_data.py (and the
prepare_owt.py flow for arbitrary text corpora).
"""

import argparse
import sys
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))

if __name__ == "__main__":
    p = argparse.ArgumentParser(description="Tokenize text into .bin")
    p.add_argument("--input", type=Path, required=True,
                   help="raw text file (e.g. tiny-shakespeare)")
    p.add_argument("--tokenizer-dir", type=Path, required=True,
                   help="checkpoint dir containing tokenizer files")
    p.add_argument("--out-dir", type=Path, default=None)
    p.add_argument("--train-frac", type=float, default=0.9)
    args = p.parse_args()

    from mdi_llm_amd.tokenizer import Tokenizer
    from mdi_llm_amd.utils.data import prepare_bin

    out = args.out_dir or args.input.parent
    tok = Tokenizer(args.tokenizer_dir)
    text = args.input.read_text(encoding="utf-8")
    train_p, val_p = prepare_bin(text, tok, out, args.train_frac)
    print(f"[prepare-data] {train_p} + {val_p}")
