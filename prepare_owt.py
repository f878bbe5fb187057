#!/usr/bin/env python3
"""OpenWebText-style corpus preparation CLI.

Capability parity with /root/reference/src/prepare_owt.py: tokenize a large
corpus into train/val .bin memmaps.  Sources: a local text/jsonl file, a
directory of text files, or a HF `datasets` dataset name (requires network
or a local datasets cache).
"""

import argparse
import sys
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))

import numpy as np


def iter_texts(args):
    src = args.source
    p = Path(src)
    if p.is_file():
        if p.suffix == ".jsonl":
            import json

            with open(p, encoding="utf-8") as fp:
                for line in fp:
                    yield json.loads(line).get(args.text_key, "")
        else:
            yield p.read_text(encoding="utf-8")
    elif p.is_dir():
        for f in sorted(p.glob("**/*.txt")):
            yield f.read_text(encoding="utf-8")
    else:
        import datasets  # HF datasets (needs network or local cache)

        ds = datasets.load_dataset(src, split=args.split)
        for row in ds:
            yield row.get(args.text_key, "")


def main(args):
    from mdi_llm_amd.tokenizer import Tokenizer
    from mdi_llm_amd.utils.console import loading_bar

    tok = Tokenizer(args.tokenizer_dir)
    out = args.out_dir
    out.mkdir(parents=True, exist_ok=True)
    eos = [tok.eos_id] if tok.eos_id is not None else []
    chunks, total = [], 0
    for i, text in enumerate(iter_texts(args)):
        ids = tok.encode(text, bos=False).tolist() + eos
        chunks.append(np.asarray(ids, dtype=np.uint32))
        total += len(ids)
        if i % 50 == 0:
            loading_bar(i % 1000, 1000, prefix=f"{total} tokens ")
    arr = np.concatenate(chunks) if chunks else np.zeros(0, dtype=np.uint32)
    dtype = np.uint16 if (arr.size and arr.max() < 2 ** 16) else np.uint32
    arr = arr.astype(dtype)
    n = int(len(arr) * args.train_frac)
    arr[:n].tofile(out / "train.bin")
    arr[n:].tofile(out / "val.bin")
    print(f"\n[prepare-owt] {len(arr)} tokens -> {out}/train.bin ({n}) + "
          f"val.bin ({len(arr)-n}), dtype {dtype.__name__}")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--source", required=True,
                   help="text/jsonl file, directory of .txt, or HF dataset")
    p.add_argument("--tokenizer-dir", type=Path, required=True)
    p.add_argument("--out-dir", type=Path, default=Path("data/owt"))
    p.add_argument("--split", default="train")
    p.add_argument("--text-key", default="text")
    p.add_argument("--train-frac", type=float, default=0.9)
    main(p.parse_args())
