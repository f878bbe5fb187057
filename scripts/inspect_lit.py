#!/usr/bin/env python3
"""Checkpoint sanity inspector.

Capability parity with /root/reference/src/scripts/inspect_lit.py:46-99:
verify the stored transformer-block count matches the config, report dtypes
and serialized sizes, dump key names.
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mdi_llm_amd.utils.checkpoint import load_from_pt  # noqa: E402
from mdi_llm_amd.utils.partition import count_transformer_blocks  # noqa: E402

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("ckpt", type=Path)
    p.add_argument("--keys", action="store_true", help="dump all key names")
    args = p.parse_args()

    config, sd = load_from_pt(args.ckpt)
    n_blocks = count_transformer_blocks(sd)
    total = sum(t.numel() * t.element_size() for t in sd.values())
    dtypes = {str(t.dtype) for t in sd.values()}
    print(f"config:        {config.name}")
    print(f"n_layer:       {config.n_layer} (checkpoint has {n_blocks})")
    print(f"params:        {sum(t.numel() for t in sd.values())/1e6:.1f} M")
    print(f"size on disk:  {total/1e9:.2f} GB")
    print(f"dtypes:        {sorted(dtypes)}")
    if n_blocks != config.n_layer:
        print("MISMATCH: block count != config.n_layer", file=sys.stderr)
        sys.exit(1)
    if args.keys:
        for k, t in sd.items():
            print(f"  {k:60s} {tuple(t.shape)}")
