#!/usr/bin/env python3
"""Weight download CLI (thin wrapper over the hub downloader).

Capability parity with /root/reference/src/download_weights.py.
"""

import argparse
import sys
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))

if __name__ == "__main__":
    p = argparse.ArgumentParser(description="Download model from HF hub")
    p.add_argument("repo_id", type=str, help="org/model")
    p.add_argument("--checkpoints-root", type=Path,
                   default=SCRIPT_DIR / "checkpoints")
    p.add_argument("--access-token", type=str, default=None)
    p.add_argument("--tokenizer-only", action="store_true")
    p.add_argument("--no-convert", action="store_true")
    args = p.parse_args()

    from mdi_llm_amd.utils.download import download_from_hub

    path = download_from_hub(
        args.repo_id,
        args.checkpoints_root,
        access_token=args.access_token,
        tokenizer_only=args.tokenizer_only,
        convert=not args.no_convert,
    )
    print(f"[download] -> {path}")
