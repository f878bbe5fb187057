#!/usr/bin/env python3
"""Model preparation CLI: download/convert a checkpoint and split it into
pipeline-stage chunks.

Capability parity with /root/reference/src/prepare_model.py: local HF dir ->
litGPT conversion, or hub download, then ``split_and_store`` into
``chunks/<N>nodes/``.
"""

import argparse
import sys
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))


def main(args):
    from mdi_llm_amd.utils.checkpoint import get_checkpoint_files, load_from_pt
    from mdi_llm_amd.utils.partition import split_and_store

    ckpt = args.ckpt
    cfg_file, model_file = get_checkpoint_files(ckpt)
    if not model_file.is_file():
        if ckpt.is_dir() and any(ckpt.glob("*.safetensors")) or \
                (ckpt / "pytorch_model.bin").is_file() or \
                any(ckpt.glob("*.bin.index.json")) or \
                any(ckpt.glob("*.safetensors.index.json")):
            from mdi_llm_amd.utils.convert_hf import convert_hf_checkpoint

            print(f"[prepare] converting HF checkpoint {ckpt}")
            convert_hf_checkpoint(ckpt, model_name=args.model_name)
        elif args.download:
            from mdi_llm_amd.utils.download import download_from_hub

            print(f"[prepare] downloading {args.download}")
            ckpt = download_from_hub(args.download, args.ckpt.parent.parent)
        else:
            raise FileNotFoundError(
                f"{model_file} missing and no HF files in {ckpt}; "
                "pass --download org/model to fetch"
            )

    if args.n_nodes > 1:
        config, sd = load_from_pt(ckpt)
        out = split_and_store(sd, args.n_nodes, ckpt, config.n_layer)
        print(f"[prepare] wrote {args.n_nodes}-node chunks -> {out}")
    else:
        print("[prepare] n_nodes=1: nothing to split")


if __name__ == "__main__":
    p = argparse.ArgumentParser(description="Prepare/partition a model")
    p.add_argument("--ckpt", type=Path, required=True,
                   help="checkpoint dir (litGPT or HF layout)")
    p.add_argument("--n-nodes", type=int, default=2)
    p.add_argument("--model-name", type=str, default=None,
                   help="registry name override when the dir name differs")
    p.add_argument("--download", type=str, default=None,
                   help="HF repo id to download (org/model)")
    main(p.parse_args())
