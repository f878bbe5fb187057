#!/usr/bin/env python3
"""Secondary (worker) node CLI.

Flag-compatible with the reference worker (/root/reference/src/secondary.py):
--nodes-config CONFIG-PATH SECONDARY-INDEX, --chunk, --device, --dtype,
--seed, -v/--verb.
"""

import argparse
import sys
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))


def main(args):
    import torch

    from mdi_llm_amd.parallel.orchestrator import MDIRuntime

    torch.manual_seed(args.seed)
    idx = int(args.nodes_config[1])
    if getattr(args, "debug", False):
        from mdi_llm_amd.utils.console import setup_debug_logging

        setup_debug_logging(f"secondary{idx}", SCRIPT_DIR / "logs")
    rt = MDIRuntime(
        f"secondary:{idx}",
        config_file=Path(args.nodes_config[0]),
        chunk_path=args.chunk,
        device=args.device,
        dtype=args.dtype,
        verb=args.verb,
    )
    rt.start()


if __name__ == "__main__":
    p = argparse.ArgumentParser(description="Secondary node - MDI (MI355X)")
    p.add_argument("-v", "--verb", action="store_true")
    p.add_argument("-d", "--debug", action="store_true",
                   help="write debug logs to logs/logs_secondary<i>.log")
    p.add_argument("--chunk", type=Path, default=None)
    p.add_argument(
        "--nodes-config",
        type=str,
        nargs=2,
        metavar=("CONFIG-PATH", "SECONDARY-INDEX"),
        default=[str(SCRIPT_DIR / "settings_distr" / "configuration.json"), 0],
    )
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--dtype", type=str, default=None)
    p.add_argument("--seed", type=int, default=10137)
    main(p.parse_args())
