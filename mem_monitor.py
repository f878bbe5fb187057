#!/usr/bin/env python3
"""Memory monitor CLI: run a command while sampling RSS + GPU memory.

Capability parity with /root/reference/src/mem_monitor.py (GPU side via
rocm-smi instead of GPUtil/jtop).
"""

import argparse
import sys
from pathlib import Path

SCRIPT_DIR = Path(__file__).resolve().parent
sys.path.insert(0, str(SCRIPT_DIR))

if __name__ == "__main__":
    p = argparse.ArgumentParser(description="Monitor memory of a command")
    p.add_argument("--out", type=Path,
                   default=SCRIPT_DIR / "logs" / "mem_monitor.csv")
    p.add_argument("--interval", type=float, default=0.5)
    p.add_argument("cmd", nargs=argparse.REMAINDER,
                   help="command to run (after --)")
    args = p.parse_args()
    cmd = [c for c in args.cmd if c != "--"]
    if not cmd:
        p.error("no command given")

    from mdi_llm_amd.utils.monitor import monitor_command

    rc = monitor_command(cmd, args.out, args.interval)
    print(f"[mem-monitor] exit={rc}, samples -> {args.out}")
    sys.exit(rc)
