"""Console UX helpers (reference utils/utils.py:133-172 parity):
a loading bar and a background "waiting" spinner."""

from __future__ import annotations

import itertools
import sys
import threading
import time

__all__ = ["loading_bar", "waiting_animation", "setup_debug_logging", "get_logger"]


def loading_bar(current: int, total: int, width: int = 30,
                prefix: str = "") -> str:
    """Return (and print) a textual progress bar."""
    frac = 0 if total <= 0 else min(max(current / total, 0.0), 1.0)
    filled = int(width * frac)
    bar = f"{prefix}[{'#' * filled}{'.' * (width - filled)}] {frac*100:5.1f}%"
    sys.stderr.write("\r" + bar)
    sys.stderr.flush()
    if current >= total:
        sys.stderr.write("\n")
    return bar


def waiting_animation(stop_event: threading.Event, message: str = "working",
                      interval: float = 0.2) -> threading.Thread:
    """Spawn a daemon spinner thread until ``stop_event`` is set."""

    def _spin():
        for ch in itertools.cycle("|/-\\"):
            if stop_event.is_set():
                sys.stderr.write("\r" + " " * (len(message) + 4) + "\r")
                sys.stderr.flush()
                return
            sys.stderr.write(f"\r{message} {ch}")
            sys.stderr.flush()
            time.sleep(interval)

    t = threading.Thread(target=_spin, daemon=True)
    t.start()
    return t


def setup_debug_logging(role: str, log_dir="logs") -> "logging.Logger":
    """Wire the named framework logger to a per-role debug log file
    (reference starter.py:36-44 / secondary.py:29-38 parity: logger
    "model_dist", DEBUG level, timestamped file handler under logs/).

    Returns the logger; CLIs call this under ``--debug``."""
    import logging
    from pathlib import Path

    log = logging.getLogger("model_dist")
    Path(log_dir).mkdir(parents=True, exist_ok=True)
    handler = logging.FileHandler(
        Path(log_dir) / f"logs_{role}.log", mode="w")
    handler.setFormatter(
        logging.Formatter("[%(asctime)s] -> %(levelname)s: %(message)s"))
    log.setLevel(logging.DEBUG)
    log.addHandler(handler)
    return log


def get_logger() -> "logging.Logger":
    import logging

    return logging.getLogger("model_dist")
