"""tokens-vs-time plotting with the reference's file conventions.

Capability parity with /root/reference/src/sub/utils/plots.py:12-51 and
plot_tok_time.py: CSV name pattern
``tokens_time_samples_<N>nodes_<model>_<S>samples.csv`` is the cross-run
join key for overlaying 1/2/.../8-stage curves.
"""

from __future__ import annotations

import csv
import re
from pathlib import Path
from typing import Iterable, List, Tuple, Union

__all__ = ["write_tok_time_csv", "tok_time_csv_name", "plot_tokens_per_time",
           "collect_csv_runs"]

PathLike = Union[str, Path]


def tok_time_csv_name(n_nodes: int, model_name: str, n_samples: int) -> str:
    return f"tokens_time_samples_{n_nodes}nodes_{model_name}_{n_samples}samples.csv"


def write_tok_time_csv(path: PathLike, tok_time: Iterable[Tuple[int, float]]):
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    with open(path, "w", newline="") as fp:
        w = csv.writer(fp)
        w.writerow(["tokens", "time"])
        for n, t in tok_time:
            w.writerow([n, f"{t:.6f}"])
    return path


def collect_csv_runs(logs_dir: PathLike, model_name: str) -> List[Path]:
    pat = re.compile(
        rf"tokens_time_samples_(\d+)nodes_{re.escape(model_name)}_(\d+)samples\.csv"
    )
    return sorted(
        p for p in Path(logs_dir).glob("*.csv") if pat.fullmatch(p.name)
    )


def plot_tokens_per_time(csv_paths: List[PathLike], out_png: PathLike,
                         model_name: str = "") -> Path:
    """Overlay time-vs-tokens curves of multiple runs (one per stage count),
    the reference's published benchmark figure format."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots(figsize=(8, 5))
    for path in csv_paths:
        path = Path(path)
        m = re.search(r"_(\d+)nodes_", path.name)
        label = f"{m.group(1)} node(s)" if m else path.stem
        xs, ys = [], []
        with open(path) as fp:
            r = csv.reader(fp)
            next(r)
            for tokens, t in r:
                xs.append(float(t))
                ys.append(int(tokens))
        ax.plot(xs, ys, label=label)
    ax.set_xlabel("time (s)")
    ax.set_ylabel("cumulative generated tokens")
    ax.set_title(f"Generation: tokens vs time — {model_name}")
    ax.grid(True, alpha=0.3)
    ax.legend()
    out_png = Path(out_png)
    out_png.parent.mkdir(parents=True, exist_ok=True)
    fig.savefig(out_png, dpi=120, bbox_inches="tight")
    plt.close(fig)
    return out_png
