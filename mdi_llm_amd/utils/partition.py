"""Layer partitioning for pipeline stages + chunked-checkpoint IO.

Capability parity with the reference's split machinery:
``N_LAYERS_NODES`` (/root/reference/src/sub/config.py:56-98),
``split_parameters`` (utils/utils.py:241-385) and ``split_and_store``
(utils/utils.py:388-438).  The on-disk chunk format is kept byte-compatible:
``<ckpt>/chunks/<N>nodes/model_starter.pth`` + ``model_secondary<i>.pth``,
starter chunk holding ``transformer.wte.* , transformer.h.0..k-1,
transformer.ln_f.*, lm_head.*`` and each secondary chunk holding its blocks
re-indexed from ``transformer.h.0``.

Beyond the reference's fixed table we provide a general near-even formula
for any (n_layer, n_nodes) — required because 8-stage Llama-3-8B (32 layers)
is a target config the table does not cover; table entries are preserved for
the sizes the reference lists.
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Dict, List, Union

import torch

__all__ = [
    "N_LAYERS_NODES",
    "layer_split",
    "split_parameters",
    "split_and_store",
    "chunk_dir",
    "chunk_file_name",
    "count_transformer_blocks",
]

# Fixed split table, value-compatible with the reference
# (/root/reference/src/sub/config.py:56-98): starter gets fewer blocks
# because it also runs wte + ln_f + lm_head.
N_LAYERS_NODES: Dict[int, Dict[int, dict]] = {
    1: {n: {"N_LAYERS_START": n} for n in (5, 7, 9, 12, 22, 24, 32, 36, 48)},
    2: {
        5: {"N_LAYERS_START": 2, "N_LAYERS_SECONDARY": 3},
        7: {"N_LAYERS_START": 3, "N_LAYERS_SECONDARY": 4},
        9: {"N_LAYERS_START": 4, "N_LAYERS_SECONDARY": 5},
        12: {"N_LAYERS_START": 5, "N_LAYERS_SECONDARY": 7},
        22: {"N_LAYERS_START": 10, "N_LAYERS_SECONDARY": 12},
        24: {"N_LAYERS_START": 10, "N_LAYERS_SECONDARY": 14},
        32: {"N_LAYERS_START": 14, "N_LAYERS_SECONDARY": 18},
        36: {"N_LAYERS_START": 16, "N_LAYERS_SECONDARY": 20},
        48: {"N_LAYERS_START": 22, "N_LAYERS_SECONDARY": 26},
    },
    3: {
        5: {"N_LAYERS_START": 1, "N_LAYERS_SECONDARY": 2},
        7: {"N_LAYERS_START": 1, "N_LAYERS_SECONDARY": 3},
        9: {"N_LAYERS_START": 1, "N_LAYERS_SECONDARY": 4},
        12: {"N_LAYERS_START": 2, "N_LAYERS_SECONDARY": 5},
        22: {"N_LAYERS_START": 6, "N_LAYERS_SECONDARY": 8},
        24: {"N_LAYERS_START": 4, "N_LAYERS_SECONDARY": 10},
        32: {"N_LAYERS_START": 8, "N_LAYERS_SECONDARY": 12},
        36: {"N_LAYERS_START": 10, "N_LAYERS_SECONDARY": 13},
        48: {"N_LAYERS_START": 14, "N_LAYERS_SECONDARY": 17},
    },
    4: {
        22: {"N_LAYERS_START": 4, "N_LAYERS_SECONDARY": 6},
        32: {"N_LAYERS_START": 5, "N_LAYERS_SECONDARY": 9},
    },
    5: {
        22: {"N_LAYERS_START": 2, "N_LAYERS_SECONDARY": 5},
        32: {"N_LAYERS_START": 4, "N_LAYERS_SECONDARY": 7},
    },
}


def layer_split(n_layer: int, n_nodes: int) -> List[int]:
    """Per-stage block counts ``[starter, sec1, ..]`` summing to n_layer.

    Uses the reference-compatible table when the (n_nodes, n_layer) entry
    exists; otherwise a near-even formula that hands the starter the
    smallest share (it additionally runs embedding + ln_f + lm_head and the
    sampler).
    """
    if n_nodes < 1:
        raise ValueError("n_nodes must be >= 1")
    if n_nodes == 1:
        return [n_layer]
    entry = N_LAYERS_NODES.get(n_nodes, {}).get(n_layer)
    if entry is not None:
        start = entry["N_LAYERS_START"]
        sec = entry["N_LAYERS_SECONDARY"]
        counts = [start] + [sec] * (n_nodes - 1)
        # table rows may not be exactly divisible; give remainder to last
        counts[-1] += n_layer - sum(counts)
        return counts
    if n_layer < n_nodes:
        raise ValueError(f"cannot split {n_layer} layers over {n_nodes} nodes")
    base, rem = divmod(n_layer, n_nodes)
    # secondaries absorb the remainder first; starter keeps the base share
    counts = [base] * n_nodes
    i = n_nodes - 1
    while rem > 0:
        counts[i] += 1
        rem -= 1
        i = i - 1 if i > 1 else n_nodes - 1
    return counts


def balanced_split(config, n_nodes: int, sampler_bytes: int = 150_000_000,
                   dtype_bytes: int = 2) -> List[int]:
    """Byte-balanced per-stage block counts for a given model config.

    Decode is HBM-bandwidth-bound, so per-stage weight BYTES are a direct
    proxy for per-stage time.  The starter additionally streams the lm_head
    every token and runs the sampler, so it gets fewer blocks — the general
    version of the reference's hand-tuned table intuition
    (/root/reference/src/sub/config.py:56-98, README.md:334-342).
    """
    if n_nodes == 1:
        return [config.n_layer]
    E, I, V = config.n_embd, config.intermediate_size, config.padded_vocab_size
    hs, nh, ng = config.head_size, config.n_head, config.n_query_groups
    layer_b = (E * (nh + 2 * ng) * hs + E * nh * hs + 3 * E * I) * dtype_bytes
    if config.mlp_class_name == "GptNeoxMLP":
        layer_b = (E * (nh + 2 * ng) * hs + E * nh * hs + 2 * E * I) * dtype_bytes
    starter_extra = V * E * dtype_bytes + sampler_bytes
    extra_layers = starter_extra / max(layer_b, 1)
    # starter share: solve x + extra = (n_layer - x) / (n_nodes - 1) balance
    x = (config.n_layer - extra_layers * (n_nodes - 1)) / n_nodes
    start = max(0, min(config.n_layer - (n_nodes - 1), round(x)))
    rest = config.n_layer - start
    base, rem = divmod(rest, n_nodes - 1)
    counts = [start] + [base + (1 if i < rem else 0)
                        for i in range(n_nodes - 1)]
    # every secondary needs at least one block
    for i in range(1, n_nodes):
        if counts[i] == 0:
            counts[i] = 1
            counts[0] -= 1
    assert sum(counts) == config.n_layer and all(c >= 0 for c in counts)
    return counts


def count_transformer_blocks(state_dict: dict) -> int:
    """Number of distinct ``transformer.h.<i>.`` indices in a state dict
    (reference utils/utils.py:470-492)."""
    seen = set()
    for k in state_dict:
        if k.startswith("transformer.h."):
            seen.add(int(k.split(".")[2]))
    return len(seen)


def split_parameters(
    state_dict: dict, n_nodes: int, n_layer: int | None = None
) -> List[dict]:
    """Split a full-model litGPT state dict into per-stage chunks.

    Chunk 0 (starter): ``transformer.wte.*``, (optional ``transformer.wpe.*``),
    ``transformer.h.0..k-1``, ``transformer.ln_f.*``, ``lm_head.*``.
    Chunk i>0: its block range re-indexed from ``transformer.h.0``
    (reference utils/utils.py:241-385).
    """
    if n_layer is None:
        n_layer = count_transformer_blocks(state_dict)
    counts = layer_split(n_layer, n_nodes)
    chunks: List[dict] = [dict() for _ in range(n_nodes)]

    for key, tensor in state_dict.items():
        if key.startswith("transformer.h."):
            parts = key.split(".")
            idx = int(parts[2])
            # find owning stage
            lo = 0
            for stage, cnt in enumerate(counts):
                if lo <= idx < lo + cnt:
                    parts[2] = str(idx - lo)
                    chunks[stage][".".join(parts)] = tensor
                    break
                lo += cnt
        elif key.startswith(("transformer.wte.", "transformer.wpe.",
                             "transformer.ln_f.", "lm_head.")):
            chunks[0][key] = tensor
        else:
            chunks[0][key] = tensor
    return chunks


def chunk_dir(ckpt_dir: Union[str, Path], n_nodes: int) -> Path:
    return Path(ckpt_dir) / "chunks" / f"{n_nodes}nodes"


def chunk_file_name(stage: int) -> str:
    return "model_starter.pth" if stage == 0 else f"model_secondary{stage - 1}.pth"


def split_and_store(
    state_dict: dict,
    n_nodes: int,
    ckpt_dir: Union[str, Path],
    n_layer: int | None = None,
) -> Path:
    """Write per-stage chunk files in the reference's on-disk layout
    (utils/utils.py:388-438) and return the chunk directory."""
    out = chunk_dir(ckpt_dir, n_nodes)
    out.mkdir(parents=True, exist_ok=True)
    chunks = split_parameters(state_dict, n_nodes, n_layer)
    for stage, chunk in enumerate(chunks):
        torch.save(chunk, out / chunk_file_name(stage))
    meta = {
        "n_nodes": n_nodes,
        "n_layer": n_layer or count_transformer_blocks(state_dict),
        "layer_split": layer_split(
            n_layer or count_transformer_blocks(state_dict), n_nodes
        ),
    }
    (out / "split_meta.json").write_text(json.dumps(meta, indent=2))
    return out
