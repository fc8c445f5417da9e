"""Metrics — top-k accuracy (reference /root/reference/src/utils.jl:20-45:
`maxk!`/`kacc`/`topkaccuracy` via partial sort)."""

from typing import Sequence

import torch


def maxk(logits: torch.Tensor, k: int) -> torch.Tensor:
    """Indices of the k largest entries per column-sample (reference maxk,
    utils.jl:28-33). Here rows are samples: returns (N, k) int64."""
    return torch.topk(logits, k, dim=-1).indices


def topkaccuracy(logits: torch.Tensor, target: torch.Tensor, k: int = 1) -> float:
    """Fraction of samples whose true class is within the top-k predictions
    (reference kacc/topkaccuracy, utils.jl:35-45). `target` is class
    indices (N,) or one-hot (N, C)."""
    if target.dim() == 2:
        target = target.argmax(dim=-1)
    k = min(k, logits.shape[-1])
    top = maxk(logits, k)
    return float((top == target.unsqueeze(-1)).any(dim=-1).float().mean())


def showpreds(logits: torch.Tensor, target: torch.Tensor,
              class_names: Sequence[str] = None, k: int = 3) -> str:
    """Pretty-print top-k predictions per sample (reference showpreds,
    utils.jl:47-71)."""
    probs = torch.softmax(logits.float(), dim=-1)
    vals, idx = torch.topk(probs, min(k, logits.shape[-1]), dim=-1)
    if target.dim() == 2:
        target = target.argmax(dim=-1)
    lines = []
    for i in range(logits.shape[0]):
        names = [
            (class_names[j] if class_names is not None else str(int(j)))
            for j in idx[i]
        ]
        tgt = class_names[target[i]] if class_names is not None else str(int(target[i]))
        preds = ", ".join(f"{n}:{float(v):.3f}" for n, v in zip(names, vals[i]))
        lines.append(f"[{i}] true={tgt}  pred=({preds})")
    return "\n".join(lines)
