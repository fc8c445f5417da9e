"""Gradient-tree algebra — MI355X re-expression of the reference's L4 layer.

The reference walks a Functors.jl model tree (`destruct` at
/root/reference/src/ddp_tasks.jl:22-26, `_accum`/`_dodiv` at
src/overloads.jl:43-54, buffer protocol at src/ddp_tasks.jl:59-109).
Here a "gradient tree" is an ordered dict {param_name: Tensor|None}: PyTorch
already names every leaf, so the typed recursion collapses to dict walks.
`None` plays the role of Julia `nothing` (no gradient / non-diff leaf).

These functions are the oracle surface for the distributed==batched tests
(reference test/single_device.jl:6-36) and the engine of the task-DDP path.
"""

from collections import OrderedDict
from typing import Dict, Iterable, Optional

import torch

GradTree = Dict[str, Optional[torch.Tensor]]


def destruct(model: torch.nn.Module, device=None) -> GradTree:
    """Zero-initialized gradient skeleton of `model` (ddp_tasks.jl:22-26)."""
    out: GradTree = OrderedDict()
    for name, p in model.named_parameters():
        if p.requires_grad:
            t = torch.zeros_like(p.data, memory_format=torch.preserve_format)
            out[name] = t.to(device) if device is not None else t
        else:
            out[name] = None
    return out


def grads_of(model: torch.nn.Module) -> GradTree:
    """Current .grad tree of a model (post-backward)."""
    return OrderedDict(
        (name, p.grad) for name, p in model.named_parameters() if p.requires_grad
    )


def accum_(dst: GradTree, src: GradTree) -> GradTree:
    """dst += src leaf-wise, in place; None behaves like Zygote's `nothing`
    (overloads.jl:43-46: accum(nothing, x) = x, accum(x, nothing) = x)."""
    for k, s in src.items():
        if s is None:
            continue
        d = dst.get(k)
        if d is None:
            dst[k] = s.clone()
        else:
            d.add_(s.to(d.device, non_blocking=True) if s.device != d.device else s)
    return dst


def dodiv_(tree: GradTree, n: float) -> GradTree:
    """tree /= n leaf-wise in place (overloads.jl:48-54)."""
    for v in tree.values():
        if v is not None:
            v.div_(n)
    return tree


def markbuffer_(dest: GradTree, src: GradTree) -> GradTree:
    """Publish a device's grads into its buffer slot (ddp_tasks.jl:65-71).

    Cross-device leaf-wise copy; on HIP this is an async P2P copyto over
    xGMI, ordered on the current stream.
    """
    for k, s in src.items():
        if s is None:
            continue
        d = dest.get(k)
        if d is None:
            dest[k] = s.detach().clone()
        else:
            d.copy_(s, non_blocking=True)
    return dest


def getbuffer_(dest: GradTree, src: GradTree) -> GradTree:
    """Copy reduced grads back into a device's grad memory (ddp_tasks.jl:73-78)."""
    return markbuffer_(dest, src)


def sync_buffer(buffer: Dict, average: bool = True) -> GradTree:
    """Reduce the per-device buffer dict {dev: GradTree} to the mean tree
    (ddp_tasks.jl:93-109). The reduction runs on the device holding the
    first entry (the reference's HOST GPU); result left there.

    On the RCCL path this whole function is replaced by a bucketed
    all-reduce (see process_ddp) — kept here for the task path and the
    CPU oracle tests.
    """
    trees = list(buffer.values())
    if not trees:
        return OrderedDict()
    final: GradTree = OrderedDict()
    first = trees[0]
    for k, v in first.items():
        final[k] = None if v is None else v.clone()
    for t in trees[1:]:
        accum_(final, t)
    if average:
        dodiv_(final, float(len(trees)))
    return final


def ensure_synced(trees: Iterable[GradTree], rtol=1e-4, atol=1e-5) -> bool:
    """Replica-consistency checker (ddp_tasks.jl:115-126)."""
    trees = list(trees)
    if len(trees) < 2:
        return True
    ref = trees[0]
    for t in trees[1:]:
        for k, v in ref.items():
            o = t.get(k)
            if (v is None) != (o is None):
                return False
            if v is not None and not torch.allclose(
                v.float().cpu(), o.float().cpu(), rtol=rtol, atol=atol
            ):
                return False
    return True


def check_nans(tree: GradTree) -> Dict[str, bool]:
    """NaN scan (ddp_tasks.jl:86-91); returns {name: has_nan} for bad leaves."""
    bad = {}
    for k, v in tree.items():
        if v is not None and not torch.isfinite(v).all():
            bad[k] = True
    return bad


def show_stats(tree: GradTree, name: str = "grads") -> str:
    """Debug dump of per-leaf mean/std/min/max (the reference's
    `_show_stats`, /root/reference/src/overloads.jl:56-59)."""
    lines = [name]
    for k, v in tree.items():
        if v is None:
            lines.append(f"  {k}: nothing")
        else:
            f = v.detach().float()
            lines.append(
                f"  {k}: shape={tuple(v.shape)} mean={f.mean():.4e} "
                f"std={f.std():.4e} min={f.min():.4e} max={f.max():.4e}")
    return "\n".join(lines)
