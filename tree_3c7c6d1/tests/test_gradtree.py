"""The reference's central oracle: distributed == batched
(/root/reference/test/single_device.jl:6-36 `check_data_parallel`).

Gradients of sum(m(x)) over a k-sample batch must equal the accum_-fold of
per-sample gradients; norm layers run in eval mode (the reference runs all
comparisons under testmode! — single_device.jl:51-57 comment).
"""

import copy

import pytest
import torch
import torch.nn as nn

from fluxdistributed_amd.parallel.gradtree import (
    destruct, grads_of, accum_, dodiv_, sync_buffer, markbuffer_,
    getbuffer_, ensure_synced, check_nans,
)
from fluxdistributed_amd.models import resnet18
from fluxdistributed_amd.models.resnet import FusedBNAct


def _grad_of(model, x):
    model.zero_grad()
    model(x).sum().backward()
    return {k: v.clone() for k, v in grads_of(model).items() if v is not None}


def _compare(a, b, rtol=1e-4, atol=1e-4):
    # the reference `compare` fixture (test/runtests.jl:6-35)
    assert set(a) == set(b)
    for k in a:
        assert torch.allclose(a[k], b[k], rtol=rtol, atol=atol), k


MODELS = {
    "conv": lambda: nn.Conv2d(3, 4, 3, padding=1),
    "dense": lambda: nn.Sequential(nn.Flatten(), nn.Linear(3 * 8 * 8, 7)),
    "chain_conv_bn": lambda: nn.Sequential(
        nn.Conv2d(3, 4, 3, padding=1), nn.BatchNorm2d(4), nn.ReLU(),
        nn.MaxPool2d(2), nn.Flatten(), nn.Linear(4 * 4 * 4, 5),
    ),
    "fused_bn": lambda: nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), FusedBNAct(8)),
}


@pytest.mark.parametrize("name", sorted(MODELS))
def test_distributed_equals_batched(name, seed):
    model = MODELS[name]()
    model.eval()  # testmode!: norm layers use running stats
    x = torch.randn(3, 3, 8, 8)

    batched = _grad_of(model, x)
    folded = destruct(model)
    for i in range(3):
        g = _grad_of(model, x[i : i + 1])
        accum_(folded, g)
    folded = {k: v for k, v in folded.items() if v is not None}
    _compare(batched, folded)


def test_distributed_equals_batched_resnet18(seed):
    model = resnet18(num_classes=10, small_input=True)
    model.eval()
    x = torch.randn(2, 3, 16, 16)
    batched = _grad_of(model, x)
    folded = destruct(model)
    for i in range(2):
        accum_(folded, _grad_of(model, x[i : i + 1]))
    _compare(batched, {k: v for k, v in folded.items() if v is not None},
             rtol=1e-3, atol=1e-3)


def test_sync_buffer_mean(seed):
    model = nn.Linear(4, 3)
    t1 = {k: torch.randn_like(v) for k, v in destruct(model).items()}
    t2 = {k: torch.randn_like(v) for k, v in destruct(model).items()}
    buf = {0: {k: v.clone() for k, v in t1.items()},
           1: {k: v.clone() for k, v in t2.items()}}
    final = sync_buffer(buf)
    for k in t1:
        assert torch.allclose(final[k], (t1[k] + t2[k]) / 2)


def test_mark_get_roundtrip(seed):
    model = nn.Linear(4, 3)
    src = {k: torch.randn_like(v) for k, v in destruct(model).items()}
    buf = destruct(model)
    markbuffer_(buf, src)
    dst = destruct(model)
    getbuffer_(dst, buf)
    for k in src:
        assert torch.equal(dst[k], src[k])


def test_none_leaves_pass_through():
    a = {"w": torch.ones(2), "frozen": None}
    b = {"w": torch.ones(2), "frozen": None}
    accum_(a, b)
    assert torch.allclose(a["w"], torch.full((2,), 2.0))
    assert a["frozen"] is None
    dodiv_(a, 2.0)
    assert torch.allclose(a["w"], torch.ones(2))


def test_ensure_synced_and_nans():
    t1 = {"w": torch.ones(3)}
    t2 = {"w": torch.ones(3)}
    assert ensure_synced([t1, t2])
    t2["w"][0] = 5.0
    assert not ensure_synced([t1, t2])
    assert check_nans({"w": torch.tensor([1.0, float("nan")])}) == {"w": True}
    assert check_nans(t1) == {}


def test_show_stats():
    from fluxdistributed_amd.parallel.gradtree import show_stats

    tree = {"a": torch.randn(4, 4), "b": None}
    s = show_stats(tree)
    assert "a: shape=(4, 4)" in s and "b: nothing" in s
