"""Concurrent-replica stem-conv correctness (round-1 advisor finding #1).

Task-DDP runs a thread per logical replica; two same-shape stems on one
device used to SHARE the cached padded-input buffer, so the interleaving
fwd(A), fwd(B) -> bwd(A), bwd(B) computed both stem wgrads from replica
B's input. The pool checkout in ops/conv.py gives each forward exclusive
ownership; this test drives exactly that interleaving on the native bf16
stem path and compares every replica's gradient to a solo oracle.

Reference contract: /root/reference/test/single_device.jl:127-133 (logical
fan-out of N replicas on one physical GPU).
"""

import threading

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("GPU-only suite", allow_module_level=True)


def _make_model(seed):
    from fluxdistributed_amd.models import build_model
    from fluxdistributed_amd.utils.precision import to_mixed_bf16

    torch.manual_seed(seed)
    m = build_model("resnet18", num_classes=16, small_input=True)
    return to_mixed_bf16(m.to("cuda:0").to(memory_format=torch.channels_last)).train()


def _batches(n):
    out = []
    for i in range(n):
        g = torch.Generator().manual_seed(500 + i)
        x = torch.randn(4, 3, 32, 32, generator=g).bfloat16().cuda() \
            .contiguous(memory_format=torch.channels_last)
        y = torch.randint(0, 16, (4,), generator=g).cuda()
        out.append((x, y))
    return out


def _grads(model):
    return {n: p.grad.detach().float().clone() for n, p in model.named_parameters()
            if p.grad is not None}


def test_threaded_replicas_match_solo_oracle():
    from fluxdistributed_amd.ops import logit_cross_entropy

    N = 2
    batches = _batches(N)

    # solo oracle: each batch through its own model, sequentially
    oracle = []
    for i, (x, y) in enumerate(batches):
        m = _make_model(7)  # identical weights across replicas
        loss = logit_cross_entropy(m(x), y)
        m.zero_grad(set_to_none=False)
        loss.backward()
        torch.cuda.synchronize()
        oracle.append(_grads(m))

    # threaded replicas with a barrier BETWEEN forward and backward:
    # both forwards complete before either backward starts — the exact
    # interleaving that corrupted the shared stem buffer.
    models = [_make_model(7) for _ in range(N)]
    barrier = threading.Barrier(N)
    errors = [None] * N

    def run(i):
        try:
            x, y = batches[i]
            m = models[i]
            loss = logit_cross_entropy(m(x), y)
            barrier.wait(timeout=60)
            m.zero_grad(set_to_none=False)
            loss.backward()
            torch.cuda.synchronize()
        except BaseException as e:  # noqa: BLE001
            errors[i] = e

    threads = [threading.Thread(target=run, args=(i,)) for i in range(N)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert all(e is None for e in errors), errors

    stem_checked = False
    for i, m in enumerate(models):
        got = _grads(m)
        assert got.keys() == oracle[i].keys()
        for name in got:
            assert torch.allclose(got[name], oracle[i][name], rtol=1e-2, atol=1e-3), (
                f"replica {i} grad mismatch in {name}: "
                f"max|d|={float((got[name] - oracle[i][name]).abs().max())}")
            if "conv1" in name and name.startswith("conv"):
                stem_checked = True
    assert stem_checked or any("conv1.weight" in k for k in oracle[0])


def test_stem_requires_grad_input_falls_back():
    """A stem-shaped conv whose INPUT needs grad must not silently return
    no dx (advisor finding #3): dispatch routes it to the library."""
    from fluxdistributed_amd.ops.conv import fda_conv2d

    x = torch.randn(2, 3, 32, 32, device="cuda").bfloat16() \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    w = torch.randn(64, 3, 3, 3, device="cuda").bfloat16() \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = fda_conv2d(x, w, stride=(1, 1), padding=(1, 1))
    y.float().sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad.float()).all()
    assert w.grad is not None
