"""Task-DDP orchestration on logical CPU devices — the reference's CPU
fake-device contract (test/single_device.jl:115-168 "Workflow" testset)."""

import copy

import torch
import torch.nn as nn

from fluxdistributed_amd.parallel.task_ddp import (
    prepare_training, train, train_step, update,
)
from fluxdistributed_amd.parallel.gradtree import sync_buffer, grads_of, ensure_synced
from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy
from fluxdistributed_amd.models import resnet18


def _mlp(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Flatten(), nn.Linear(12, 16), nn.ReLU(), nn.Linear(16, 4))


def _loss(out, y):
    return logit_cross_entropy(out, y)


def test_grad_syncing_in_train(seed):
    """reference test_grad_syncing_in_train (single_device.jl:66-97):
    drive the real train_step -> sync_buffer machinery on shards and
    compare to the batched gradient."""
    model = _mlp()
    x = torch.randn(4, 3, 2, 2)
    y = torch.randint(0, 4, (4,))

    # batched gradient (loss is a mean -> matches the averaged shard fold)
    batched = copy.deepcopy(model)
    out = batched(x)
    _loss(out, y).backward()
    bg = {k: v.clone() for k, v in grads_of(batched).items()}

    st = prepare_training(
        model, None, devices=[0, 1],
        opt_factory=lambda m: FusedSGDMomentum(m.parameters(), lr=0.1),
    )
    for i, r in enumerate(st.replicas):
        sl = slice(2 * i, 2 * i + 2)
        train_step(_loss, st.buffer, r, x[sl], y[sl])
    final = sync_buffer(st.buffer)
    for k, v in bg.items():
        assert torch.allclose(final[k], v, rtol=1e-4, atol=1e-5), k


def test_replicas_stay_in_sync(seed):
    """reference check_distributed_opt (single_device.jl:99-113, 160-167):
    after shared-grad updates all replicas must be identical."""
    model = _mlp()
    st = prepare_training(
        model, None, devices=[0, 1, 2],
        opt_factory=lambda m: FusedSGDMomentum(m.parameters(), lr=0.05, momentum=0.9),
    )

    def batches(j):
        torch.manual_seed(100 + j)
        return [
            (torch.randn(2, 3, 2, 2), torch.randint(0, 4, (2,)))
            for _ in range(3)
        ]

    train(_loss, st, steps=3, batches=batches, log_every=0, val_every=0)
    trees = [
        {k: p.detach() for k, p in r.model.named_parameters()}
        for r in st.replicas
    ]
    assert ensure_synced(trees, rtol=1e-6, atol=1e-7)


def test_task_ddp_matches_large_batch(seed):
    """2 replicas with per-replica batch B == 1 replica with batch 2B
    (same data, mean loss): identical parameters afterwards."""
    model = _mlp(7)
    xs = [torch.randn(4, 3, 2, 2) for _ in range(3)]
    ys = [torch.randint(0, 4, (4,)) for _ in range(3)]

    # single-device large batch
    solo = copy.deepcopy(model)
    opt = FusedSGDMomentum(solo.parameters(), lr=0.05, momentum=0.9)
    for x, y in zip(xs, ys):
        opt.zero_grad()
        _loss(solo(x), y).backward()
        opt.step()

    st = prepare_training(
        model, None, devices=[0, 1],
        opt_factory=lambda m: FusedSGDMomentum(m.parameters(), lr=0.05, momentum=0.9),
    )
    train(_loss, st, steps=3, log_every=0, val_every=0,
          batches=lambda j: [(xs[j][:2], ys[j][:2]), (xs[j][2:], ys[j][2:])])

    for (k, p_solo), (_, p_ddp) in zip(
        solo.named_parameters(), st.replicas[0].model.named_parameters()
    ):
        assert torch.allclose(p_solo, p_ddp, rtol=1e-4, atol=1e-5), k


def test_resnet_cifar_loss_decreases(seed):
    """BASELINE config 1: ResNet-18 / CIFAR-shape, CPU, loss must decrease."""
    model = resnet18(num_classes=10, small_input=True)
    x = torch.randn(16, 3, 32, 32)
    y = torch.randint(0, 10, (16,))
    st = prepare_training(
        model, None, devices=[0],
        opt_factory=lambda m: FusedSGDMomentum(m.parameters(), lr=0.05, momentum=0.9),
    )
    r = st.replicas[0]
    losses = []
    for _ in range(8):
        l = train_step(_loss, st.buffer, r, x, y)
        final = sync_buffer(st.buffer)
        update(r, final)
        losses.append(float(l))
    assert losses[-1] < losses[0], losses


def test_oom_skip_counts_missed(seed, monkeypatch):
    """the reference's OOM-skip path with a WORKING num_missed counter."""
    model = _mlp()
    st = prepare_training(
        model, None, devices=[0, 1],
        opt_factory=lambda m: FusedSGDMomentum(m.parameters(), lr=0.1),
    )
    calls = {"n": 0}
    real_step = train_step

    def flaky(loss_fn, buffer, replica, x, y):
        calls["n"] += 1
        if calls["n"] == 1:
            raise RuntimeError("HIP out of memory: simulated")
        return real_step(loss_fn, buffer, replica, x, y)

    monkeypatch.setattr("fluxdistributed_amd.parallel.task_ddp.train_step", flaky)
    train(_loss, st, steps=2, log_every=0, val_every=0,
          batches=lambda j: [
              (torch.randn(2, 3, 2, 2), torch.randint(0, 4, (2,)))
              for _ in range(2)
          ])
    assert st.num_missed == 1
