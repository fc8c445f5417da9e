"""Ensemble ("pick best") trainer tests — reference test.jl:26-63 parity."""

import torch

from fluxdistributed_amd.parallel.ensemble import (
    train_ensemble, make_replicas, lr_div5_every10,
)


def _tiny():
    return torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                               torch.nn.Linear(16, 4))


def test_ensemble_converges_and_syncs():
    torch.manual_seed(0)
    model = _tiny()
    reps = make_replicas(model, devices=[0, 1, 2],
                         opt_factory=lambda m: torch.optim.SGD(m.parameters(), lr=0.1))
    xs = torch.randn(64, 8)
    ys = (xs.sum(dim=1) > 0).long() % 4
    val = (xs[:16], ys[:16])

    def batches(cyc, i):
        g = torch.Generator().manual_seed(cyc * 10 + i)
        idx = torch.randperm(64, generator=g)[:16]
        return [(xs[idx], ys[idx])]

    loss_fn = torch.nn.functional.cross_entropy
    hist, reps = train_ensemble(loss_fn, reps, val, cycles=3,
                                steps_per_cycle=1, batches=batches)
    assert len(hist) == 3
    # after the final broadcast all replicas hold the same weights
    s0 = reps[0].model.state_dict()
    for r in reps[1:]:
        for k, v in r.model.state_dict().items():
            assert torch.equal(v, s0[k])


def test_lr_schedule_div5_every10():
    m = _tiny()
    opt = torch.optim.SGD(m.parameters(), lr=1.0)
    for c in range(1, 21):
        lr_div5_every10([opt], c)
    assert abs(opt.param_groups[0]["lr"] - 1.0 / 25.0) < 1e-9
