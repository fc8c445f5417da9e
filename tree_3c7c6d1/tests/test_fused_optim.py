import copy

import torch
import torch.nn as nn

from fluxdistributed_amd.ops import FusedSGDMomentum, FusedAdam


def _mlp(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))


def test_grads_land_in_flat_buffer(seed):
    m = _mlp()
    opt = FusedSGDMomentum(m.parameters(), lr=0.1)
    (g,) = opt.flat_grads()
    assert g.abs().sum() == 0
    m(torch.randn(5, 8)).sum().backward()
    assert g.abs().sum() > 0
    # every param's .grad is a view of the flat buffer
    for p in m.parameters():
        assert p.grad is not None and p.grad.data_ptr() >= g.data_ptr()
        assert p.grad.data_ptr() < g.data_ptr() + g.numel() * g.element_size()


def test_sgd_matches_torch(seed):
    m1, m2 = _mlp(1), _mlp(1)
    opt1 = FusedSGDMomentum(m1.parameters(), lr=0.05, momentum=0.9, weight_decay=1e-4)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.05, momentum=0.9,
                           weight_decay=1e-4, dampening=0.0)
    x = torch.randn(6, 8)
    for _ in range(5):
        opt1.zero_grad()
        m1(x).pow(2).sum().backward()
        opt1.step()
        opt2.zero_grad()
        m2(x).pow(2).sum().backward()
        opt2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, rtol=1e-5, atol=1e-6)


def test_adam_matches_torch(seed):
    m1, m2 = _mlp(2), _mlp(2)
    opt1 = FusedAdam(m1.parameters(), lr=1e-2)
    opt2 = torch.optim.Adam(m2.parameters(), lr=1e-2)
    x = torch.randn(6, 8)
    for _ in range(5):
        opt1.zero_grad()
        m1(x).pow(2).sum().backward()
        opt1.step()
        opt2.zero_grad()
        m2(x).pow(2).sum().backward()
        opt2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, rtol=1e-4, atol=1e-6)


def test_state_dict_roundtrip(seed):
    m = _mlp(3)
    opt = FusedSGDMomentum(m.parameters(), lr=0.05, momentum=0.9)
    x = torch.randn(4, 8)
    for _ in range(3):
        opt.zero_grad()
        m(x).sum().backward()
        opt.step()
    sd = opt.state_dict()
    params_after3 = [p.detach().clone() for p in m.parameters()]

    m2 = _mlp(3)
    opt2 = FusedSGDMomentum(m2.parameters(), lr=0.05, momentum=0.9)
    # fast-forward m2 to the checkpointed weights + optimizer state
    with torch.no_grad():
        for p2, p1 in zip(m2.parameters(), params_after3):
            p2.copy_(p1)
    opt2.load_state_dict(sd)

    # one more identical step on both must agree exactly
    for mm, oo in ((m, opt), (m2, opt2)):
        oo.zero_grad()
        mm(x).sum().backward()
        oo.step()
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, rtol=0, atol=0)


def test_bf16_master_weights(seed):
    m = _mlp(4).to(torch.bfloat16)
    opt = FusedSGDMomentum(m.parameters(), lr=0.01, momentum=0.9)
    g = opt.groups[0]
    assert g.master is not None and g.master.dtype == torch.float32
    x = torch.randn(4, 8, dtype=torch.bfloat16)
    opt.zero_grad()
    m(x).float().sum().backward()
    opt.step()
    # master and bf16 params stay consistent
    assert torch.allclose(g.P.float(), g.master, rtol=1e-2, atol=1e-2)
