import pytest
import torch

from fluxdistributed_amd.models import build_model, resnet18, resnet34, resnet50


def test_resnet34_param_count():
    # reference flagship: ResNet-34 ~21.8M params
    m = resnet34(num_classes=1000)
    n = sum(p.numel() for p in m.parameters())
    assert 21e6 < n < 22.5e6, n


@pytest.mark.parametrize("name,expansion", [("resnet18", 1), ("resnet50", 4)])
def test_forward_shapes(name, expansion, seed):
    m = build_model(name, num_classes=17)
    x = torch.randn(2, 3, 64, 64)
    assert m(x).shape == (2, 17)


def test_small_input_stem(seed):
    # CIFAR stem (BASELINE config 1): 3x3 stride-1 conv, no maxpool
    m = resnet18(num_classes=10, small_input=True)
    x = torch.randn(4, 3, 32, 32)
    assert m(x).shape == (4, 10)


def test_backward_produces_grads(seed):
    m = resnet18(num_classes=5)
    x = torch.randn(2, 3, 32, 32)
    m(x).sum().backward()
    grads = [p.grad for p in m.parameters()]
    assert all(g is not None for g in grads)
    assert all(torch.isfinite(g).all() for g in grads)


def test_bn_running_stats_update(seed):
    m = resnet18(num_classes=5)
    rm0 = m.bn1.running_mean.clone()
    m.train()
    m(torch.randn(4, 3, 32, 32) + 3.0)
    assert not torch.allclose(m.bn1.running_mean, rm0)
    # counter is flushed lazily into the buffer at state_dict time
    assert int(m.state_dict()["bn1.num_batches_tracked"]) == 1
