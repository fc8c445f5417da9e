import torch
import torch.nn.functional as F

from fluxdistributed_amd.ops import logit_cross_entropy, fused_add_relu, batch_norm_act
from fluxdistributed_amd.utils.metrics import topkaccuracy, showpreds


def test_logit_ce_matches_torch(seed):
    x = torch.randn(5, 11, requires_grad=True)
    y = torch.randint(0, 11, (5,))
    loss = logit_cross_entropy(x, y)
    ref = F.cross_entropy(x, y)
    assert torch.allclose(loss, ref, rtol=1e-5, atol=1e-6)
    loss.backward()
    x2 = x.detach().clone().requires_grad_()
    F.cross_entropy(x2, y).backward()
    assert torch.allclose(x.grad, x2.grad, rtol=1e-5, atol=1e-6)


def test_fused_add_relu(seed):
    a = torch.randn(4, 6, requires_grad=True)
    b = torch.randn(4, 6, requires_grad=True)
    out = fused_add_relu(a, b)
    assert torch.allclose(out, torch.relu(a + b))
    out.sum().backward()
    mask = ((a + b) > 0).float()
    assert torch.allclose(a.grad, mask)
    assert torch.allclose(b.grad, mask)


def test_batch_norm_act_cpu_matches_composed(seed):
    x = torch.randn(4, 8, 5, 5)
    w, b = torch.rand(8) + 0.5, torch.randn(8)
    rm, rv = torch.zeros(8), torch.ones(8)
    res = torch.randn(4, 8, 5, 5)
    out = batch_norm_act(x, w, b, rm.clone(), rv.clone(), True, 0.1, 1e-5,
                         relu=True, residual=res)
    ref = torch.relu(F.batch_norm(x, rm.clone(), rv.clone(), w, b, True, 0.1, 1e-5) + res)
    assert torch.allclose(out, ref, rtol=1e-5, atol=1e-5)


def test_topk_accuracy():
    logits = torch.tensor([[0.1, 0.9, 0.0], [0.8, 0.1, 0.1]])
    y = torch.tensor([1, 2])
    assert topkaccuracy(logits, y, k=1) == 0.5
    assert topkaccuracy(logits, y, k=3) == 1.0
    onehot = F.one_hot(y, 3).float()
    assert topkaccuracy(logits, onehot, k=1) == 0.5


def test_showpreds_runs():
    s = showpreds(torch.randn(2, 5), torch.tensor([0, 1]),
                  class_names=[f"c{i}" for i in range(5)])
    assert "true=c0" in s
