"""GPU numerics: every HIP kernel vs a plain fp32 PyTorch reference."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _native():
    from fluxdistributed_amd.ops.native import load_native

    mod = load_native()
    assert mod is not None, "native extension must be present on GPU box"
    return mod


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-5), (torch.bfloat16, 2e-2)])
def test_ce_fwd(dtype, tol, seed):
    C = _native()
    x = torch.randn(96, 1000, device=DEV, dtype=dtype)
    y = torch.randint(0, 1000, (96,), device=DEV)
    loss, dlogits = C.ce_fwd(x.contiguous(), y)
    xr = x.float().cpu().requires_grad_()
    ref = F.cross_entropy(xr, y.cpu())
    ref.backward()
    assert abs(float(loss) - float(ref)) < tol * max(1.0, abs(float(ref)))
    assert torch.allclose(dlogits.float().cpu(), xr.grad, rtol=tol, atol=tol * 1e-2)


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-6), (torch.bfloat16, 1e-2)])
@pytest.mark.parametrize("n", [4096, 4099])  # vector path + scalar tail
def test_add_relu(dtype, tol, n, seed):
    C = _native()
    a = torch.randn(n, device=DEV, dtype=dtype)
    b = torch.randn(n, device=DEV, dtype=dtype)
    out = C.add_relu_fwd(a, b)
    ref = torch.relu(a.float() + b.float())
    assert torch.allclose(out.float(), ref.to(out.dtype).float(), rtol=tol, atol=tol)
    g = torch.randn(n, device=DEV, dtype=dtype)
    gx = C.add_relu_bwd(g, out)
    mask = (out.float() > 0)
    assert torch.allclose(gx.float(), torch.where(mask, g.float(), torch.zeros(())),
                          rtol=tol, atol=tol)


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-4), (torch.bfloat16, 3e-2)])
@pytest.mark.parametrize("Cc,training,relu,res", [
    (64, True, True, False),
    (128, True, True, True),
    (256, True, False, False),
    (512, False, True, False),
    (2048, True, True, True),
])
def test_bn_act_fwd(dtype, tol, Cc, training, relu, res, seed):
    C = _native()
    N, H, W = 4, 7, 7
    x = torch.randn(N, Cc, H, W, device=DEV, dtype=dtype).contiguous(
        memory_format=torch.channels_last)
    w = (torch.rand(Cc, device=DEV) + 0.5)
    b = torch.randn(Cc, device=DEV)
    rm = torch.randn(Cc, device=DEV) * 0.1
    rv = torch.rand(Cc, device=DEV) + 0.5
    rm2, rv2 = rm.clone(), rv.clone()
    residual = (torch.randn_like(x).contiguous(memory_format=torch.channels_last)
                if res else torch.empty(0, device=DEV, dtype=dtype))

    out, mean, invstd = C.bn_act_fwd(x, w, b, rm, rv, training, 0.1, 1e-5, relu,
                                     residual)

    xf = x.float()
    ref = F.batch_norm(xf, rm2, rv2, w, b, training, 0.1, 1e-5)
    if res:
        ref = ref + residual.float()
    if relu:
        ref = torch.relu(ref)
    assert torch.allclose(out.float(), ref, rtol=tol, atol=tol), \
        (out.float() - ref).abs().max()
    # running stats must match PyTorch's update
    assert torch.allclose(rm, rm2, rtol=1e-4, atol=1e-5)
    assert torch.allclose(rv, rv2, rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-3), (torch.bfloat16, 5e-2)])
@pytest.mark.parametrize("Cc,relu", [(64, True), (256, False)])
def test_bn_act_bwd(dtype, tol, Cc, relu, seed):
    C = _native()
    N, H, W = 4, 9, 9
    x = torch.randn(N, Cc, H, W, device=DEV, dtype=dtype).contiguous(
        memory_format=torch.channels_last)
    w = (torch.rand(Cc, device=DEV) + 0.5)
    b = torch.randn(Cc, device=DEV)
    rm, rv = torch.zeros(Cc, device=DEV), torch.ones(Cc, device=DEV)
    out, mean, invstd = C.bn_act_fwd(x, w, b, rm, rv, True, 0.1, 1e-5, relu,
                                     torch.empty(0, device=DEV, dtype=dtype))
    gout = torch.randn_like(x).contiguous(memory_format=torch.channels_last)
    gx, gw, gb, _ = C.bn_act_bwd(gout, x, w, mean, invstd, out, relu, True)

    # fp32 autograd reference
    xr = x.float().detach().requires_grad_()
    wr = w.detach().requires_grad_()
    br = b.detach().requires_grad_()
    ref = F.batch_norm(xr, torch.zeros(Cc, device=DEV), torch.ones(Cc, device=DEV),
                       wr, br, True, 0.1, 1e-5)
    if relu:
        ref = torch.relu(ref)
    ref.backward(gout.float())
    assert torch.allclose(gx.float(), xr.grad, rtol=tol, atol=tol), \
        (gx.float() - xr.grad).abs().max()
    assert torch.allclose(gw, wr.grad, rtol=tol, atol=tol * 10)
    assert torch.allclose(gb, br.grad, rtol=tol, atol=tol * 10)


def test_sgd_step_matches_ref(seed):
    C = _native()
    n = 64 * 10
    P = torch.randn(n, device=DEV, dtype=torch.bfloat16)
    M = P.float()
    G = torch.randn(n, device=DEV, dtype=torch.bfloat16)
    V = torch.randn(n, device=DEV).abs()
    Mr, Vr = M.clone(), V.clone()
    C.sgd_step(P, G, M, V, 0.05, 0.9, 1e-4, False)
    g = G.float() + 1e-4 * Mr
    Vref = 0.9 * Vr + g
    Mref = Mr - 0.05 * Vref
    assert torch.allclose(V, Vref, rtol=1e-5, atol=1e-6)
    assert torch.allclose(M, Mref, rtol=1e-5, atol=1e-6)
    assert torch.allclose(P.float(), Mref, rtol=1e-2, atol=1e-2)


def test_adam_step_matches_ref(seed):
    C = _native()
    n = 64 * 10
    P = torch.randn(n, device=DEV)
    M = P  # fp32: master aliases P
    G = torch.randn(n, device=DEV)
    V = torch.zeros(n, device=DEV)
    S = torch.zeros(n, device=DEV)
    P0 = P.clone()
    b1, b2, lr, eps = 0.9, 0.999, 1e-2, 1e-8
    C.adam_step(P, G, M, V, S, lr, b1, b2, eps, 0.0, 1 - b1, 1 - b2)
    Vref = (1 - b1) * G
    Sref = (1 - b2) * G * G
    Pref = P0 - lr * (Vref / (1 - b1)) / ((Sref / (1 - b2)).sqrt() + eps)
    assert torch.allclose(V, Vref, rtol=1e-5, atol=1e-7)
    assert torch.allclose(P, Pref, rtol=1e-5, atol=1e-7)


def test_fused_optimizer_end_to_end_gpu(seed):
    from fluxdistributed_amd.ops import FusedSGDMomentum

    torch.manual_seed(0)
    m1 = torch.nn.Linear(32, 8).to(DEV)
    m2 = torch.nn.Linear(32, 8).to(DEV)
    m2.load_state_dict(m1.state_dict())
    o1 = FusedSGDMomentum(m1.parameters(), lr=0.05, momentum=0.9)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.05, momentum=0.9)
    x = torch.randn(16, 32, device=DEV)
    for _ in range(5):
        o1.zero_grad()
        m1(x).pow(2).sum().backward()
        o1.step()
        o2.zero_grad()
        m2(x).pow(2).sum().backward()
        o2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 0), (torch.bfloat16, 0)])
@pytest.mark.parametrize("shape,k,s,p", [
    ((2, 64, 112, 112), 3, 2, 1),   # ResNet stem pool
    ((2, 128, 17, 17), 3, 2, 1),    # odd sizes
    ((2, 64, 16, 16), 2, 2, 0),
])
def test_maxpool(dtype, tol, shape, k, s, p, seed):
    C = _native()
    x = torch.randn(*shape, device=DEV, dtype=dtype).contiguous(
        memory_format=torch.channels_last)
    out, idx = C.maxpool_fwd(x, k, k, s, p)
    ref = F.max_pool2d(x.float(), k, s, p)
    assert out.shape == ref.shape
    assert torch.equal(out.float(), ref.to(out.dtype).float())
    # backward: compare against autograd on the fp32 reference
    xr = x.float().detach().requires_grad_()
    refo = F.max_pool2d(xr, k, s, p)
    g = torch.randn_like(refo)
    refo.backward(g)
    gx = C.maxpool_bwd(g.to(out.dtype), idx, shape[2], shape[3], k, k, s, p)
    # ties may pick different argmax; tolerate tiny fraction of mismatches
    diff = (gx.float() - xr.grad).abs()
    frac_bad = float((diff > 1e-2).float().mean())
    assert frac_bad < 2e-3, frac_bad


def test_global_avg_pool(seed):
    from fluxdistributed_amd.ops.functional import global_avg_pool

    x = torch.randn(6, 512, 7, 7, device=DEV, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = global_avg_pool(x)
    ref = x.float().mean(dim=(2, 3))
    assert (y.float() - ref).abs().max().item() < 2e-2
    gy = torch.randn_like(y)
    y.backward(gy)
    gref = (gy.float() / 49).unsqueeze(-1).unsqueeze(-1).expand(6, 512, 7, 7)
    assert (x.grad.float() - gref).abs().max().item() < 1e-3
