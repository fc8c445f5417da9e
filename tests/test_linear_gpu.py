"""FdaLinear numerics vs fp32 PyTorch oracle (the reference's Dense FC,
/root/reference FC 512->1000; SURVEY.md §2.4 Dense row). The native path
runs the FC as a 1x1 conv on the MFMA implicit-GEMM kernels with
out_features padded to %64 in-house (ops/linear.py)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("GPU-only suite", allow_module_level=True)

SHAPES = [
    (96, 512, 1000),    # ResNet-18/34 head, flagship batch
    (96, 2048, 1000),   # ResNet-50/152 head
    (32, 512, 32),      # CIFAR tests head
    (7, 512, 1000),     # M not multiple of anything
]


@pytest.mark.parametrize("shape", SHAPES)
@pytest.mark.parametrize("with_bias", [True, False])
def test_fda_linear_matches_fp32_oracle(shape, with_bias):
    from fluxdistributed_amd.ops.linear import FdaLinear, _fda_linear_supported

    M, Cin, N = shape
    torch.manual_seed(11)
    x32 = torch.randn(M, Cin, device="cuda", requires_grad=True)
    lin32 = torch.nn.Linear(Cin, N, bias=with_bias).cuda()
    y32 = lin32(x32)
    gy = torch.randn_like(y32)
    y32.backward(gy)

    lin = FdaLinear(Cin, N, bias=with_bias).cuda().bfloat16()
    with torch.no_grad():
        lin.weight.copy_(lin32.weight)
        if with_bias:
            lin.bias.copy_(lin32.bias)
    xb = x32.detach().bfloat16().requires_grad_(True)
    assert _fda_linear_supported(xb, lin.weight)
    y = lin(xb)
    assert y.shape == (M, N)
    scale = y32.abs().max().item()
    assert (y.float() - y32).abs().max().item() < 0.03 * max(scale, 1.0)

    y.backward(gy.bfloat16())
    for got, ref, name in [
        (xb.grad.float(), x32.grad, "dx"),
        (lin.weight.grad.float(), lin32.weight.grad, "dw"),
    ] + ([(lin.bias.grad.float(), lin32.bias.grad, "db")] if with_bias else []):
        s = ref.abs().max().item()
        err = (got - ref).abs().max().item()
        assert err < 0.04 * max(s, 1.0), f"{name}: err={err} scale={s}"


def test_fda_linear_direct_grad_into_flat_buffer():
    """With a flat fused optimizer the FC grads land in G directly (no
    AccumulateGrad); grads must match the non-flat path."""
    from fluxdistributed_amd.ops.linear import FdaLinear
    from fluxdistributed_amd.ops import FusedSGDMomentum

    torch.manual_seed(3)
    lin_a = FdaLinear(512, 1000).cuda().bfloat16()
    lin_b = FdaLinear(512, 1000).cuda().bfloat16()
    with torch.no_grad():
        lin_b.weight.copy_(lin_a.weight)
        lin_b.bias.copy_(lin_a.bias)
    opt = FusedSGDMomentum(lin_b.parameters(), lr=0.0)

    x = torch.randn(96, 512, device="cuda").bfloat16()
    gy = torch.randn(96, 1000, device="cuda").bfloat16()

    ya = lin_a(x)
    ya.backward(gy)
    opt.zero_grad()
    yb = lin_b(x)
    yb.backward(gy)
    torch.cuda.synchronize()
    assert torch.equal(ya.float(), yb.float())
    assert torch.allclose(lin_a.weight.grad.float(), lin_b.weight.grad.float(),
                          rtol=1e-2, atol=1e-3)
    assert torch.allclose(lin_a.bias.grad.float(), lin_b.bias.grad.float(),
                          rtol=1e-2, atol=1e-3)
