"""Numerics tests for the implicit-GEMM conv kernels (conv_igemm.hip).

Oracle: plain fp32 PyTorch conv of the same op (SURVEY.md §4 — every HIP
kernel is compared against a plain PyTorch fp32 reference).
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("GPU-only suite", allow_module_level=True)

from fluxdistributed_amd.ops.conv import fda_conv2d, _native_supported  # noqa: E402
from fluxdistributed_amd.ops.native import require_native  # noqa: E402


# (N, C, H, W, K, R, stride) — ResNet-34 body shapes at small batch plus
# edge cases (odd M tiles: 7x7 spatial).
SHAPES = [
    (4, 64, 56, 56, 64, 3, 1),     # layer1 3x3
    (4, 64, 56, 56, 128, 1, 2),    # downsample 1x1 s2
    (4, 128, 28, 28, 128, 3, 1),
    (4, 128, 28, 28, 256, 3, 2),   # 3x3 s2
    (4, 256, 14, 14, 512, 1, 2),
    (96, 512, 7, 7, 512, 3, 1),    # M=4704 not a multiple of 128
    (2, 64, 9, 11, 64, 3, 1),      # odd spatial
    (3, 64, 113, 113, 128, 3, 2),  # odd spatial + stride 2 (parity classes)
    (2, 128, 14, 14, 512, 1, 1),   # bottleneck expand 1x1
    (2, 256, 27, 27, 64, 1, 2),    # odd spatial 1x1 s2
    (2, 64, 56, 56, 256, 1, 1),    # bottleneck downsample-free expand
    # ResNet-50/152 bottleneck channel extremes (round-1 verdict next #7):
    (2, 1024, 14, 14, 256, 1, 1),  # r50 layer3 reduce 1x1 C=1024
    (2, 2048, 7, 7, 512, 1, 1),    # r50/152 layer4 reduce 1x1 C=2048
    (96, 512, 7, 7, 2048, 1, 1),   # layer4 expand 1x1 K=2048 at full batch
                                   # (M=4704 -> wgrad takes the FT=4 tile)
    (2, 1024, 14, 14, 2048, 1, 2), # layer4 downsample 1x1 s2 K=2048
    (4, 512, 14, 14, 512, 3, 2),   # r50 layer4 3x3 s2 C=K=512
    (96, 64, 56, 56, 64, 3, 1),    # full-batch layer1 (BK32 128x64 config)
    (96, 128, 28, 28, 128, 3, 1),  # full-batch layer2 (BK32 128x128 config)
]


def _mk(n, c, h, w, k, r, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(n, c, h, w, generator=g).cuda()
    wt = (torch.randn(k, c, r, r, generator=g) * (c * r * r) ** -0.5).cuda()
    return x, wt


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_fwd_matches_fp32(shape):
    n, c, h, w, k, r, s = shape
    x, wt = _mk(n, c, h, w, k, r)
    pad = r // 2
    ref = torch.nn.functional.conv2d(x, wt, stride=s, padding=pad)
    xb = x.bfloat16().contiguous(memory_format=torch.channels_last)
    wb = wt.bfloat16().contiguous(memory_format=torch.channels_last)
    assert _native_supported(xb, wb, (s, s), (pad, pad), (1, 1), 1)
    C = require_native("conv_igemm_fwd")
    y = C.conv_igemm_fwd(xb, wb, s, s, pad, pad)
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.03 * max(scale, 1.0), f"{shape}: err={err} scale={scale}"


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_dgrad_matches_fp32(shape):
    n, c, h, w, k, r, s = shape
    x, wt = _mk(n, c, h, w, k, r, seed=1)
    pad = r // 2
    x32 = x.clone().requires_grad_(True)
    ref = torch.nn.functional.conv2d(x32, wt, stride=s, padding=pad)
    gy = torch.randn_like(ref)
    ref.backward(gy)
    dx_ref = x32.grad

    C = require_native("conv_igemm_dgrad")
    gyb = gy.bfloat16().contiguous(memory_format=torch.channels_last)
    wtt = (wt.bfloat16().permute(2, 3, 1, 0).reshape(r * r * c, k).contiguous())
    dx = C.conv_igemm_dgrad(gyb, wtt, c, h, w, r, r, s, s, pad, pad)
    err = (dx.float() - dx_ref).abs().max().item()
    scale = dx_ref.abs().max().item()
    assert err < 0.03 * max(scale, 1.0), f"{shape}: err={err} scale={scale}"


def test_conv_autograd_end_to_end():
    n, c, h, w, k, r, s = 4, 64, 14, 14, 128, 3, 2
    x, wt = _mk(n, c, h, w, k, r, seed=2)
    pad = r // 2
    # fp32 reference
    x32 = x.clone().requires_grad_(True)
    w32 = wt.clone().requires_grad_(True)
    out = torch.nn.functional.conv2d(x32, w32, stride=s, padding=pad)
    loss = (out.float() ** 2).mean()
    loss.backward()

    xb = x.bfloat16().contiguous(memory_format=torch.channels_last).requires_grad_(True)
    wb = wt.bfloat16().contiguous(memory_format=torch.channels_last).requires_grad_(True)
    yb = fda_conv2d(xb, wb, (s, s), (pad, pad))
    lb = (yb.float() ** 2).mean()
    lb.backward()

    for got, ref, name in [(xb.grad, x32.grad, "dx"), (wb.grad, w32.grad, "dw")]:
        err = (got.float() - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err < 0.05 * max(scale, 1e-3), f"{name}: err={err} scale={scale}"


def test_model_uses_native_conv():
    """The flagship model's body convs must dispatch to conv_igemm."""
    from fluxdistributed_amd.models import build_model
    from fluxdistributed_amd.ops.conv import FdaConv2d

    m = build_model("resnet34")
    fda = [mod for mod in m.modules() if isinstance(mod, FdaConv2d)]
    assert len(fda) >= 35, f"expected >=35 FdaConv2d body convs, got {len(fda)}"


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_wgrad_matches_fp32(shape):
    n, c, h, w, k, r, s = shape
    x, wt = _mk(n, c, h, w, k, r, seed=3)
    pad = r // 2
    w32 = wt.clone().requires_grad_(True)
    ref = torch.nn.functional.conv2d(x, w32, stride=s, padding=pad)
    gy = torch.randn_like(ref)
    ref.backward(gy)
    dw_ref = w32.grad

    C = require_native("conv_igemm_wgrad")
    gyb = gy.bfloat16().contiguous(memory_format=torch.channels_last)
    xb = x.bfloat16().contiguous(memory_format=torch.channels_last)
    ws = C.conv_igemm_wgrad(gyb, xb, r, r, s, s, pad, pad)
    dw = ws.view(k, r, r, c).permute(0, 3, 1, 2)
    err = (dw - dw_ref).abs().max().item()
    scale = dw_ref.abs().max().item()
    # reduction over N*P*Q in bf16 products, fp32 accumulate
    assert err < 0.02 * max(scale, 1.0), f"{shape}: err={err} scale={scale}"


STEM_SHAPES = [
    (4, 3, 224, 224, 64, 7, 2, 3),   # ImageNet stem
    (8, 3, 32, 32, 64, 3, 1, 1),     # CIFAR stem
]


@pytest.mark.parametrize("shape", STEM_SHAPES)
def test_stem_conv_fwd_and_wgrad(shape):
    from fluxdistributed_amd.ops.conv import _FdaStemConv2d, _stem_supported

    n, c, h, w, k, r, s, pad = shape
    x, wt = _mk(n, c, h, w, k, r, seed=5)
    w32 = wt.clone().requires_grad_(True)
    ref = torch.nn.functional.conv2d(x, w32, stride=s, padding=pad)
    gy = torch.randn_like(ref)
    ref.backward(gy)

    xb = x.bfloat16().contiguous(memory_format=torch.channels_last)
    wb = wt.bfloat16().contiguous(memory_format=torch.channels_last).requires_grad_(True)
    assert _stem_supported(xb, wb, (s, s), (pad, pad), (1, 1), 1)
    y = _FdaStemConv2d.apply(xb, wb, (s, s), (pad, pad), False)
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.03 * max(scale, 1.0), f"fwd err={err} scale={scale}"

    y.backward(gy.bfloat16().contiguous(memory_format=torch.channels_last))
    errw = (wb.grad.float() - w32.grad).abs().max().item()
    scalew = w32.grad.abs().max().item()
    assert errw < 0.02 * max(scalew, 1.0), f"wgrad err={errw} scale={scalew}"


def test_stem_dispatch_in_model_forward():
    from fluxdistributed_amd.ops.conv import fda_conv2d

    x = torch.randn(2, 3, 64, 64).cuda().bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(64, 3, 7, 7) * 0.05).cuda().bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    y = fda_conv2d(x, w, (2, 2), (3, 3))
    ref = torch.nn.functional.conv2d(x.float(), w.float(), stride=2, padding=3)
    assert (y.float() - ref).abs().max().item() < 0.05 * ref.abs().max().item()
