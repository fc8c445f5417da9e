"""conv->BN stats-fusion numerics: the conv epilogue's per-tile sum/sumsq
partials (bn_finalize_from_partials) must give the same batch stats,
output, running stats and gradients as BN's own stats pass. This path was
dead until r2 (grad mode is invisible inside Function.forward) — this is
its dedicated hardware oracle."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("GPU-only suite", allow_module_level=True)


@pytest.mark.parametrize("shape", [
    (4, 64, 56, 56, 64, 3, 1),
    (96, 256, 14, 14, 512, 3, 2),   # split-K fwd shape (combine emits stats)
    (4, 3, 64, 64, 64, 7, 2),       # stem kernel stats
    (96, 64, 56, 56, 64, 3, 1),     # full-batch layer1: BK32 128x64 stats
])
def test_conv_bn_fused_stats_match_unfused(shape, monkeypatch):
    from fluxdistributed_amd.models.resnet import FusedBNAct
    from fluxdistributed_amd.ops.conv import fda_conv2d

    n, c, h, w, k, r, s = shape
    pad = r // 2
    torch.manual_seed(4)
    x = torch.randn(n, c, h, w).cuda().bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(k, c, r, r) * (c * r * r) ** -0.5).cuda().bfloat16() \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)

    def run(fuse):
        monkeypatch.setenv("FLUXDIST_BN_FUSE", "1" if fuse else "0")
        torch.manual_seed(9)
        bn = FusedBNAct(k, relu=True).cuda().train()
        wt.grad = None
        y = fda_conv2d(x, wt, (s, s), (pad, pad))
        out = bn(y)
        out.float().square().mean().backward()
        torch.cuda.synchronize()
        return (out.float(), bn.running_mean.clone(), bn.running_var.clone(),
                wt.grad.float().clone())

    out1, rm1, rv1, gw1 = run(True)
    out0, rm0, rv0, gw0 = run(False)
    assert torch.allclose(rm1, rm0, rtol=1e-3, atol=1e-4), \
        f"running_mean: {float((rm1-rm0).abs().max())}"
    assert torch.allclose(rv1, rv0, rtol=1e-3, atol=1e-4), \
        f"running_var: {float((rv1-rv0).abs().max())}"
    assert torch.allclose(out1, out0, rtol=2e-2, atol=2e-3), \
        f"out: {float((out1-out0).abs().max())}"
    assert torch.allclose(gw1, gw0, rtol=2e-2, atol=2e-3), \
        f"gw: {float((gw1-gw0).abs().max())}"

    # and against the fp32 oracle
    xf = x.float()
    wf = wt.detach().float()
    yf = torch.nn.functional.conv2d(xf, wf, stride=s, padding=pad)
    bnf = torch.nn.BatchNorm2d(k).cuda().train()
    reff = torch.relu(bnf(yf))
    assert torch.allclose(rm1, bnf.running_mean, rtol=5e-2, atol=5e-3)
    scale = float(reff.abs().max())
    assert float((out1 - reff).abs().max()) < 0.05 * max(scale, 1.0)
