"""Residual-junction grad fusion: the identity-shortcut add in backward is
folded into conv1's dgrad += epilogue (ops/conv.py junction stash). Checks
numerics vs the unfused path and vs the fp32 oracle, and that the stash
contract holds (consumed every backward, loud otherwise)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("GPU-only suite", allow_module_level=True)


def _grads(model, x, y):
    from fluxdistributed_amd.ops import logit_cross_entropy

    model.zero_grad(set_to_none=False)
    loss = logit_cross_entropy(model(x), y)
    loss.backward()
    torch.cuda.synchronize()
    return {n: p.grad.detach().float().clone()
            for n, p in model.named_parameters() if p.grad is not None}


@pytest.mark.parametrize("arch", ["resnet18", "resnet50"])
def test_junction_fusion_matches_unfused(arch, monkeypatch):
    from fluxdistributed_amd.models import build_model
    from fluxdistributed_amd.models import resnet as rn
    from fluxdistributed_amd.ops.conv import junction_stash_empty
    from fluxdistributed_amd.utils.precision import to_mixed_bf16

    torch.manual_seed(21)
    m = build_model(arch, num_classes=16, small_input=True)
    m = to_mixed_bf16(m.to("cuda:0").to(memory_format=torch.channels_last))
    m.train()
    g = torch.Generator().manual_seed(77)
    x = torch.randn(4, 3, 32, 32, generator=g).bfloat16().cuda() \
        .contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 16, (4,), generator=g).cuda()

    engaged = []
    orig = rn._junction_fusible

    def spy(xx, conv):
        r = orig(xx, conv)
        engaged.append(r)
        return r

    monkeypatch.setattr(rn, "_junction_fusible", spy)
    fused = _grads(m, x, y)
    assert junction_stash_empty(), "deferred gres left unconsumed"
    assert any(engaged), "fusion never engaged on an identity block"

    monkeypatch.setattr(rn, "_junction_fusible", lambda *_: False)
    unfused = _grads(m, x, y)
    assert fused.keys() == unfused.keys()
    for k in fused:
        # fused epilogue adds gres at fp32 before the single bf16 round, so
        # allclose (it is MORE precise than bf16+bf16), not equal
        assert torch.allclose(fused[k], unfused[k], rtol=2e-2, atol=2e-3), (
            f"{k}: max|d|={float((fused[k] - unfused[k]).abs().max())}")
