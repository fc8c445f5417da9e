"""Mixed-precision policy for MI355X training.

bf16 compute for conv/linear (MFMA-shaped work), fp32 for BatchNorm
parameters and statistics (the BN kernels compute in fp32 internally and
keep params fp32 for numerics), fp32 master weights in the fused optimizer.
"""

import torch

from ..models.resnet import FusedBNAct


def to_mixed_bf16(model: torch.nn.Module) -> torch.nn.Module:
    """Cast conv/linear params to bf16; keep norm-layer params + buffers fp32."""
    for mod in model.modules():
        if isinstance(mod, FusedBNAct):
            continue
        if isinstance(mod, (torch.nn.Conv2d, torch.nn.Linear)):
            for name, p in list(mod._parameters.items()):
                if p is not None:
                    mod._parameters[name] = torch.nn.Parameter(
                        p.data.to(torch.bfloat16), requires_grad=p.requires_grad
                    )
    return model


def to_channels_last(model: torch.nn.Module) -> torch.nn.Module:
    return model.to(memory_format=torch.channels_last)
