"""ResNet family (18/34/50/101/152), written from scratch for this framework.

Parity: the reference trains Metalhead ResNets — ResNet-34 flagship
(/root/reference/README.md:27), ResNet-50 in the process path
(/root/reference/src/sync.jl:159). Blocks here use the fused BN(+add)+ReLU
op (one HIP kernel on GPU) instead of three separate launches.

Conv weights are bias-free, as in the reference (Flux convs with
`bias=Flux.Zeros` — see SURVEY.md §2.4).
"""

from typing import List, Optional, Type

import torch
import torch.nn as nn

from ..ops.linear import FdaLinear
from ..ops.functional import (batch_norm_act, MaxPool2d,
                              GlobalAvgPool)


class FusedBNAct(nn.Module):
    """BatchNorm2d + optional residual-add + optional ReLU as one op.

    Running stats are per-replica and never synchronized across data-parallel
    replicas — the reference's exact semantic (SURVEY.md §7 hard-part 3:
    grads averaged, running stats drift independently).
    """

    def __init__(self, num_features: int, relu: bool = True,
                 momentum: float = 0.1, eps: float = 1e-5):
        super().__init__()
        self.num_features = num_features
        self.relu = relu
        self.momentum = momentum
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))
        # batch counter is kept host-side and flushed into the buffer lazily:
        # a per-step GPU scalar add per BN layer was 29 launches/step of pure
        # overhead (profiles/README.md)
        self._nbt_pending = 0

    def _flush_nbt(self):
        if self._nbt_pending:
            self.num_batches_tracked += self._nbt_pending
            self._nbt_pending = 0

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        self._flush_nbt()
        super()._save_to_state_dict(destination, prefix, keep_vars)

    def _load_from_state_dict(self, *args, **kw):
        self._nbt_pending = 0
        super()._load_from_state_dict(*args, **kw)

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor] = None,
                defer_gres: bool = False):
        if self.training:
            self._nbt_pending += 1
        return batch_norm_act(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.training, self.momentum, self.eps, self.relu, residual,
            defer_gres,
        )


def _junction_fusible(x: torch.Tensor, conv: nn.Conv2d) -> bool:
    """True when the identity-shortcut junction gradient can be folded into
    `conv`'s native dgrad += epilogue: the SAME tensor x must reach the
    native dgrad (channels_last, native-supported shape) and need a grad.
    The contract is load-bearing — a deferred gres that no dgrad consumes
    raises in ops/conv.py rather than silently dropping gradient."""
    import os

    from ..ops.conv import _native_supported

    if os.environ.get("FLUXDIST_JUNCTION", "1") == "0":
        return False
    if not (torch.is_grad_enabled() and x.requires_grad):
        return False
    if not (x.is_cuda and x.is_contiguous(memory_format=torch.channels_last)):
        return False
    return _native_supported(x, conv.weight, conv.stride, conv.padding,
                             conv.dilation, conv.groups)


def conv3x3(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    from ..ops.conv import FdaConv2d

    return FdaConv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    from ..ops.conv import FdaConv2d

    return FdaConv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin: int, cout: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = conv3x3(cin, cout, stride)
        self.bn1 = FusedBNAct(cout, relu=True)
        self.conv2 = conv3x3(cout, cout)
        self.bn2 = FusedBNAct(cout, relu=True)  # fused add+relu via residual arg
        self.downsample = downsample

    def forward(self, x):
        if self.downsample is not None:
            # downsample runs FIRST: python arg-evaluation order otherwise
            # clobbers conv2's BN-stats stash with the downsample conv's
            # (the 3 residual bn_stats launches in the r2 profile)
            identity = self.downsample(x)
            out = self.bn1(self.conv1(x))
            return self.bn2(self.conv2(out), residual=identity)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=x,
                        defer_gres=_junction_fusible(x, self.conv1))


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, cout: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = conv1x1(cin, cout)
        self.bn1 = FusedBNAct(cout, relu=True)
        self.conv2 = conv3x3(cout, cout, stride)
        self.bn2 = FusedBNAct(cout, relu=True)
        self.conv3 = conv1x1(cout, cout * self.expansion)
        self.bn3 = FusedBNAct(cout * self.expansion, relu=True)
        self.downsample = downsample

    def forward(self, x):
        if self.downsample is not None:
            identity = self.downsample(x)   # first — see BasicBlock note
            out = self.bn1(self.conv1(x))
            out = self.bn2(self.conv2(out))
            return self.bn3(self.conv3(out), residual=identity)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=x,
                        defer_gres=_junction_fusible(x, self.conv1))


class Downsample(nn.Module):
    """1x1 strided conv + BN (no activation) on the identity path."""

    def __init__(self, cin: int, cout: int, stride: int):
        super().__init__()
        self.conv = conv1x1(cin, cout, stride)
        self.bn = FusedBNAct(cout, relu=False)

    def forward(self, x):
        return self.bn(self.conv(x))


class ResNet(nn.Module):
    def __init__(self, block: Type[nn.Module], layers: List[int],
                 num_classes: int = 1000, small_input: bool = False):
        super().__init__()
        self.small_input = small_input
        self.cin = 64
        from ..ops.conv import FdaConv2d

        if small_input:  # CIFAR-style stem (BASELINE config 1)
            self.conv1 = FdaConv2d(3, 64, 3, stride=1, padding=1, bias=False)
            self.maxpool = nn.Identity()
        else:
            self.conv1 = FdaConv2d(3, 64, 7, stride=2, padding=3, bias=False)
            self.maxpool = MaxPool2d(3, stride=2, padding=1)
        self.bn1 = FusedBNAct(64, relu=True)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = GlobalAvgPool()
        self.fc = FdaLinear(512 * block.expansion, num_classes)
        self._init_weights()

    def _make_layer(self, block, cout, n, stride=1):
        downsample = None
        if stride != 1 or self.cin != cout * block.expansion:
            downsample = Downsample(self.cin, cout * block.expansion, stride)
        blocks = [block(self.cin, cout, stride, downsample)]
        self.cin = cout * block.expansion
        blocks += [block(self.cin, cout) for _ in range(n - 1)]
        return nn.Sequential(*blocks)

    def _init_weights(self):
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, 0, 0.01)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x)
        return self.fc(x)


def resnet18(**kw):
    return ResNet(BasicBlock, [2, 2, 2, 2], **kw)


def resnet34(**kw):
    return ResNet(BasicBlock, [3, 4, 6, 3], **kw)


def resnet50(**kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], **kw)


def resnet101(**kw):
    return ResNet(Bottleneck, [3, 4, 23, 3], **kw)


def resnet152(**kw):
    return ResNet(Bottleneck, [3, 8, 36, 3], **kw)


_ZOO = {
    "resnet18": resnet18,
    "resnet34": resnet34,
    "resnet50": resnet50,
    "resnet101": resnet101,
    "resnet152": resnet152,
}


def build_model(name: str, num_classes: int = 1000, small_input: bool = False) -> ResNet:
    if name not in _ZOO:
        raise ValueError(f"unknown model '{name}'; choose from {sorted(_ZOO)}")
    return _ZOO[name](num_classes=num_classes, small_input=small_input)
