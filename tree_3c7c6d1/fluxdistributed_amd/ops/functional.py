"""Fused training ops: CPU reference path + gfx950 native path.

Each op exists because it is on the ResNet hot path of the reference
(/root/reference/src/ddp_tasks.jl:28 loss; Metalhead basic blocks BN+ReLU;
residual adds) — see SURVEY.md §2.4 for the full kernel inventory.

Numerics contract (tests/test_ops_gpu.py): every native kernel is compared
against the plain fp32 PyTorch composition of the same op.
"""

from typing import Optional

import torch
import torch.nn.functional as F

from .native import require_native


def _on_gpu(*tensors) -> bool:
    return any(t is not None and isinstance(t, torch.Tensor) and t.is_cuda for t in tensors)


# --------------------------------------------------------------------------
# Fused logit cross-entropy (reference: Flux.Losses.logitcrossentropy,
# /root/reference/src/ddp_tasks.jl:28). Forward computes the mean loss AND
# d(loss)/d(logits) in one read of the logits; backward just scales.
# --------------------------------------------------------------------------


class _LogitCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor):
        # target: int64 class indices, shape (N,)
        if logits.is_cuda:
            C = require_native("logit_cross_entropy")
            loss, dlogits = C.ce_fwd(logits, target)
        else:
            x = logits.float()
            lse = torch.logsumexp(x, dim=1, keepdim=True)
            logp = x - lse
            n = logits.shape[0]
            loss = -logp[torch.arange(n), target].mean()
            sm = torch.exp(logp)
            sm[torch.arange(n), target] -= 1.0
            dlogits = (sm / n).to(logits.dtype)
        # loss is returned fp32 on both paths (logged losses must not land
        # on the bf16 grid); dlogits carries the compute dtype.
        ctx.save_for_backward(dlogits)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        (dlogits,) = ctx.saved_tensors
        return dlogits * grad_out.to(dlogits.dtype), None


def logit_cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over un-normalized logits; target = class indices."""
    return _LogitCrossEntropy.apply(logits, target.long())


# --------------------------------------------------------------------------
# Fused residual add + ReLU (tail of every ResNet block).
# --------------------------------------------------------------------------


class _AddReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, residual: torch.Tensor):
        if x.is_cuda:
            C = require_native("fused_add_relu")
            out = C.add_relu_fwd(x, residual)
        else:
            out = torch.relu(x + residual)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (out,) = ctx.saved_tensors
        if grad_out.is_cuda:
            C = require_native("fused_add_relu")
            gx = C.add_relu_bwd(grad_out.contiguous(memory_format=torch.channels_last)
                                if grad_out.dim() == 4 else grad_out.contiguous(), out)
        else:
            gx = grad_out * (out > 0).to(grad_out.dtype)
        return gx, gx


def fused_add_relu(x: torch.Tensor, residual: torch.Tensor) -> torch.Tensor:
    return _AddReLU.apply(x, residual)


# --------------------------------------------------------------------------
# MaxPool2d NHWC with saved argmax (reference: ResNet stem 3x3 s2 pool;
# SURVEY.md §2.4 "hand kernel + index mask").
# --------------------------------------------------------------------------


class _MaxPool2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, padding):
        C = require_native("max_pool2d")
        out, idx = C.maxpool_fwd(x, kernel, kernel, stride, padding)
        ctx.save_for_backward(idx)
        ctx.geom = (x.shape[2], x.shape[3], kernel, stride, padding)
        return out

    @staticmethod
    def backward(ctx, gout):
        (idx,) = ctx.saved_tensors
        H, W, k, s, p = ctx.geom
        C = require_native("max_pool2d")
        gx = C.maxpool_bwd(gout, idx, H, W, k, k, s, p)
        return gx, None, None, None


def max_pool2d(x: torch.Tensor, kernel: int = 3, stride: int = 2,
               padding: int = 1) -> torch.Tensor:
    if _on_gpu(x):
        return _MaxPool2d.apply(x, kernel, stride, padding)
    return F.max_pool2d(x, kernel, stride, padding)


class MaxPool2d(torch.nn.Module):
    def __init__(self, kernel_size: int = 3, stride: int = 2, padding: int = 1):
        super().__init__()
        self.kernel_size, self.stride, self.padding = kernel_size, stride, padding

    def forward(self, x):
        return max_pool2d(x, self.kernel_size, self.stride, self.padding)


# --------------------------------------------------------------------------
# Fused BatchNorm (+ optional residual add) + ReLU, NHWC, training & eval.
# Matches the reference semantic (SURVEY.md §7 hard-part 3): per-replica
# running stats, never synced across data-parallel replicas.
# --------------------------------------------------------------------------


class _BNAct(torch.autograd.Function):
    @staticmethod
    def forward(
        ctx,
        x: torch.Tensor,
        weight: torch.Tensor,
        bias: torch.Tensor,
        running_mean: torch.Tensor,
        running_var: torch.Tensor,
        training: bool,
        momentum: float,
        eps: float,
        relu: bool,
        residual: Optional[torch.Tensor],
    ):
        C = require_native("batch_norm_act")
        from .conv import take_conv_stats

        conv_part = take_conv_stats(x) if training else None
        x = x.contiguous(memory_format=torch.channels_last)
        out, save_mean, save_invstd = C.bn_act_fwd(
            x, weight, bias, running_mean, running_var,
            training, momentum, eps, relu,
            residual if residual is not None else torch.empty(0, device=x.device, dtype=x.dtype),
            conv_part,
        )
        ctx.save_for_backward(x, weight, save_mean, save_invstd, out)
        ctx.bn_bias = bias
        ctx.relu = relu
        ctx.has_residual = residual is not None
        ctx.training = training
        return out

    @staticmethod
    def backward(ctx, grad_out):
        x, weight, save_mean, save_invstd, out = ctx.saved_tensors
        bias = ctx.bn_bias
        C = require_native("batch_norm_act")
        # direct grads: when the params live in a fused-optimizer flat
        # buffer, the finalize kernel += 's into the G slices and no
        # AccumulateGrad kernels run (ops/fused_optim.FLAT_SLICES).
        from .fused_optim import flat_grad_slice

        gw_sl = flat_grad_slice(weight)
        gb_sl = flat_grad_slice(bias) if bias is not None else None
        direct = gw_sl is not None and gb_sl is not None
        want_gres = ctx.has_residual and ctx.relu
        gx, gw, gb, gres_k = C.bn_act_bwd(
            grad_out.contiguous(memory_format=torch.channels_last),
            x, weight, save_mean, save_invstd, out, ctx.relu, ctx.training,
            gw_sl if direct else None, gb_sl if direct else None, want_gres,
        )
        if direct:
            from ..parallel.bucketing import notify_grad_written

            notify_grad_written(weight)
            notify_grad_written(bias)
            gw = gb = None
        gres = None
        if ctx.has_residual:
            # d(out)/d(residual) = relu-mask * grad_out: a byproduct of the
            # bwd-apply kernel's own mask computation (gres output).
            gres = gres_k if ctx.relu else grad_out
        return gx, gw, gb, None, None, None, None, None, None, gres


def batch_norm_act(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor,
    running_mean: torch.Tensor,
    running_var: torch.Tensor,
    training: bool,
    momentum: float = 0.1,
    eps: float = 1e-5,
    relu: bool = True,
    residual: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """BN (train or eval stats) + optional residual add + optional ReLU.

    GPU: one fused native kernel pair (NHWC). CPU: composed PyTorch ops —
    identical math, used as the oracle.
    """
    if _on_gpu(x):
        return _BNAct.apply(
            x, weight, bias, running_mean, running_var,
            training, momentum, eps, relu, residual,
        )
    # CPU reference path: compute in fp32 (handles mixed bf16-x/fp32-params)
    out = F.batch_norm(
        x.float(), running_mean, running_var, weight.float(), bias.float(),
        training, momentum, eps,
    )
    if residual is not None:
        out = out + residual.float()
    if relu:
        out = torch.relu(out)
    return out.to(x.dtype)


# --------------------------------------------------------------------------
# Global average pool (the reference's AdaptiveMeanPool before the FC —
# SURVEY.md §2.4 "warp reduction").
# --------------------------------------------------------------------------


class _GlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        C = require_native("global_avg_pool")
        ctx.geom = (x.shape[2], x.shape[3])
        return C.gap_fwd(x.contiguous(memory_format=torch.channels_last))

    @staticmethod
    def backward(ctx, gy):
        C = require_native("global_avg_pool")
        H, W = ctx.geom
        return C.gap_bwd(gy.contiguous(), H, W)


def global_avg_pool(x: torch.Tensor) -> torch.Tensor:
    """[N,C,H,W] -> [N,C] mean over H,W."""
    if _on_gpu(x):
        return _GlobalAvgPool.apply(x)
    return x.float().mean(dim=(2, 3)).to(x.dtype)


class GlobalAvgPool(torch.nn.Module):
    def forward(self, x):
        return global_avg_pool(x)
