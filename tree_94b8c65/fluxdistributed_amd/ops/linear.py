"""Dense (FC) layer on the native MFMA conv kernels — the last hipBLASLt
user in the ResNet step (round-1 profile `Cijk_*` rows; SURVEY.md §2.4
Dense row: the reference's Flux `Dense(512 => 1000)` FC).

MI355X-native design: an FC is a 1x1 conv over a [M, Cin, 1, 1]
channels_last image, so the battle-tested implicit-GEMM kernels do all
three GEMMs. conv_igemm requires output channels % 64 == 0 and the FC has
1000 classes, so out_features is padded to Kp = ceil(N/64)*64 in-house:

  fwd:   y[M][Kp]  = conv_fwd(x4d, wpad)           wpad rows >= N are zero
         y += bpad (bias_add kernel)               -> narrow view [M][N]
  dgrad: dx[M][Cin] = conv_dgrad(dypad, wt)        wt from the _WtArena
  wgrad: dwpad[Kp][Cin] = conv_wgrad(dypad, x4d);  G += first N rows
  bias:  db = colsum(dypad)[:N] direct into flat G

The padded weight/bias copies live in a per-device arena refreshed once
per optimizer step (same freshness marker as the dgrad transpose arena).
All padding/copy kernels are in-tree (csrc/elementwise.hip) — zero aten
kernels in the FC path.
"""

import torch
import torch.nn.functional as F

from .native import load_native, require_native


def _as_nhwc_4d(t2d: torch.Tensor) -> torch.Tensor:
    """[M, C] contiguous -> [M, C, 1, 1] that IS channels_last."""
    M, C = t2d.shape
    return t2d.view(M, 1, 1, C).permute(0, 3, 1, 2)


class _FcPadArena:
    """Per-(device, weight) padded weight/bias: wpad [Kp, Cin] bf16 with
    zero rows >= N, bpad [Kp]. Refreshed when the optimizer marker moves
    (raw-kernel param updates are invisible to version counters) or the
    parameter storage is rebound."""

    def __init__(self):
        self.entries = {}   # id(weight) -> dict

    def get(self, weight, bias):
        from .conv import _WT_MARKER, _arena_for

        ent = self.entries.get(id(weight))
        N, Cin = weight.shape
        Kp = (N + 63) // 64 * 64
        if ent is None:
            wpad = torch.zeros(Kp, Cin, dtype=weight.dtype,
                               device=weight.device)
            bpad = torch.zeros(Kp, dtype=weight.dtype, device=weight.device)
            ent = dict(wpad=wpad, bpad=bpad, key=None, Kp=Kp)
            self.entries[id(weight)] = ent
            # register the padded weight (as a 4D view) for the batched
            # dgrad transpose: wt[(c)][k] with k contiguous
            _arena_for(ent["wpad4"] if False else wpad.view(Kp, Cin, 1, 1)) \
                .register(wpad.view(Kp, Cin, 1, 1))
            ent["wpad4"] = wpad.view(Kp, Cin, 1, 1)
        key = (_WT_MARKER[0], weight._version,
               bias._version if bias is not None else 0,
               weight.data_ptr())
        if ent["key"] != key:
            C = require_native("fc_pad")
            C.pad_rows_bf16_into(ent["wpad"], weight.detach().contiguous())
            if bias is not None:
                ent["bpad"][: N] = bias.detach()
            ent["key"] = key
        return ent


_FC_PAD = _FcPadArena()


class _FdaLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        C = require_native("fc_fwd")
        from .conv import _arena_for

        N, Cin = weight.shape
        ent = _FC_PAD.get(weight, bias)
        Kp = ent["Kp"]
        xc = x.contiguous()
        y = C.conv_igemm_fwd(_as_nhwc_4d(xc),
                             ent["wpad"].view(Kp, 1, 1, Cin).permute(0, 3, 1, 2),
                             1, 1, 0, 0)          # [M, Kp, 1, 1]
        y2 = y.view(x.shape[0], Kp)
        if bias is not None:
            C.bias_add_rows_bf16(y2, ent["bpad"])
        ctx.save_for_backward(xc, weight)
        ctx.bias = bias
        ctx.Kp = Kp
        return y2[:, :N]

    @staticmethod
    def backward(ctx, gy):
        C = require_native("fc_bwd")
        from .conv import _arena_for
        from .fused_optim import flat_grad_slice
        from ..parallel.bucketing import notify_grad_written

        x, weight = ctx.saved_tensors
        bias = ctx.bias
        Kp = ctx.Kp
        N, Cin = weight.shape
        M = x.shape[0]
        dyp = C.pad_rows_bf16(gy.contiguous(), Kp)     # [M, Kp], pad zero
        dyp4 = _as_nhwc_4d(dyp)
        dx = dw = db = None
        ent = _FC_PAD.entries[id(weight)]
        if ctx.needs_input_grad[0]:
            wt = _arena_for(ent["wpad4"]).get(ent["wpad4"])  # [Cin, Kp]
            dx4 = C.conv_igemm_dgrad(dyp4, wt, Cin, 1, 1, 1, 1, 1, 1, 0, 0)
            dx = dx4.permute(0, 2, 3, 1).reshape(M, Cin)
        if ctx.needs_input_grad[1]:
            ws = C.conv_igemm_wgrad(dyp4, _as_nhwc_4d(x), 1, 1, 1, 1, 0, 0)
            g_sl = flat_grad_slice(weight)
            if g_sl is not None:
                C.grad_accum_bf16(g_sl, ws.reshape(-1)[: N * Cin])
                notify_grad_written(weight)
            else:
                dw = ws[:N].to(torch.bfloat16)
        if bias is not None and ctx.needs_input_grad[2]:
            gb_sl = flat_grad_slice(bias)
            if gb_sl is not None:
                C.colsum_accum_bf16(gb_sl, dyp)
                notify_grad_written(bias)
            else:
                db = dyp.float().sum(dim=0)[:N].to(torch.bfloat16)
        return dx, dw, db


def _fda_linear_supported(x, weight) -> bool:
    import os

    if os.environ.get("FLUXDIST_FC", "") == "lib":
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16 and x.dim() == 2):
        return False
    return weight.shape[1] % 64 == 0 and load_native() is not None


class FdaLinear(torch.nn.Linear):
    """nn.Linear dispatched to the native MFMA path for bf16 GPU tensors;
    parameter shapes/layout unchanged (checkpoint-compatible)."""

    def forward(self, x):
        if _fda_linear_supported(x, self.weight):
            return _FdaLinear.apply(x, self.weight, self.bias)
        return F.linear(x, self.weight, self.bias)
