"""Fused flat-buffer optimizers (reference: the Optimisers.jl Momentum/ADAM
updates applied leaf-wise at /root/reference/src/ddp_tasks.jl:168 and
src/overloads.jl:1-12 — here re-designed MI355X-first).

Instead of ~110 per-leaf elementwise launches, all parameters live in ONE
contiguous flat buffer per (device, dtype) group:

  P (param dtype, what the model computes with)
  G (param dtype, gradients — autograd accumulates into views of it, so a
     data-parallel all-reduce is a single flat collective over G)
  M (fp32 master copy, only when P is low precision)
  V/S (fp32 optimizer state)

One kernel launch updates the whole model (~22M elems for ResNet-34): pure
HBM-bound streaming, vectorized 16B/lane (guide Appendix B elementwise).
"""


from typing import List

import torch

from .native import require_native


def _bump_wt_marker():
    # raw-kernel param writes are invisible to autograd version counters;
    # invalidate the dgrad transposed-weight arena (ops/conv.py)
    from .conv import bump_conv_wt_marker

    bump_conv_wt_marker()


# id(param) -> (flat G tensor, offset, numel): lets backward ops write
# gradients straight into the flat buffer (no AccumulateGrad kernel) —
# see ops/conv.py and ops/functional.py "direct grad" paths.
FLAT_SLICES = {}


def flat_grad_slice(param):
    """Contiguous flat-G slice backing param.grad, or None."""
    ent = FLAT_SLICES.get(id(param))
    if ent is None:
        return None
    G, off, n = ent
    return G[off : off + n]


class _FlatGroup:
    """All parameters of one (device, dtype) flattened into shared storage."""

    ALIGN = 64  # elements; keeps every param slice 128B-aligned for vector IO

    def __init__(self, params: List[torch.nn.Parameter]):
        self.params = params
        dev, dt = params[0].device, params[0].dtype
        self.device, self.dtype = dev, dt
        offs, total = [], 0
        for p in params:
            offs.append(total)
            total += (p.numel() + self.ALIGN - 1) // self.ALIGN * self.ALIGN
        self.offsets, self.numel = offs, total
        self.P = torch.zeros(total, device=dev, dtype=dt)
        self.G = torch.zeros(total, device=dev, dtype=dt)
        for p, off in zip(params, offs):
            # Rebind: the Parameter stays a leaf; its storage is now the flat
            # buffer. channels_last params keep their physical layout.
            v = self._view_like(self.P, off, p.data)
            v.copy_(p.data)
            p.data = v
            p.grad = self._view_like(self.G, off, p.data)
            FLAT_SLICES[id(p)] = (self.G, off, p.numel())
        self.master = self.P.float() if dt != torch.float32 else None

    @staticmethod
    def _view_like(flat, off, like):
        v = flat[off : off + like.numel()]
        if like.is_contiguous(memory_format=torch.channels_last) and like.dim() == 4 \
                and not like.is_contiguous():
            return v.view(like.shape[0], like.shape[2], like.shape[3], like.shape[1]) \
                .permute(0, 3, 1, 2)
        return v.view(like.shape)

    def grad_views(self):
        return [p.grad for p in self.params]


def _flatten_param_groups(params) -> List[_FlatGroup]:
    params = [p for p in params if p.requires_grad]
    if not params:
        return []
    by_key = {}
    for p in params:
        by_key.setdefault((p.device, p.dtype), []).append(p)
    return [_FlatGroup(ps) for ps in by_key.values()]


class _FlatOptimizer(torch.optim.Optimizer):
    """Base: flattens params, exposes flat G for DDP, state_dict round-trips."""

    def __init__(self, params, defaults):
        params = list(params)
        super().__init__(params, defaults)
        flat_params = [p for g in self.param_groups for p in g["params"]]
        self.groups = _flatten_param_groups(flat_params)

    @torch.no_grad()
    def zero_grad(self, set_to_none: bool = False):
        # set_to_none must stay False: autograd accumulates into the flat views.
        for g in self.groups:
            g.G.zero_()

    def flat_grads(self) -> List[torch.Tensor]:
        return [g.G for g in self.groups]

    @torch.no_grad()
    def refresh_master(self):
        """Re-seed fp32 masters from current (possibly just-loaded) params."""
        for g in self.groups:
            if g.master is not None:
                g.master.copy_(g.P.float())

    def state_dict(self):
        d = super().state_dict()
        d["flat_state"] = [
            {
                "master": (g.master.clone() if g.master is not None else None),
                **{k: v.clone() for k, v in self._flat_buffers(g).items()},
            }
            for g in self.groups
        ]
        return d

    def load_state_dict(self, d):
        flat = d.pop("flat_state", None)
        super().load_state_dict(d)
        if flat is not None:
            for g, s in zip(self.groups, flat):
                if g.master is not None and s["master"] is not None:
                    g.master.copy_(s["master"])
                for k, v in self._flat_buffers(g).items():
                    v.copy_(s[k])

    def _flat_buffers(self, g):  # pragma: no cover - overridden
        return {}


class FusedSGDMomentum(_FlatOptimizer):
    """SGD with (heavy-ball) momentum — reference flagship optimizer
    Momentum(0.01, 0.9) (/root/reference/README.md:37-38)."""

    def __init__(self, params, lr=0.01, momentum=0.9, weight_decay=0.0, nesterov=False):
        super().__init__(params, dict(lr=lr, momentum=momentum,
                                      weight_decay=weight_decay, nesterov=nesterov))
        for g in self.groups:
            g.V = torch.zeros(g.numel, device=g.device, dtype=torch.float32)

    def _flat_buffers(self, g):
        return {"V": g.V}

    @torch.no_grad()
    def step(self, closure=None):
        hp = self.param_groups[0]
        lr, mom, wd = hp["lr"], hp["momentum"], hp["weight_decay"]
        nesterov = hp["nesterov"]
        for g in self.groups:
            if g.P.is_cuda:
                C = require_native("fused_sgd")
                C.sgd_step(g.P, g.G, g.master if g.master is not None else g.P,
                           g.V, lr, mom, wd, nesterov)
                _bump_wt_marker()
            else:
                master = g.master if g.master is not None else g.P
                grad = g.G.float()
                if wd:
                    grad = grad.add(master, alpha=wd)
                g.V.mul_(mom).add_(grad)
                upd = grad.add(g.V, alpha=mom) if nesterov else g.V
                master.add_(upd, alpha=-lr)
                if g.master is not None:
                    g.P.copy_(master.to(g.dtype))
        return None


class FusedAdam(_FlatOptimizer):
    """Adam — reference process-DDP optimizer (/root/reference/bin/driver.jl:27)."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=0.0):
        super().__init__(params, dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay))
        for g in self.groups:
            g.V = torch.zeros(g.numel, device=g.device, dtype=torch.float32)  # m1
            g.S = torch.zeros(g.numel, device=g.device, dtype=torch.float32)  # m2
        self._step = 0

    def _flat_buffers(self, g):
        return {"V": g.V, "S": g.S}

    @torch.no_grad()
    def step(self, closure=None):
        hp = self.param_groups[0]
        lr, (b1, b2), eps, wd = hp["lr"], hp["betas"], hp["eps"], hp["weight_decay"]
        self._step += 1
        bc1 = 1.0 - b1 ** self._step
        bc2 = 1.0 - b2 ** self._step
        for g in self.groups:
            if g.P.is_cuda:
                C = require_native("fused_adam")
                C.adam_step(g.P, g.G, g.master if g.master is not None else g.P,
                            g.V, g.S, lr, b1, b2, eps, wd, bc1, bc2)
                _bump_wt_marker()
            else:
                master = g.master if g.master is not None else g.P
                grad = g.G.float()
                if wd:
                    grad = grad.add(master, alpha=wd)
                g.V.mul_(b1).add_(grad, alpha=1 - b1)
                g.S.mul_(b2).addcmul_(grad, grad, value=1 - b2)
                denom = (g.S / bc2).sqrt_().add_(eps)
                master.addcdiv_(g.V / bc1, denom, value=-lr)
                if g.master is not None:
                    g.P.copy_(master.to(g.dtype))
        return None

    def state_dict(self):
        d = super().state_dict()
        d["step_count"] = self._step
        return d

    def load_state_dict(self, d):
        self._step = d.pop("step_count", 0)
        super().load_state_dict(d)
