"""Independent-ensemble ("pick best") trainer.

Parity with the reference's experimental per-worker scheme
(/root/reference/src/test.jl:26-63, dead code there — not `include`d by the
module): each replica trains INDEPENDENTLY for a cycle; at
the end of each cycle the replica with the lowest validation loss is
broadcast to all others ("pick best" instead of gradient averaging), with
the reference's LR/5-every-10-cycles schedule available as a helper.

MI355X-native notes: replicas here are (device, model, optimizer) tuples
exactly like task_ddp.Replica; the broadcast is a flat state_dict copy
(device-to-device over xGMI when replicas sit on different GPUs).
"""

import copy
from typing import Callable, List, Optional, Sequence

import torch

from .task_ddp import Replica, log_loss_and_acc
from ..utils.device import device_ctx, to_device
from ..utils.logging import get_logger

log = get_logger(__name__)


def lr_div5_every10(optimizer_lrs: Sequence, cycle: int) -> None:
    """The reference's schedule: LR /= 5 every 10 cycles (test.jl:50)."""
    if cycle > 0 and cycle % 10 == 0:
        for opt in optimizer_lrs:
            for g in opt.param_groups:
                g["lr"] /= 5.0


def _broadcast_state(src: Replica, dst: Replica) -> None:
    with torch.no_grad():
        sd = src.model.state_dict()
        for name, p in dst.model.state_dict().items():
            p.copy_(to_device(sd[name], p.device))


def train_ensemble(
    loss_fn: Callable,
    replicas: List[Replica],
    val: tuple,
    cycles: int,
    steps_per_cycle: int,
    batches: Optional[Callable[[int, int], Sequence]] = None,
    sched: Optional[Callable[[Sequence, int], None]] = lr_div5_every10,
):
    """Run `cycles` rounds; each round trains every replica independently for
    `steps_per_cycle` steps, then keeps the min-val-loss model
    (test.jl:58 `findmin`) and broadcasts it to every replica.

    `batches(cycle, replica_index)` returns an iterable of (x, y); defaults
    to each replica's own loader. Returns (best_index_history, replicas).
    """
    history = []
    for cyc in range(cycles):
        for i, r in enumerate(replicas):
            it = (batches(cyc, i) if batches is not None
                  else (next(r.loader) for _ in range(steps_per_cycle)))
            with device_ctx(r.device):
                for x, y in it:
                    r.optimizer.zero_grad()
                    loss = loss_fn(r.model(to_device(x, r.device)),
                                   to_device(y, r.device))
                    loss.backward()
                    r.optimizer.step()
        # pick best on validation loss
        losses = []
        for r in replicas:
            vl, _ = log_loss_and_acc(loss_fn, r, val, ks=(1,))
            losses.append(vl)
        best = min(range(len(replicas)), key=lambda i: losses[i])
        history.append(best)
        log.info("ensemble cycle %d: best replica %d (val %.4f)",
                 cyc, best, losses[best])
        for i, r in enumerate(replicas):
            if i != best:
                _broadcast_state(replicas[best], r)
        if sched is not None:
            sched([r.optimizer for r in replicas], cyc + 1)
    return history, replicas


def make_replicas(model: torch.nn.Module, devices: Sequence,
                  opt_factory: Callable) -> List[Replica]:
    """Independent deep copies (no shared grads — unlike prepare_training)."""
    reps = []
    for i, dev in enumerate(devices):
        m = copy.deepcopy(model)
        if isinstance(dev, torch.device) or (isinstance(dev, str) and "cuda" in str(dev)):
            # let placement failures propagate: a silently-CPU replica whose
            # Replica.device says cuda trains on the wrong device
            m = m.to(dev)
        reps.append(Replica(i, dev, m, opt_factory(m)))
    return reps
