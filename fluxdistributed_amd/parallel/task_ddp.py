"""Task-based DDP: one process, N devices, thread per device.

MI355X re-expression of the reference's live orchestrator
(/root/reference/src/ddp_tasks.jl:174-289): model replicated per device,
per-device threads run forward+backward, gradients are published to a
buffer, reduced to their mean, broadcast back, and each replica takes an
identical optimizer step — fully synchronous SGD, math-equivalent to
large-batch.

Differences from the reference (deliberate, MI355X-first):
- replicas are keyed by index, not device object (a logical fan-out may put
  two replicas on one GPU — reference test/single_device.jl:127-133);
- the buffer lives on the first replica's device ("HOST" in the reference,
  ddp_tasks.jl:250) and buffer allocation is completed before return —
  the reference's unawaited alloc tasks are a known latent race
  (SURVEY.md "known reference bugs");
- `num_missed` OOM counter actually increments (reference bug
  ddp_tasks.jl:180,240);
- per-stage timers + images/sec (observability the reference lacks,
  SURVEY.md §5.1).

For >1 real GPU prefer the process-DDP path (process_ddp.py, RCCL over
xGMI); this path remains fully functional on N GPUs in one process and is
the CPU-testable orchestration oracle.
"""

import threading
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Sequence

import torch

from .gradtree import (
    destruct, grads_of, markbuffer_, getbuffer_, sync_buffer,
)
from ..utils.device import device_ctx, synchronize, to_device, is_real_gpu
from ..utils.timers import StageTimers, Throughput
from ..utils.metrics import topkaccuracy
from ..utils.logging import get_logger

log = get_logger(__name__)


@dataclass
class Replica:
    index: int
    device: object
    model: torch.nn.Module
    optimizer: torch.optim.Optimizer
    loader: Optional[object] = None


@dataclass
class TrainState:
    replicas: List[Replica]
    buffer: Dict[int, dict]          # replica index -> GradTree on HOST device
    host_device: object
    cycles: int = 0
    num_missed: int = 0
    timers: StageTimers = field(default_factory=StageTimers)
    throughput: Throughput = field(default_factory=Throughput)


def prepare_training(
    model: torch.nn.Module,
    data: Optional[Callable[[int], object]],
    devices: Sequence,
    opt_factory: Callable[[object], torch.optim.Optimizer],
    nsamples: int = 32,
    buffersize: int = 5,
    loader_factory: Optional[Callable] = None,
) -> TrainState:
    """Replicate model + optimizer + loader per device; allocate grad buffers.

    Mirrors /root/reference/src/ddp_tasks.jl:249-289. `data` is a callable
    `data(nsamples) -> (x, y)` batch sampler (the reference's
    `minibatch(data_tree, shard; nsamples)` closure), OR a sequence of one
    such callable per device — the reference shards the key into disjoint
    per-device chunks (ddp_tasks.jl:257-258), so real datasets should pass
    per-replica samplers built over `data.imagenet.shard_key` shards
    (train.py does). Pass None to drive training with explicit batches.
    """
    from ..data.loader import PrefetchLoader

    host = devices[0]
    if data is not None and not callable(data):
        per_replica = list(data)
        if len(per_replica) != len(devices):
            raise ValueError(
                f"got {len(per_replica)} data callables for {len(devices)} devices")
    else:
        per_replica = [data] * len(devices)
    replicas: List[Replica] = []
    for i, dev in enumerate(devices):
        import copy

        m = copy.deepcopy(model)
        if is_real_gpu(dev):
            m = m.to(dev)
        opt = opt_factory(m)
        loader = None
        if per_replica[i] is not None:
            fn = per_replica[i]
            fac = loader_factory or (lambda f, d: PrefetchLoader(f, device=d, buffersize=buffersize))
            loader = fac(lambda n=nsamples, _fn=fn: _fn(n), dev)
        replicas.append(Replica(i, dev, m, opt, loader))

    buffer: Dict[int, dict] = {}
    with device_ctx(host):
        for r in replicas:
            buffer[r.index] = destruct(
                model, device=host if is_real_gpu(host) else None
            )
    synchronize(host)  # buffers fully allocated before training starts
    return TrainState(replicas=replicas, buffer=buffer, host_device=host)


def train_step(loss_fn, buffer, replica: Replica, x, y):
    """Forward+backward on one replica, then publish grads to its buffer slot
    (/root/reference/src/ddp_tasks.jl:80-84)."""
    from ..ops.gradflush import pop_no_defer, push_no_defer

    with device_ctx(replica.device):
        replica.optimizer.zero_grad()
        out = replica.model(x)
        loss = loss_fn(out, y)
        # concurrent replica backwards share the autograd device worker:
        # direct-grad flushes must stay per-layer immediate (gradflush.py)
        push_no_defer()
        try:
            loss.backward()
        finally:
            pop_no_defer()
        markbuffer_(buffer[replica.index], grads_of(replica.model))
        synchronize(replica.device)
    return loss.detach()


def update(replica: Replica, final):
    """Write reduced grads into the replica's grad memory and step
    (/root/reference/src/ddp_tasks.jl:163-172)."""
    with device_ctx(replica.device):
        getbuffer_(grads_of(replica.model), final)
        synchronize(replica.device)
        replica.optimizer.step()
        synchronize(replica.device)


def train(
    loss_fn,
    state: TrainState,
    steps: int,
    val: Optional[tuple] = None,
    sched: Optional[Callable[[int], None]] = None,
    log_every: int = 10,
    val_every: int = 50,
    batches: Optional[Callable[[int], Sequence]] = None,
    on_cycle_end: Optional[Callable[[int, TrainState], None]] = None,
):
    """Main synchronous loop (/root/reference/src/ddp_tasks.jl:174-247).

    `batches(j)` may supply the per-replica batch list for step j (used by
    tests to drive exact shards); otherwise each replica's loader is used.
    """
    reps = state.replicas
    for j in range(steps):
        state.cycles += 1
        if batches is not None:
            mbs = batches(j)
        else:
            mbs = [next(r.loader) for r in reps]

        losses: List[Optional[torch.Tensor]] = [None] * len(reps)
        errors: List[Optional[BaseException]] = [None] * len(reps)

        with state.timers.stage("fwd_bwd"):
            def run(i, r, xb, yb):
                try:
                    losses[i] = train_step(loss_fn, state.buffer, r, xb, yb)
                except BaseException as e:  # noqa: BLE001 — surfaced below
                    errors[i] = e

            threads = [
                threading.Thread(target=run, args=(i, r, xb, yb), daemon=True)
                for i, (r, (xb, yb)) in enumerate(zip(reps, mbs))
            ]
            for t in threads:
                t.start()
            for t in threads:
                t.join()  # barrier — wait.(gs) at ddp_tasks.jl:208

        oom = [e for e in errors if e is not None and _is_oom(e)]
        hard = [e for e in errors if e is not None and not _is_oom(e)]
        if hard:
            raise hard[0]
        if oom:
            # per-batch OOM tolerance (ddp_tasks.jl:230-238) with a WORKING
            # missed-batch counter (reference bug: declared, never bumped).
            state.num_missed += 1
            log.warning("OOM on step %d — skipping batch (missed=%d)",
                        j, state.num_missed)
            for r in reps:
                if is_real_gpu(r.device):
                    torch.cuda.empty_cache()
            continue

        with state.timers.stage("allreduce"):
            final = sync_buffer(state.buffer, average=True)

        with state.timers.stage("optimizer"):
            uth = [
                threading.Thread(target=update, args=(r, final), daemon=True)
                for r in reps
            ]
            for t in uth:
                t.start()
            for t in uth:
                t.join()

        n_imgs = sum(int(xb.shape[0]) for xb, _ in mbs)
        state.throughput.add(n_imgs)

        if log_every and (j + 1) % log_every == 0:
            lv = [float(l) for l in losses if l is not None]
            log.info("cycle %d loss=%.4f imgs/s=%.1f", state.cycles,
                     sum(lv) / max(len(lv), 1), state.throughput.rate())
        if val is not None and val_every and (j + 1) % val_every == 0:
            log_loss_and_acc(loss_fn, reps[0], val)
        if sched is not None:
            sched(state.cycles)
        if on_cycle_end is not None:
            on_cycle_end(state.cycles, state)

    return [(r.device, r.model) for r in reps]


def log_loss_and_acc(loss_fn, replica: Replica, val, ks=(1, 5, 10)):
    """Eval loss + top-k accuracy on the first replica
    (/root/reference/src/ddp_tasks.jl:128-140)."""
    xv, yv = val
    model = replica.model
    was_training = model.training
    model.eval()
    with torch.no_grad(), device_ctx(replica.device):
        xd = to_device(xv, replica.device)
        out = model(xd).float().cpu()
        loss = float(loss_fn(out, yv))
        accs = {k: topkaccuracy(out, yv, k=k) for k in ks}
    if was_training:
        model.train()
    log.info("val loss=%.4f " + " ".join(f"top{k}={accs[k]:.3f}" for k in ks), loss)
    return loss, accs


def _is_oom(e: BaseException) -> bool:
    return isinstance(e, torch.cuda.OutOfMemoryError) or (
        isinstance(e, RuntimeError) and "out of memory" in str(e).lower()
    )
