"""Direct-grad flush scheduling: fp32 wgrad workspaces -> bf16 flat-G.

With a DDP bucketer attached every flush runs immediately (the bucket's
all-reduce launches as soon as its last grad lands, so the grad must be
written then). Without one (single-GPU / task-DDP replica), the ~36
per-layer flush launches of a ResNet-34 backward are queued and executed
as ONE batched kernel at backward end via autograd's engine callback —
per-layer launches measured ~180 us/step of mostly launch+tail overhead.

The queue holds references to the workspace tensors so the caching
allocator cannot recycle them before the batched kernel reads them.
"""

import threading

import torch

from .native import require_native

_TLS = threading.local()
_META_CACHE = {}


def _bucketer_active() -> bool:
    from ..parallel.bucketing import _NOTIFY

    return bool(_NOTIFY)


# Replica threads flag their backwards as non-deferrable (task_ddp sets
# this around train_step); autograd device workers inherit nothing from
# the launching thread, so the flag lives in a process-global counter.
_NO_DEFER = [0]


def push_no_defer():
    _NO_DEFER[0] += 1


def pop_no_defer():
    _NO_DEFER[0] -= 1


def _deferral_ok() -> bool:
    return _NO_DEFER[0] == 0


def queue_or_flush(param, g_sl: torch.Tensor, ws_flat: torch.Tensor) -> None:
    """G slice += cast(ws): now (DDP / replica threads) or batched at
    backward end (single main-thread training).

    The deferred path is gated to backwards whose GRAPHS were launched
    from the main thread: with several task-DDP replica threads running
    concurrent backwards, all device nodes execute interleaved on ONE
    autograd device worker, and one graph's end-callback would flush the
    other graph's still-accumulating queue entries — correct but
    unverifiable ordering; replicas keep the per-layer immediate flush.
    """
    C = require_native("grad_accum")
    if _bucketer_active() or not _deferral_ok():
        from ..parallel.bucketing import notify_grad_written

        C.grad_accum_bf16(g_sl, ws_flat)
        notify_grad_written(param)
        return
    q = getattr(_TLS, "queue", None)
    if q is None:
        q = _TLS.queue = []
    if not q:
        torch.autograd.Variable._execution_engine.queue_callback(_flush)
    q.append((g_sl, ws_flat))


def _flush() -> None:
    q = getattr(_TLS, "queue", None)
    _TLS.queue = []
    if not q:
        return
    C = require_native("grad_accum")
    if len(q) == 1:
        C.grad_accum_bf16(q[0][0], q[0][1])
        return
    key = tuple((g.data_ptr(), w.data_ptr(), g.numel()) for g, w in q)
    ent = _META_CACHE.get(key)
    if ent is None:
        dev = q[0][0].device
        mk = lambda vals: torch.tensor(vals, dtype=torch.long, device=dev)
        ent = (mk([g.data_ptr() for g, _ in q]),
               mk([w.data_ptr() for _, w in q]),
               mk([g.numel() for g, _ in q]),
               max(g.numel() for g, _ in q))
        if len(_META_CACHE) > 64:
            _META_CACHE.clear()
        _META_CACHE[key] = ent
    C.grad_accum_batch(*ent)
