"""Training engines: eager step and hipGraph-captured step.

The MI355X replacement for the reference's per-step Julia task scheduling:
the whole fwd+bwd+optimizer step is captured once into a hipGraph
(torch.cuda.CUDAGraph == hipGraph on ROCm) and replayed every iteration —
launch-bound gaps between the ~200 kernels of a ResNet step disappear.

Works because the entire step is in-place over stable storage:
- parameters/grads/optimizer state live in flat buffers (ops/fused_optim),
- the loader copies each batch into static input tensors,
- BN running stats update in place.

The DDP all-reduce can be captured too (RCCL supports hipGraph capture);
`GraphedTrainStep` captures the non-overlapped flat all-reduce. For
overlapped bucketed comm use the eager step (bucket hooks are host-side).
"""

from typing import Callable

import torch


class EagerTrainStep:
    """Eager step with rocprof-visible stage markers: torch.cuda.nvtx maps
    to roctx ranges on ROCm, so `rocprofv3 --marker-trace` shows
    fwd/bwd/allreduce/step spans (observability the reference lacks,
    SURVEY.md §5.1)."""

    def __init__(self, model, optimizer, loss_fn: Callable, ddp=None,
                 markers: bool = False):
        self.model = model
        self.optimizer = optimizer
        self.loss_fn = loss_fn
        self.ddp = ddp
        self.markers = markers and torch.cuda.is_available()

    def __call__(self, x, y):
        nvtx = torch.cuda.nvtx if self.markers else None
        if nvtx: nvtx.range_push("fwd")
        out = self.model(x)
        loss = self.loss_fn(out, y)
        if nvtx: nvtx.range_pop(); nvtx.range_push("bwd")
        self.optimizer.zero_grad()
        loss.backward()
        if nvtx: nvtx.range_pop(); nvtx.range_push("allreduce")
        if self.ddp is not None:
            self.ddp.finalize_backward()
        if nvtx: nvtx.range_pop(); nvtx.range_push("step")
        self.optimizer.step()
        if nvtx: nvtx.range_pop()
        return loss


class GraphedTrainStep:
    """Capture fwd+bwd(+flat all-reduce)+optimizer into one hipGraph.

    Call with same-shaped batches; they are copied into the capture's
    static input tensors and the graph is replayed.
    """

    def __init__(self, model, optimizer, loss_fn: Callable,
                 example_x: torch.Tensor, example_y: torch.Tensor,
                 ddp=None, warmup: int = 3):
        assert example_x.is_cuda, "graph capture needs GPU tensors"
        self.optimizer = optimizer
        self.static_x = example_x.clone()
        self.static_y = example_y.clone()
        self.ddp = ddp
        if ddp is not None and ddp.overlap:
            # hooks are host-side callbacks; capture uses the flat collective
            ddp.bucketer.detach()
            ddp.overlap = False

        def whole_step():
            out = model(self.static_x)
            loss = loss_fn(out, self.static_y)
            optimizer.zero_grad()
            loss.backward()
            if ddp is not None:
                ddp.finalize_backward()
            optimizer.step()
            return loss

        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup):
                loss = whole_step()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_loss = whole_step()

    def __call__(self, x, y):
        self.static_x.copy_(x, non_blocking=True)
        self.static_y.copy_(y, non_blocking=True)
        self.graph.replay()
        return self.static_loss


def make_train_step(model, optimizer, loss_fn, example_batch=None, ddp=None,
                    use_graph: bool = True, warmup: int = 3,
                    markers: bool = False):
    """Pick graph or eager. Graph requires CUDA + an example batch; falls
    back to eager if capture fails (e.g. an op is not capture-safe)."""
    if use_graph and example_batch is not None and example_batch[0].is_cuda:
        try:
            return GraphedTrainStep(model, optimizer, loss_fn,
                                    example_batch[0], example_batch[1],
                                    ddp=ddp, warmup=warmup)
        except Exception as e:  # noqa: BLE001
            import warnings

            warnings.warn(f"hipGraph capture failed ({e}); using eager step")
    return EagerTrainStep(model, optimizer, loss_fn, ddp=ddp,
                          markers=markers)
