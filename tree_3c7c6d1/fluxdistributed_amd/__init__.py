"""fluxdistributed_amd — an MI355X-native data-parallel training framework.

A from-scratch re-design of the capabilities of DhairyaLGandhi/FluxDistributed.jl
(reference mounted at /root/reference) for AMD Instinct MI355X (gfx950):

- PyTorch-ROCm is the tensor substrate; hot ops are hand-written HIP/CDNA4
  kernels (fused logit-cross-entropy, fused BN+ReLU, residual add+ReLU,
  multi-tensor fused SGD/Momentum and Adam) built for gfx950 only.
- Data parallelism comes in the reference's two flavors, both re-expressed
  natively:
  * task-DDP   (reference src/ddp_tasks.jl): one process, N devices, thread
    per device, explicit gradient buffer + reduce — `prepare_training`/`train`.
  * process-DDP (reference src/sync.jl + bin/driver.jl): one process per GPU,
    bucketed all-reduce over RCCL/xGMI overlapped with backward — `DDPModel`.
- The async minibatch pipeline (reference's Flux-fork DataLoader with
  `buffersize`) is a prefetching loader staging pinned host batches onto a
  side HIP stream.

Public API (parity with reference src/FluxDistributed.jl:11-12 exports):
  prepare_training, train, minibatch, train_solutions, sync_buffer (the
  `syncgrads` equivalent lives in parallel.process_ddp).
"""

__version__ = "0.1.0"

from .parallel.gradtree import (  # noqa: F401
    destruct,
    accum_,
    dodiv_,
    sync_buffer,
    markbuffer_,
    getbuffer_,
    ensure_synced,
    check_nans,
)
from .parallel.task_ddp import prepare_training, train, train_step  # noqa: F401
from .data.imagenet import minibatch, train_solutions, labels  # noqa: F401
from .utils.metrics import topkaccuracy  # noqa: F401

from . import models, ops, parallel, data, utils  # noqa: F401
