"""Async prefetching loader with H2D staging on a side HIP stream.

Re-expresses the reference's Flux-fork DataLoader contract — a loader
FUNCTION plus a `buffersize` prefetch depth (/root/reference/src/
ddp_tasks.jl:277-284; the fork's channel-backed loader):
a background thread produces host batches into a bounded queue
(buffersize), batches are pinned, and the device copy is issued
asynchronously on a dedicated side stream (pinned hipMemcpyAsync), double-
buffered so step k's H2D overlaps step k-1's compute.
"""

import queue
import threading
from typing import Callable, Optional

import torch

from ..utils.device import is_real_gpu


class PrefetchLoader:
    """Iterable: next() yields device-resident (x, y) batches.

    make_batch: () -> (x, y) host tensors (the loader function).
    """

    _SENTINEL = object()

    def __init__(self, make_batch: Callable[[], tuple], device=None,
                 buffersize: int = 5, pin: Optional[bool] = None):
        self.make_batch = make_batch
        self.device = device
        self.gpu = is_real_gpu(device)
        self.pin = self.gpu if pin is None else pin
        self.q: "queue.Queue" = queue.Queue(maxsize=max(1, buffersize))
        self.copy_stream = torch.cuda.Stream(device) if self.gpu else None
        self._stop = threading.Event()
        self._exc = None
        self._prev_batch = None  # keeps pinned source alive through the copy
        self._thread = threading.Thread(target=self._producer, daemon=True)
        self._thread.start()

    def _producer(self):
        try:
            while not self._stop.is_set():
                x, y = self.make_batch()
                if self.pin and not x.is_pinned():
                    x = x.pin_memory()
                    y = y.pin_memory()
                self.q.put((x, y))
        except BaseException as e:  # noqa: BLE001
            self._exc = e
            self.q.put(self._SENTINEL)

    def __iter__(self):
        return self

    def __next__(self):
        item = self.q.get()
        if item is self._SENTINEL:
            raise RuntimeError("loader thread failed") from self._exc
        x, y = item
        if not self.gpu:
            return x, y
        with torch.cuda.stream(self.copy_stream):
            xd = x.to(self.device, non_blocking=True)
            yd = y.to(self.device, non_blocking=True)
        evt = torch.cuda.Event()
        evt.record(self.copy_stream)
        cur = torch.cuda.current_stream(self.device)
        cur.wait_event(evt)
        # xd/yd are ALLOCATED on copy_stream but consumed on the current
        # stream: record_stream is the allocator contract for that — without
        # it, freeing xd returns the block to copy_stream's pool and a later
        # prefetch H2D can overwrite it while the consumer's kernels still
        # read it. Latent under eager timing; exposed as input aliasing by
        # back-to-back hipGraph replays (tools/graph_parity.py history).
        xd.record_stream(cur)
        yd.record_stream(cur)
        # pinned host tensors must outlive the async copy
        self._prev_batch = (x, y)
        return xd, yd

    def close(self):
        self._stop.set()
        try:
            while True:
                self.q.get_nowait()
        except queue.Empty:
            pass
        self._thread.join(timeout=2.0)

    def __del__(self):
        try:
            self._stop.set()
        except Exception:
            pass
