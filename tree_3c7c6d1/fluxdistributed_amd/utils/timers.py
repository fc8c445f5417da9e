"""Per-stage timers + throughput counters — observability the reference
lacks entirely (print/@info only, no timers, no imgs/sec).

GPU timing uses events only when explicitly asked (sync=False keeps the
timers non-invasive on the hot path: wall-clock around already-synchronous
stages).
"""

import contextlib
import time
from collections import defaultdict


class StageTimers:
    def __init__(self):
        self.totals = defaultdict(float)
        self.counts = defaultdict(int)

    @contextlib.contextmanager
    def stage(self, name: str):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self.totals[name] += time.perf_counter() - t0
            self.counts[name] += 1

    def summary(self):
        return {
            k: {"total_s": self.totals[k], "count": self.counts[k],
                "mean_ms": 1e3 * self.totals[k] / max(1, self.counts[k])}
            for k in self.totals
        }

    def reset(self):
        self.totals.clear()
        self.counts.clear()


class Throughput:
    """Steady-state items/sec. The FIRST add() only starts the clock and its
    items are deliberately NOT counted — it is the warmup batch (kernel
    compilation, allocator growth); rate() therefore reports steady-state
    throughput over batches 2..n."""

    def __init__(self):
        self.n = 0
        self.t0 = None

    def add(self, n: int):
        if self.t0 is None:
            self.t0 = time.perf_counter()
            return
        self.n += n

    def rate(self) -> float:
        if self.t0 is None or self.n == 0:
            return 0.0
        dt = time.perf_counter() - self.t0
        return self.n / dt if dt > 0 else 0.0
