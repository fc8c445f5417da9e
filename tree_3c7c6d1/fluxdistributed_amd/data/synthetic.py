"""Synthetic data for benchmarking (synthetic 224×224×3,
random-init weights — there is no network for datasets in this image).

Batches are pregenerated once into a small pinned pool and cycled, so the
benchmark measures the training engine, not the host RNG.
"""

from typing import Tuple

import torch


class SyntheticBatcher:
    def __init__(self, batch_size: int, num_classes: int = 1000,
                 size: int = 224, dtype=torch.float32, pool: int = 4,
                 channels_last: bool = True, pin: bool = False, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.pool = []
        for _ in range(pool):
            x = torch.randn(batch_size, 3, size, size, generator=g, dtype=torch.float32)
            x = x.to(dtype)
            if channels_last:
                x = x.contiguous(memory_format=torch.channels_last)
            y = torch.randint(0, num_classes, (batch_size,), generator=g)
            if pin and torch.cuda.is_available():
                x, y = x.pin_memory(), y.pin_memory()
            self.pool.append((x, y))
        self.i = 0

    def __call__(self, nsamples: int = None) -> Tuple[torch.Tensor, torch.Tensor]:
        x, y = self.pool[self.i % len(self.pool)]
        self.i += 1
        return x, y
