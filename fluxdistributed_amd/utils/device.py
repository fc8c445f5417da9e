"""Device shims — the portability contract of the reference
(/root/reference/src/utils.jl:1-18: `@device!` and `synchronize` degrade to
no-ops without a GPU; "devices" may be plain integers keying dicts).

A device here is either a torch.device (real HIP device) or any hashable
logical id (ints in the CPU tests — reference test/single_device.jl:144-150).
"""

import contextlib

import torch


def is_real_gpu(dev) -> bool:
    # accepts torch.device OR its string form ("cuda", "cuda:0"): a string
    # device silently classifying as logical left replicas on the CPU and
    # fed host pointers to device kernels (GPU fault, r2)
    if isinstance(dev, torch.device):
        return dev.type == "cuda"
    return isinstance(dev, str) and dev.startswith("cuda")


@contextlib.contextmanager
def device_ctx(dev):
    """`@device! dev do ... end` equivalent: selects the HIP device for the
    scope; no-op for logical/CPU devices."""
    if is_real_gpu(dev) and torch.cuda.is_available():
        with torch.cuda.device(dev):
            yield
    else:
        yield


def synchronize(dev=None):
    if torch.cuda.is_available():
        if is_real_gpu(dev):
            torch.cuda.synchronize(dev)
        elif dev is None:
            torch.cuda.synchronize()


def to_device(x, dev, non_blocking: bool = True):
    """Move tensors / nested containers to a device; identity for logical ids."""
    if not is_real_gpu(dev):
        return x
    if isinstance(x, torch.Tensor):
        return x.to(dev, non_blocking=non_blocking)
    if isinstance(x, (list, tuple)):
        return type(x)(to_device(v, dev, non_blocking) for v in x)
    if isinstance(x, dict):
        return {k: to_device(v, dev, non_blocking) for k, v in x.items()}
    return x


def resolve_devices(n: int = None):
    """N real GPUs if available, else N logical integer devices (CPU)."""
    if torch.cuda.is_available():
        count = torch.cuda.device_count() if n is None else min(n, torch.cuda.device_count())
        devs = [torch.device(f"cuda:{i}") for i in range(count)]
        if n is not None and n > count:
            # logical fan-out on one physical GPU (reference single-GPU trick,
            # test/single_device.jl:127-133) — reuse device 0.
            devs += [torch.device("cuda:0")] * (n - count)
        return devs
    return list(range(n if n is not None else 1))
