"""Hand-written HIP/CDNA4 kernels (gfx950) with CPU reference fallbacks.

Dispatch policy (deliberate, see repo root README):
- CPU tensors      -> pure-PyTorch reference implementation (used by CI and
                      as the numerics oracle for every HIP kernel).
- GPU/HIP tensors  -> the in-tree native extension `fluxdistributed_amd._C`.
                      If the extension is missing on a GPU machine we raise
                      instead of silently falling back: a silent eager
                      fallback would fake GPU coverage.
"""

from .native import load_native, native_available, require_native  # noqa: F401
from .functional import (  # noqa: F401
    logit_cross_entropy,
    fused_add_relu,
    batch_norm_act,
    max_pool2d,
    MaxPool2d,
    global_avg_pool,
    GlobalAvgPool,
)
from .fused_optim import FusedSGDMomentum, FusedAdam  # noqa: F401
from .conv import fda_conv2d, FdaConv2d  # noqa: F401
