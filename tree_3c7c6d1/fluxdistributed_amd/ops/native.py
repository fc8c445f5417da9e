"""Loader for the in-tree HIP extension (fluxdistributed_amd/_C*.so).

The extension is compiled for gfx950 only (see csrc/ and setup.py); it is
built in-tree so the .so travels with the repo snapshot to GPU machines.
"""

import os



_NATIVE = None
_TRIED = False


def load_native():
    """Import the compiled extension, caching the result. Returns module or None."""
    global _NATIVE, _TRIED
    if _TRIED:
        return _NATIVE
    _TRIED = True
    try:
        from fluxdistributed_amd import _C  # noqa: F401  (in-tree .so)

        _NATIVE = _C
    except ImportError:
        _NATIVE = None
    return _NATIVE


def native_available() -> bool:
    return load_native() is not None


def require_native(opname: str):
    """Return the native module; raise loudly if we are on a GPU without it.

    On a GPU box the HIP path must be the one that runs — a silent eager
    fallback would let GPU tests pass without exercising native code.
    """
    mod = load_native()
    if mod is None:
        raise RuntimeError(
            f"fluxdistributed_amd native extension is required for op '{opname}' "
            "on GPU tensors but fluxdistributed_amd._C is not importable. "
            "Build it with: python setup.py build_ext --inplace "
            "(PYTORCH_ROCM_ARCH=gfx950)."
        )
    return mod


def allow_eager_gpu_fallback() -> bool:
    """Escape hatch for debugging only (FLUXDIST_ALLOW_EAGER=1)."""
    return os.environ.get("FLUXDIST_ALLOW_EAGER", "0") == "1"
