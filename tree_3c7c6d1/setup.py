"""Build the in-tree gfx950 HIP extension: python setup.py build_ext --inplace

The .so lands at fluxdistributed_amd/_C*.so (in-tree so it travels with the
repo snapshot to GPU machines; git-ignored so history stays source-only).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "fluxdistributed_amd", "csrc")

ext = CUDAExtension(
    name="fluxdistributed_amd._C",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "elementwise.hip"),
        os.path.join(CSRC, "bn_act.hip"),
        os.path.join(CSRC, "pool.hip"),
        os.path.join(CSRC, "conv_igemm.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="fluxdistributed_amd",
    version="0.1.0",
    packages=["fluxdistributed_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
