"""In-tree build of the defer_amd gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

produces defer_amd/_hip_ops.*.so next to the package sources (the .so
travels with the repo snapshot to GPU boxes; no JIT cache involved).
"""

import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).parent
CSRC = ROOT / "defer_amd" / "csrc"

sources = [str(CSRC / f) for f in
           ["bindings.cpp", "conv.hip", "prebn.hip", "elementwise.hip", "pool.hip",
            "codec.hip", "lz4.hip"]
           if (CSRC / f).exists()]

setup(
    name="defer_amd",
    version="0.1.0",
    packages=["defer_amd"],
    ext_modules=[
        CUDAExtension(
            name="defer_amd._hip_ops",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
