"""Build the in-tree gfx950 HIP extension:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands at wap_amd/_wap_hip*.so (git-ignored, but it travels with the
repo snapshot to GPU boxes).  Source is hand-written HIP (wap_amd/csrc) —
there is no CUDA anywhere, so hipify is a no-op pass-through.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="wap_amd_ext",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="wap_amd._wap_hip",
            sources=["wap_amd/csrc/wap_ext.hip", "wap_amd/csrc/conv_mfma.hip",
                     "wap_amd/csrc/fattn.hip", "wap_amd/csrc/convbwd.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
