"""Build the creditcore HIP extension in-tree for gfx950 (MI355X).

    python setup.py build_ext --inplace

The .so lands at creditcore/_ccore*.so so it travels with the repo snapshot
(it is git-ignored; sources in csrc/ are the history). No CUDA path exists:
this builds only with ROCm (hipcc) for gfx950.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="creditcore-kernels",
    ext_modules=[
        CUDAExtension(
            name="creditcore._ccore",
            sources=["csrc/creditcore_kernels.hip"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                # CREDITCORE_ASAN=1: host-side AddressSanitizer debug build
                # (SURVEY.md §5.2); run with LD_PRELOAD of libasan.
                "nvcc": ["-O3", "-std=c++17"]
                + (["-fsanitize=address", "-shared-libasan", "-g"]
                   if os.environ.get("CREDITCORE_ASAN") == "1" else []),
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
