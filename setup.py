"""In-tree build of the CDNA4 HIP extension (_tfmx_C).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Cross-compiles on a GPU-less box (hipcc needs no device).  The .so lands in
transformer_amd/ops/ so it travels with the repo snapshot to GPU boxes.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import CUDAExtension, BuildExtension  # noqa: E402

CSRC = os.path.join("transformer_amd", "ops", "csrc")

sources = [os.path.join(CSRC, f) for f in sorted(os.listdir(CSRC))
           if f.endswith((".hip", ".cpp"))]

setup(
    name="transformer_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="transformer_amd.ops._tfmx_C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
