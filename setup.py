"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces vit_10b_fsdp_example_amd/_C*.so next to the package sources so
the built artifact travels with the repo snapshot to GPU boxes.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ext = CUDAExtension(
    name="vit_10b_fsdp_example_amd._C",
    sources=[
        "csrc/bindings.cpp",
        "csrc/ltgemm.cpp",
        "csrc/layernorm.hip",
        "csrc/adamw.hip",
        "csrc/cross_entropy.hip",
        "csrc/fmha.hip",
        "csrc/wgemm.hip",
        "csrc/fgemm.hip",
    ],
    libraries=["hipblaslt"],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": [
            "-O3",
            "-std=c++17",
            "--offload-arch=gfx950",
            "-Rpass-analysis=kernel-resource-usage",
        ],
    },
)

setup(
    name="vit_10b_fsdp_example_amd_ext",
    version="0.1.0",
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
