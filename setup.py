"""Build the in-tree HIP extension for gfx950 (MI355X).

    python setup.py build_ext --inplace

Produces brainiak_amd/ops/_hip_ops*.so next to its Python wrapper so the
artifact travels with the repo snapshot to GPU boxes (no JIT cache).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_DIR = os.path.join("brainiak_amd", "ops", "hip")

ext = CUDAExtension(
    name="brainiak_amd.ops._hip_ops",
    sources=[
        os.path.join(HIP_DIR, "ext.cpp"),
        os.path.join(HIP_DIR, "fcma_kernels.hip"),
        os.path.join(HIP_DIR, "svm_kernels.hip"),
        os.path.join(HIP_DIR, "isfc_kernels.hip"),
        os.path.join(HIP_DIR, "stencil_kernels.hip"),
        os.path.join(HIP_DIR, "procrustes.hip"),
        os.path.join(HIP_DIR, "tfa_kernels.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3"],
        "nvcc": ["-O3", "--offload-arch=gfx950"],
    },
)

setup(
    name="brainiak_amd",
    version="0.1.0",
    packages=["brainiak_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
