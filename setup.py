"""In-tree build of the atomo_amd HIP extension for MI355X (gfx950).

    python setup.py build_ext --inplace

The .so lands at atomo_amd/ops/_atomo_hip*.so so it travels with the repo
snapshot onto GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="atomo_amd.ops._atomo_hip",
    sources=[
        "atomo_amd/ops/csrc/bindings.cpp",
        "atomo_amd/ops/csrc/atomo_kernels.hip",
        "atomo_amd/ops/csrc/svd_batched.hip",
        "atomo_amd/ops/csrc/jacobi_eigh.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
    libraries=["rocsolver", "rocblas"],
)

setup(
    name="atomo_amd",
    version="0.1.0",
    packages=["atomo_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
