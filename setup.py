"""In-tree build of the gfx950 HIP extension.

Build with:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The resulting _ps_hip.so lands next to pytorch_ps_mpi_amd/ops/ and travels
with the repo snapshot (it is git-ignored but NOT gpurun-ignored).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "pytorch_ps_mpi_amd", "ops", "csrc")

setup(
    name="pytorch_ps_mpi_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="pytorch_ps_mpi_amd.ops._ps_hip",
            sources=[
                os.path.join(CSRC, "bindings.cpp"),
                os.path.join(CSRC, "ps_kernels.hip"),
                os.path.join(CSRC, "bn_kernels.hip"),
                os.path.join(CSRC, "ln_kernels.hip"),
                os.path.join(CSRC, "ce_kernels.hip"),
                os.path.join(CSRC, "attn_kernels.hip"),
                os.path.join(CSRC, "mfma_attn_kernels.hip"),
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
