"""In-tree build of the HIP/CDNA4 extension (gfx950).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands at distributed_sac_amd/ops/_hip_ops*.so and travels
with the repo snapshot to GPU boxes (it is git-ignored but not
gpurun-ignored).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="distributed_sac_amd",
    version="0.1.0",
    packages=["distributed_sac_amd"],
    ext_modules=[
        CUDAExtension(
            name="distributed_sac_amd.ops._hip_ops",
            sources=["distributed_sac_amd/ops/_hip/dsac_kernels.hip",
                     "distributed_sac_amd/ops/_hip/shm_ring.cpp",
                     "distributed_sac_amd/ops/_hip/bf16_gemm.hip",
                     "distributed_sac_amd/ops/_hip/chain_gemm.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
