"""Build the gfx950 HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""
import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

setup(
    name="multihop_offload_amd",
    version="0.1.0",
    packages=["multihop_offload_amd"],
    ext_modules=[
        CUDAExtension(
            name="multihop_offload_amd._hip_ops",
            sources=[
                "multihop_offload_amd/ops/hip/bindings.cpp",
                "multihop_offload_amd/ops/hip/fw.hip",
                "multihop_offload_amd/ops/hip/episode.hip",
                "multihop_offload_amd/ops/hip/queueing.hip",
                "multihop_offload_amd/ops/hip/chebconv.hip",
                "multihop_offload_amd/ops/hip/optimizer.hip",
                "multihop_offload_amd/ops/hip/chebconv_large.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
