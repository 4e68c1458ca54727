"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands next to the package (aurora_amd/_hip_ops*.so) so the gpurun
snapshot carries it to the GPU box; no JIT cache involved.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

SRC = [
    "aurora_amd/ops/csrc/bindings.cpp",
    "aurora_amd/ops/csrc/elementwise.hip",
    "aurora_amd/ops/csrc/attention_decode.hip",
    "aurora_amd/ops/csrc/attention_prefill.hip",
    "aurora_amd/ops/csrc/encoder.hip",
    "aurora_amd/ops/csrc/skinny_gemm.hip",
    "aurora_amd/ops/csrc/allreduce.hip",
]

setup(
    name="aurora_amd_hip_ops",
    ext_modules=[
        CUDAExtension(
            name="aurora_amd._hip_ops",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
