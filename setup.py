"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built _C*.so lands next to neuronx_distributed_training_amd/ops/ and
travels to the GPU box with the repo snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "neuronx_distributed_training_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in (
        "bindings.cpp",
        "rmsnorm.hip",
        "swiglu.hip",
        "rope.hip",
        "adamw.hip",
        "flash_attn_fwd.hip",
        "flash_attn_fwd_dbuf.hip",
        "flash_attn_fwd_v3.hip",
        "flash_attn_bwd_v3.hip",
        "flash_attn_bwd.hip",
        "cross_entropy.hip",
        "moe_gemm.hip",
        "probe.hip",
    )
]

import pybind11
from setuptools import Extension

setup(
    name="neuronx_distributed_training_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="neuronx_distributed_training_amd.ops._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        ),
        Extension(
            name="neuronx_distributed_training_amd.data._helpers_cpp",
            sources=[
                os.path.join(
                    HERE, "neuronx_distributed_training_amd", "data",
                    "csrc_cpu", "helpers.cpp",
                )
            ],
            include_dirs=[pybind11.get_include()],
            extra_compile_args=["-O3", "-std=c++17"],
            language="c++",
        ),
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
