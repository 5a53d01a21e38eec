"""Build the in-tree gfx950 HIP extension.

Usage (in-tree so the .so ships to GPU boxes with the source snapshot):
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

sources = [
    "csrc/bind.cpp",
    "csrc/ops/layernorm.hip",
    "csrc/ops/bias_act.hip",
    "csrc/ops/fused_residual.hip",
    "csrc/ops/embedding.hip",
    "csrc/ops/cross_entropy.hip",
    "csrc/ops/attention.hip",
    "csrc/ops/wgrad.hip",
    "csrc/ops/mlm_head.hip",
    "csrc/ops/gemm_epilogue.cpp",
    "csrc/optim/multi_tensor.hip",
    "csrc/tok/tokenizer.cpp",
]

setup(
    name="bert_pytorch_amd_ext",
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="bert_pytorch_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
            libraries=["hipblaslt"],
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
