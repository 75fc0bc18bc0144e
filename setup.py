"""Build the unicore_amd framework + its gfx950 HIP kernel extension.

Single-target build: MI355X (gfx950, CDNA4) via hipcc.  The extension is
built IN-TREE (``python setup.py build_ext --inplace``) so the resulting
``unicore_amd/_kernels*.so`` travels with the source checkout.

Unlike the reference (which ships 8 separate CUDA extensions behind an
``--enable-cuda-ext`` flag and an sm_70/80/90 gencode matrix, reference
setup.py:17-24,141-387), there is exactly one extension and one
architecture; a CPU-only install simply skips ``build_ext`` — every op has
an eager fallback on CPU.
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_SOURCES = [
    "csrc/bindings.cpp",
    "csrc/softmax_dropout.hip",
    "csrc/norms.hip",
    "csrc/adam.hip",
    "csrc/multi_tensor.hip",
    "csrc/rounding.hip",
    "csrc/qkv.hip",
    "csrc/gelu_dropout.hip",
    "csrc/mfma_probe.hip",
    "csrc/flash_attn.hip",
    "csrc/dropout_add.hip",
    "csrc/dropout_add_ln.hip",
    "csrc/embedding.hip",
    "csrc/cross_entropy.hip",
    "csrc/gaussian.hip",
    "csrc/gated.hip",
]

setup(
    name="unicore_amd",
    version="0.1.0",
    description="MI355X-native training framework (Uni-Core capabilities)",
    packages=find_packages(include=["unicore_amd", "unicore_amd.*", "unicore_cli"]),
    ext_modules=[
        CUDAExtension(
            name="unicore_amd._kernels",
            sources=HIP_SOURCES,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
    entry_points={
        "console_scripts": [
            "unicore-train = unicore_cli.train:cli_main",
        ],
    },
    python_requires=">=3.8",
)
