"""In-tree build of the turboprune_amd HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands inside turboprune_amd/ (imported as turboprune_amd._C) so
it travels to GPU boxes with the source tree.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "turboprune_amd", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "sgd_fused.hip"),
    os.path.join(CSRC, "kth_select.hip"),
    os.path.join(CSRC, "ce_loss.hip"),
    os.path.join(CSRC, "augment.hip"),
    os.path.join(CSRC, "gemm_masked.hip"),
    os.path.join(CSRC, "batchnorm.hip"),
    os.path.join(CSRC, "transpose.hip"),
    os.path.join(CSRC, "maxpool.hip"),
    os.path.join(CSRC, "cifar_aug.hip"),
    os.path.join(CSRC, "layernorm_gelu.hip"),
    os.path.join(CSRC, "rrc.hip"),
    os.path.join(CSRC, "conv_implicit.hip"),
    os.path.join(CSRC, "conv_wrw.hip"),
    os.path.join(CSRC, "gemm_256_8phase.hip"),
    os.path.join(CSRC, "sgd_multi.hip"),
    os.path.join(CSRC, "attention.hip"),
    os.path.join(CSRC, "schedulefree.hip"),
    os.path.join(CSRC, "conv_implicit_256.hip"),
]

setup(
    name="turboprune_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="turboprune_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
