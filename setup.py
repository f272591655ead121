"""Build the fast_autoaugment_amd HIP extension in-tree.

  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles for gfx950 without a GPU present; the resulting
fast_autoaugment_amd/ops/_C*.so travels with the source tree.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "fast_autoaugment_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "ext.cpp"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "loss.hip"),
    os.path.join(CSRC, "step.hip"),
    os.path.join(CSRC, "aug_kernels.hip"),
    os.path.join(CSRC, "bnrelu.hip"),
    os.path.join(CSRC, "conv_mfma.hip"),
    os.path.join(CSRC, "depthwise.hip"),
]

setup(
    name="fast_autoaugment_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="fast_autoaugment_amd.ops._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
