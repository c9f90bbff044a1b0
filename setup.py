"""In-tree build of the fedkit CDNA4 kernel extension (fedkit/_C.so).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles gfx950 without a GPU; the built .so lives in-tree so a
repo snapshot carries it to the GPU box (no JIT cache involved).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

SOURCES = [
    "csrc/module.cpp",
    "csrc/elementwise.hip",
    "csrc/flat_ops.hip",
    "csrc/loss.hip",
    "csrc/losses_extra.hip",
    "csrc/linear.hip",
    "csrc/adam.hip",
    "csrc/batchnorm.hip",
    "csrc/conv2d_mfma.hip",
    "csrc/conv_small.hip",
    "csrc/pool.hip",
]

setup(
    name="fedkit-kernels",
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="fedkit._C",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
