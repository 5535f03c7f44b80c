"""In-tree build of the mi355x HIP extension (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles without a GPU; the built mi355x/_C*.so travels to the
GPU box with the repo snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

SRC = [
    "mi355x/csrc/bindings.cpp",
    "mi355x/csrc/elementwise.hip",
    "mi355x/csrc/conv.hip",
    "mi355x/csrc/conv_mfma.hip",
    "mi355x/csrc/mfma_probe.hip",
    "mi355x/csrc/wcast.hip",
    "mi355x/csrc/bn.hip",
    "mi355x/csrc/pool.hip",
    "mi355x/csrc/gemm.hip",
    "mi355x/csrc/gemm_mfma.hip",
    "mi355x/csrc/stem_mfma.hip",
    "mi355x/csrc/loss.hip",
    "mi355x/csrc/rccl_comm.cpp",
]

setup(
    name="mi355x",
    version="0.1.0",
    packages=["mi355x"],
    ext_modules=[
        CUDAExtension(
            name="mi355x._C",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
            libraries=["rccl"],
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
