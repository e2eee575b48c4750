"""In-tree build of the hefl._C HIP extension for MI355X (gfx950).

Build: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(hipcc cross-compiles without a GPU; the .so lands at hefl/_C*.so and
travels with the repo snapshot to the GPU box).
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="hefl",
    version="0.1.0",
    packages=find_packages(include=["hefl", "hefl.*"]),
    ext_modules=[
        CUDAExtension(
            name="hefl._C",
            sources=[
                "hefl/csrc/bindings.cpp",
                "hefl/csrc/ntt.hip",
                "hefl/csrc/cnn.hip",
                "hefl/csrc/fft.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
