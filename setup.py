"""In-tree build of the gfx950 kernel extension amgx_amd._core.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands inside amgx_amd/ so it travels with the repo snapshot to
GPU machines (no JIT cache dependence).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

here = os.path.dirname(os.path.abspath(__file__))
csrc = os.path.join(here, "amgx_amd", "csrc")

ext = CUDAExtension(
    name="amgx_amd._core",
    sources=[
        os.path.join(csrc, "bindings.cpp"),
        os.path.join(csrc, "kernels_solve.hip"),
        os.path.join(csrc, "kernels_setup.hip"),
        os.path.join(csrc, "kernels_classical.hip"),
        os.path.join(csrc, "kernels_mfma.hip"),
        os.path.join(csrc, "kernels_spgemm.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="amgx_amd",
    version="0.1.0",
    packages=["amgx_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
