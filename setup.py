"""Build the gfx950 HIP kernel extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built _yams_hip*.so lands next to ops/ (inside the package) so it ships
to GPU boxes with the source tree (it is .gitignore'd, not gpurun-ignored).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "yet_another_mobilenet_series_amd", "ops", "csrc")

sources = [os.path.join(CSRC, f) for f in
           ("bindings.cpp", "gemm.hip", "dwconv.hip", "bn.hip", "misc.hip",
            "se.hip", "update.hip")]

setup(
    name="yams_hip",
    ext_modules=[
        CUDAExtension(
            name="yet_another_mobilenet_series_amd.ops._yams_hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
