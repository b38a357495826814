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

# Debug builds:
#   YAMS_ASAN=1   host-side AddressSanitizer on the binding/launcher code
#   YAMS_DEBUG=1  -g -O1 device code for AMD_SERIALIZE_KERNEL bisection
cxx_flags = ["-O3", "-std=c++17"]
dev_flags = ["-O3", "-std=c++17", "--offload-arch=gfx950"]
if os.environ.get("YAMS_ASAN") == "1":
    cxx_flags += ["-fsanitize=address", "-fno-omit-frame-pointer", "-g"]
if os.environ.get("YAMS_DEBUG") == "1":
    dev_flags = ["-g", "-O1", "-std=c++17", "--offload-arch=gfx950"]

setup(
    name="yams_hip",
    ext_modules=[
        CUDAExtension(
            name="yet_another_mobilenet_series_amd.ops._yams_hip",
            sources=sources,
            extra_compile_args={
                "cxx": cxx_flags,
                "nvcc": dev_flags,
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
