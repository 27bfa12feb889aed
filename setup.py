"""In-tree build of the gfx950 HIP extension.

  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built _dtx_hip*.so lands in datatunerx_amd/ops/ so it travels with
the repo snapshot to GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import (BuildExtension, CUDAExtension,  # noqa: E402
                                       CppExtension)

ROOT = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(ROOT, "datatunerx_amd", "ops", "hip")

sources = [os.path.join(SRC, f) for f in sorted(os.listdir(SRC))
           if f.endswith((".cpp", ".hip")) and not f.endswith(("_hip.hip",
                                                               "_hip.cpp"))]

setup(
    name="datatunerx_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="datatunerx_amd.ops._dtx_hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        ),
        CppExtension(
            name="datatunerx_amd.native._dtx_native",
            sources=[os.path.join(ROOT, "datatunerx_amd", "native",
                                  "supervisor.cpp")],
            extra_compile_args={"cxx": ["-O3", "-std=c++17"]},
        ),
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
