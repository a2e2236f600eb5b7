"""Build the CDNA4 HIP extension in-tree (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

produces mapreduce_amd/_hip_ops*.so, which travels with the repo snapshot
to GPU boxes (no JIT cache involved).
"""

import os
import shutil

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

# The ninja rule for hipcc emits no depfile, so edits to #include'd .hip
# kernels do NOT trigger a recompile (a stale .so once shipped an old
# kernel).  The extension is one TU (~90 s): always build fresh.
shutil.rmtree(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                           "build", "temp.linux-x86_64-3.10"),
              ignore_errors=True)

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))

setup(
    name="mapreduce_amd",
    version="0.1.0",
    packages=["mapreduce_amd"],
    ext_modules=[
        CUDAExtension(
            name="mapreduce_amd._hip_ops",
            sources=["mapreduce_amd/ops/hip/ops_ext.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
