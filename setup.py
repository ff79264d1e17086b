"""
Build the dragnet_amd gfx950 HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces dragnet_amd/ops/_dragnet_hip.*.so (git-ignored; travels with
the working tree to GPU machines).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

# torch's hipify step copies ext.hip -> ext_hip.hip at first build and
# ninja's depfile then tracks the COPY, so edits to the included kernel
# sources do not trigger recompilation.  Force a clean kernel build.
import glob
import shutil

for stale in glob.glob("dragnet_amd/ops/hip/*_hip.hip"):
    os.unlink(stale)
shutil.rmtree("build/temp.linux-x86_64-3.10", ignore_errors=True)

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="dragnet_amd_ops",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="dragnet_amd.ops._dragnet_hip",
            sources=["dragnet_amd/ops/hip/ext.hip"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": [os.environ.get("DN_OPT", "-O3"), "-std=c++17"]
                        + (["-DDN_DIRECT_BYTES"]
                           if os.environ.get("DN_DIRECT_BYTES") else []),
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
