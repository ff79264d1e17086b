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

import pybind11  # noqa: E402
from setuptools import Extension  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

# The image ships libsqlite3.so.0 without a dev symlink; link the
# versioned .so by absolute path (same image on the GPU boxes).
_SQLITE_SO = "/usr/lib/x86_64-linux-gnu/libsqlite3.so.0"

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
        ),
        Extension(
            name="dragnet_amd.index._points",
            sources=["dragnet_amd/index/points_fast.cpp"],
            include_dirs=[pybind11.get_include()],
            extra_compile_args=["-O3", "-std=c++17",
                                "-fvisibility=hidden"],
            language="c++",
        ),
        Extension(
            name="dragnet_amd.index._csink",
            sources=["dragnet_amd/index/csink.cpp"],
            include_dirs=[pybind11.get_include()],
            extra_compile_args=["-O3", "-std=c++17",
                                "-fvisibility=hidden"],
            extra_link_args=[_SQLITE_SO],
            language="c++",
        ),
    ],
    cmdclass={"build_ext": BuildExtension},
)
