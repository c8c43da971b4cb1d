"""Build the in-tree native extensions.

  python setup.py build_ext --inplace

Extensions:
  greptimedb_amd._native   — host ingest path (line parser, WAL). pybind11-only,
                             compiles in seconds.
  greptimedb_amd._hip_ops  — CDNA4 HIP kernels for gfx950 (MI355X), driven by
                             hipcc through torch.utils.cpp_extension.

Both .so files land inside greptimedb_amd/ so they travel with the repo
snapshot to GPU boxes (they are git-ignored but NOT gpurun-ignored).
"""

import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", "8")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

ext_modules = [
    CppExtension(
        "greptimedb_amd._native",
        sources=["csrc/native.cpp", "csrc/pagedec.cpp"],
        # zstd header/lib: conda toolchain ships them; the runtime links
        # against the system libzstd.so.1 compatible ABI
        include_dirs=["/opt/conda/include"],
        extra_compile_args=["-O3", "-std=c++17"],
        libraries=["zstd"],
        library_dirs=["/opt/conda/lib"],
        extra_link_args=["-Wl,-rpath,/opt/conda/lib"],
    ),
    CUDAExtension(
        "greptimedb_amd._hip_ops",
        sources=["csrc/hip_ops.cpp", "csrc/kernels.hip"],
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
        },
    ),
]

setup(
    name="greptimedb_amd",
    version="0.1.0",
    packages=[
        "greptimedb_amd",
        "greptimedb_amd.utils",
        "greptimedb_amd.models",
        "greptimedb_amd.ops",
        "greptimedb_amd.engine",
        "greptimedb_amd.query",
        "greptimedb_amd.parallel",
        "greptimedb_amd.servers",
        "greptimedb_amd.meta",
    ],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
