"""In-tree build of the arkflow_amd._native gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands inside arkflow_amd/ so it travels with the repo snapshot
to GPU boxes (no JIT cache dependence).
"""
import os

from setuptools import Extension, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "arkflow_amd", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in ("bindings.cpp", "filter.hip", "hash_agg.hip", "hash_join.hip",
              "gemm_bf16.hip", "gemm_8phase.hip", "rowops.hip", "attention.hip", "proto_decode.hip", "radix_sort.hip", "json_decode.hip",
              "stepfused.hip")
]

setup(
    name="arkflow_amd_native",
    ext_modules=[
        CUDAExtension(
            name="arkflow_amd._native",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        ),
        # CPU-only native WAL frame codec (plain CPython extension, no torch)
        Extension(
            name="arkflow_amd._wal_native",
            sources=[os.path.join(CSRC, "wal_codec.cpp")],
            extra_compile_args=["-O3", "-std=c++17"],
        ),
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
