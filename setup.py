"""In-tree build of the nerrf-amd CDNA4 HIP extension.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(driven by __graft_entry__.build()).  gfx950-only by design.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "nerrf_amd", "ops", "hip")

sources = [
    os.path.join(HIP_DIR, "bindings.cpp"),
    os.path.join(HIP_DIR, "gather_sage.hip"),
    os.path.join(HIP_DIR, "lstm_cell.hip"),
    os.path.join(HIP_DIR, "lstm_step_fused.hip"),
    os.path.join(HIP_DIR, "lstm_rec_fused.hip"),
    os.path.join(HIP_DIR, "stream_gemm.hip"),
    os.path.join(HIP_DIR, "rec_gemm.hip"),
    os.path.join(HIP_DIR, "mcts.hip"),
    os.path.join(HIP_DIR, "event_scatter.hip"),
    os.path.join(HIP_DIR, "sage_fused.hip"),
    os.path.join(HIP_DIR, "sage_ln_act.hip"),
]
sources = [s for s in sources if os.path.exists(s)]

import pybind11  # noqa: E402
from setuptools import Extension  # noqa: E402

setup(
    name="nerrf_amd_kernels",
    ext_modules=[
        CUDAExtension(
            name="nerrf_amd._kernels",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": [
                    "-O3",
                    "-std=c++17",
                    "--offload-arch=gfx950",
                ],
            },
        ),
        # plain pybind11 (no torch dependency): trace ingest must load in
        # collector-side processes that never import torch
        Extension(
            name="nerrf_amd._ingest",
            sources=[os.path.join(ROOT, "tracker", "daemon", "ingest_ext.cpp")],
            include_dirs=[pybind11.get_include()],
            extra_compile_args=["-O3", "-std=c++17"],
            language="c++",
        ),
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
