"""Build the ddlbench_amd native HIP extension in-tree for gfx950.

    python setup.py build_ext --inplace

The .so lands at ddlbench_amd/ops/_hip_ops*.so (git-ignored; it travels
with gpurun snapshots). MI355X (gfx950) only — no fat binaries.
"""

import os
import sys
from pathlib import Path

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
# Keep ninja parallelism sane in the build container.
os.environ.setdefault("MAX_JOBS", str(min(8, os.cpu_count() or 4)))

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "ddlbench_amd" / "ops" / "csrc"

# Hand-written kernel sources, pinned explicitly (torch's hipify step
# generates *_hip.hip siblings at build time; those are never tracked).
KERNEL_SOURCES = [
    "bindings.hip",
    "bn_act.hip",
    "conv_mfma.hip",
    "conv_mfma2.hip",
    "conv_wgrad.hip",
    "cross_entropy.hip",
    "depthwise_conv.hip",
    "fused_sgd.hip",
    "maxpool.hip",
    "seq_utils.hip",
]
sources = [str(CSRC / s) for s in KERNEL_SOURCES]

setup(
    name="ddlbench_amd_ops",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="ddlbench_amd.ops._hip_ops",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
