#!/usr/bin/env python3
"""Build the in-tree CDNA4 HIP extension: harmony_amd/ops/_hip_ops.

  python setup_ops.py build_ext --inplace

Cross-compiles for gfx950 (MI355X) — no GPU needed to build. The .so lands
inside harmony_amd/ops/ and travels with the repo snapshot to GPU boxes.
"""

import os
import sys
from pathlib import Path

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", "8")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).resolve().parent
SRC = sorted(str(p) for p in (ROOT / "harmony_amd/ops/csrc").glob("*.hip")) + \
      sorted(str(p) for p in (ROOT / "harmony_amd/ops/csrc").glob("*.cpp"))

if __name__ == "__main__":
    if len(sys.argv) == 1:
        sys.argv += ["build_ext", "--inplace"]
    setup(
        name="harmony_amd_hip_ops",
        ext_modules=[
            CUDAExtension(
                name="harmony_amd.ops._hip_ops",
                sources=SRC,
                extra_compile_args={
                    "cxx": ["-O3"],
                    "nvcc": ["-O3", "--offload-arch=gfx950"],
                },
            )
        ],
        cmdclass={"build_ext": BuildExtension},
    )
