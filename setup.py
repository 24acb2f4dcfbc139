"""Build the distllm-amd HIP extension in-tree for gfx950 (MI355X).

    python setup.py build_ext --inplace

The resulting distllm_amd/ops/_hip_ops*.so travels with the repo
snapshot to GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""

import os
from pathlib import Path

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

CSRC = Path(__file__).parent / "distllm_amd" / "ops" / "csrc"
# *_hip.hip files are torch-hipify build artifacts — exclude them.
SOURCES = [str(CSRC / "ext.cpp"), str(CSRC / "blt_gemm.cpp")] + sorted(
    str(p) for p in CSRC.glob("*.hip") if not p.name.endswith("_hip.hip")
)

setup(
    name="distllm_amd_ops",
    ext_modules=[
        CUDAExtension(
            name="distllm_amd.ops._hip_ops",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
            },
            libraries=["hipblaslt"],
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
