"""In-tree build of the CDNA4 (gfx950) HIP extension.

``python setup.py build_ext --inplace`` produces
``vantage6_amd/ops/_hip_ops*.so`` — built in-tree so the .so travels with
the repo snapshot to GPU boxes (JIT caches under ~/.cache do not).
"""
import os
from pathlib import Path

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "vantage6_amd" / "ops" / "csrc"

sources = [str(CSRC / f) for f in
           ("bindings.hip", "elementwise.hip", "norms.hip", "gemm.hip", "bn.hip",
            "conv.hip", "attention.hip")]

setup(
    name="vantage6_amd",
    version="0.1.0",
    description="MI355X-native privacy-preserving federated learning "
                "framework (vantage6-capability surface)",
    packages=find_packages(include=["vantage6_amd", "vantage6_amd.*",
                                    "algorithms"]),
    entry_points={"console_scripts": ["v6amd = vantage6_amd.cli.main:cli"]},
    ext_modules=[
        CUDAExtension(
            name="vantage6_amd.ops._hip_ops",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
