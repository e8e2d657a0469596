"""Build the trlx_amd HIP extension in-tree for gfx950 (MI355X).

Usage: python setup.py build_ext --inplace
"""

import os
from pathlib import Path

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = Path(__file__).parent / "trlx_amd" / "csrc"

# *_hip.hip files are hipify byproducts regenerated at build time — never sources.
sources = [str(CSRC / "bindings.cpp")] + sorted(
    str(p) for p in CSRC.glob("*.hip") if not p.name.endswith("_hip.hip")
)

ext = CUDAExtension(
    name="trlx_amd._C",
    sources=sources,
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="trlx_amd",
    version="0.1.0",
    description="MI355X-native RLHF fine-tuning framework (trlX-capable API)",
    packages=find_packages(exclude=["tests"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
