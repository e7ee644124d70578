"""Builds the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands in cyclegan_amd/ops/_hip/ (in-tree, travels with the
repo snapshot to GPU boxes)."""

import glob
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(HERE, "cyclegan_amd", "ops", "_hip")
# Exclude torch-hipify build artifacts (*_hip.hip) — they are generated
# copies of the hand-written sources and must not be compiled or tracked.
sources = sorted(s for s in (glob.glob(os.path.join(HIP_DIR, "*.hip")) +
                             glob.glob(os.path.join(HIP_DIR, "*.cpp")))
                 if not s.endswith("_hip.hip"))

ext_modules = []
if sources:
    ext_modules.append(cpp_extension.CUDAExtension(
        name="cyclegan_amd.ops._hip._cyclegan_hip",
        sources=sources,
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": (["-O3", "-std=c++17", "--offload-arch=gfx950"] +
                     os.environ.get("CYG_EXTRA_FLAGS", "").split()),
        },
    ))

setup(
    name="cyclegan_amd",
    version="0.1.0",
    packages=["cyclegan_amd", "cyclegan_amd.models", "cyclegan_amd.ops",
              "cyclegan_amd.parallel", "cyclegan_amd.data",
              "cyclegan_amd.utils"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
