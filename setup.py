"""Build the gfx950 HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

(equivalently `python -m perceiver_amd.ops.build`). The .so is placed at
perceiver_amd/ops/_perceiver_hip.so and loaded directly — no JIT cache involved.
"""
import os
import pathlib

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

try:
    from torch.utils.cpp_extension import BuildExtension, CUDAExtension

    csrc = pathlib.Path(__file__).parent / "perceiver_amd" / "ops" / "csrc"
    sources = sorted(str(p) for p in csrc.glob("*.hip") if not p.name.endswith("_hip.hip"))
    sources += sorted(str(p) for p in csrc.glob("*.cpp"))
    ext_modules = [
        CUDAExtension(
            name="perceiver_amd.ops._perceiver_hip",
            sources=sources,
            extra_compile_args={"cxx": ["-O3", "-std=c++17"], "nvcc": ["-O3", "-std=c++17"]},
        )
    ]
    cmdclass = {"build_ext": BuildExtension}
except ImportError:  # metadata-only build
    ext_modules, cmdclass = [], {}

setup(name="perceiver-mi355x", ext_modules=ext_modules, cmdclass=cmdclass)
