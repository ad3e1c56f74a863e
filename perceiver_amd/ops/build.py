"""Ahead-of-time build of the in-tree HIP extension for gfx950.

Usage: python -m perceiver_amd.ops.build
Builds perceiver_amd/ops/_perceiver_hip.so via torch.utils.cpp_extension (hipcc,
PYTORCH_ROCM_ARCH=gfx950). The .so is committed/gpurun-snapshotted in-tree so GPU
boxes load it directly without a JIT cache.
"""
from __future__ import annotations

import os
import pathlib
import shutil


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils import cpp_extension

    here = pathlib.Path(__file__).parent
    csrc = here / "csrc"
    build_dir = here / "_build"
    build_dir.mkdir(exist_ok=True)

    # exclude torch-hipify-generated *_hip.hip duplicates from previous builds
    sources = sorted(str(p) for p in csrc.glob("*.hip") if not p.name.endswith("_hip.hip"))
    sources += sorted(str(p) for p in csrc.glob("*.cpp"))
    mod_path = cpp_extension.load(
        name="_perceiver_hip",
        sources=sources,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        build_directory=str(build_dir),
        verbose=verbose,
        is_python_module=False,
        keep_intermediates=True,
    )
    # cpp_extension with is_python_module=False loads into the process; we want the
    # artifact: copy it next to the package so imports find it in-tree.
    so = build_dir / "_perceiver_hip.so"
    dst = here / "_perceiver_hip.so"
    shutil.copy2(so, dst)
    return str(dst)


if __name__ == "__main__":
    print(build())
