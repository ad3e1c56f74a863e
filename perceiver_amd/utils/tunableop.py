"""Arm PyTorch TunableOp with the pre-tuned GEMM-algorithm table for gfx950.

The table (repo-root ``tuned/tunableop_gfx950.csv``) comes from a one-off
``PYTORCH_TUNABLEOP_TUNING=1`` pass over every flagship bench shape on MI355X
and is validator-locked to this image's torch/hipBLASLt/rocBLAS versions.
Loading it (tuning OFF) replaces hipBLASLt's heuristic algorithm picks with the
measured-best ones: +6% on the 201M MLM flagship step, +10% on the WikiText
Perceiver-AR step. Uncovered shapes silently use the default heuristics.
"""
from __future__ import annotations

import os
import shutil


def arm_tunableop(rank: int = 0) -> bool:
    """Idempotent; returns True if the table was armed. Must run before the
    first GEMM. TunableOp appends the rank before the file extension, so every
    rank gets a private copy of the canonical table."""
    canonical = os.path.join(os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__)))), "tuned", "tunableop_gfx950.csv")
    if "PYTORCH_TUNABLEOP_ENABLED" in os.environ or not os.path.exists(canonical):
        return False
    base = f"/tmp/tunableop_{os.getpid()}.csv"
    # TunableOp inserts an ordinal (device index) before the extension when it
    # reads/writes; cover every possible ordinal on an 8-GPU node
    for r in range(8):
        shutil.copy(canonical, f"/tmp/tunableop_{os.getpid()}{r}.csv")
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = base
    return True
