"""Loader and Python bindings for the in-tree HIP/CDNA4 extension (_perceiver_hip.so).

The extension is built ahead of time for gfx950 by ``python -m perceiver_amd.ops.build``
(or ``__graft_entry__.build()``) and committed in-tree so the gpurun snapshot carries it.

Policy: on a GPU box the HIP path must actually run — if a CUDA/ROCm device is present
and the extension cannot be loaded, importing this module raises unless
``PERCEIVER_AMD_ALLOW_EAGER=1`` is set (CI / CPU containers are unaffected).
"""
from __future__ import annotations

import os
from typing import Optional

import torch

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        import importlib.util
        import pathlib

        here = pathlib.Path(__file__).parent
        # canonical artifact first (what ops.build writes); fall back to any
        # interpreter-tagged name a setup.py build may have produced
        so = here / "_perceiver_hip.so"
        if not so.exists():
            so = next(iter(sorted(here.glob("_perceiver_hip*.so"))), None)
        if so is None:
            raise ImportError("no _perceiver_hip*.so found (run python -m perceiver_amd.ops.build)")
        spec = importlib.util.spec_from_file_location("_perceiver_hip", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _ext = mod
    except Exception as e:  # noqa: BLE001
        _ext = None
        if torch.cuda.is_available() and os.environ.get("PERCEIVER_AMD_ALLOW_EAGER") != "1":
            raise RuntimeError(
                f"GPU present but the perceiver_amd HIP extension failed to load ({e}). "
                "Build it with `python -m perceiver_amd.ops.build`, or set "
                "PERCEIVER_AMD_ALLOW_EAGER=1 to explicitly allow the eager fallback."
            ) from e
    return _ext


def is_available() -> bool:
    return torch.cuda.is_available() and _load() is not None


def can_use_flash(q, k, v, dropout_p: float = 0.0, training: bool = False) -> bool:
    """Shape/dtype gate for the fused flash kernel."""
    if not is_available():
        return False
    ext = _load()
    if ext is None:
        return False
    d_qk, d_v = q.shape[-1], v.shape[-1]
    # bf16 only (the kernels are bf16-I/O, fp32-accumulate); fp32 inputs are allowed
    # under autocast where casting to bf16 is the contract anyway
    bf16_ok = q.dtype == torch.bfloat16 or (
        q.dtype == torch.float32 and torch.is_autocast_enabled()
    )
    if not bf16_ok:
        return False
    ok = bool(ext.flash_supported(d_qk, d_v, int(training and dropout_p > 0)))
    if not ok:
        _warn_fallback_once(d_qk, d_v)
    return ok


_warned_shapes: set = set()


def _warn_fallback_once(d_qk: int, d_v: int) -> None:
    """One warning per head-dim combination when a CUDA-device attention call
    silently drops to the eager composition — so a new config never loses the
    fused kernels without a trace in the logs."""
    key = (d_qk, d_v)
    if key not in _warned_shapes:
        _warned_shapes.add(key)
        import warnings
        warnings.warn(
            f"flash kernel shape gate: head dims qk={d_qk}, v={d_v} unsupported; "
            "attention falls back to the eager composition for this shape",
            stacklevel=3,
        )


def flash_attention(q, k, v, pad_mask=None, causal: bool = False,
                    dropout_p: float = 0.0, training: bool = False):
    from perceiver_amd.ops.flash import FlashAttention

    return FlashAttention.apply(q, k, v, pad_mask, causal, dropout_p, training)


def ext():
    """The raw extension module (raises if unavailable)."""
    m = _load()
    if m is None:
        raise RuntimeError("perceiver_amd HIP extension not available")
    return m
