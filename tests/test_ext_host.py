"""CPU-side checks of the in-tree HIP extension's host API.

The .so cross-compiles for gfx950 and imports without a GPU; these run in the
CPU gate and catch a stale/broken in-tree build before any GPU tier does.
"""
import pytest


def _ext():
    from perceiver_amd.ops import hip

    try:
        return hip.ext()
    except ImportError:
        pytest.skip("in-tree extension not built")


def test_extension_loads_and_exposes_api():
    ext = _ext()
    for name in ["flash_fwd", "flash_bwd", "flash_supported", "ln_fwd", "ln_bwd",
                 "adamw_step", "rotary_apply", "dropout_add_fwd", "dropout_add_bwd",
                 "colsum_bf16", "gemm_bt", "gemm_bt_gelu", "gemm_bt_applicable",
                 "transpose_bf16", "gelu_bias_fwd", "gelu_bias_bwd"]:
        assert hasattr(ext, name), name


def test_gemm_bt_applicability_host_rules():
    ext = _ext()
    # flagship projection shapes (fwd orientation)
    assert ext.gemm_bt_applicable(16384, 1280, 1280)
    assert ext.gemm_bt_applicable(16384, 1792, 1280)
    assert ext.gemm_bt_applicable(65536, 256, 768)
    # the BN=160 exact-fill family
    assert ext.gemm_bt_applicable(16384, 1280, 2816)
    assert ext.gemm_bt_applicable(512, 160, 192)
    # rejections: M tile, N tile, K tile, ring depth
    assert not ext.gemm_bt_applicable(100, 128, 192)
    assert not ext.gemm_bt_applicable(256, 100, 192)
    assert not ext.gemm_bt_applicable(256, 128, 100)
    assert not ext.gemm_bt_applicable(256, 128, 128)


def test_flash_supported_host_gate():
    ext = _ext()
    assert ext.flash_supported(32, 160, 0)
    assert ext.flash_supported(64, 64, 0)
