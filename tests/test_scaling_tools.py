"""Scaling-study tooling (reference examples/scaling/clm/{flops,laws}.py):
FLOPs estimator consistency and power-law fit recovery."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np

from examples.scaling.clm.flops import ComputeEstimator
from examples.scaling.clm.laws import fit_power_law


def _est(**kw):
    base = dict(vocab_size=262, max_seq_len=4096, max_latents=512,
                num_channels=512, num_layers=9, cross_attention_dropout=0.5)
    base.update(kw)
    return ComputeEstimator(**base)


def test_flops_scale_with_depth_and_width():
    e = _est()
    assert _est(num_layers=18).forward_flops_per_seq() > 1.8 * e.forward_flops_per_seq()
    assert _est(num_channels=1024).forward_flops_per_seq() > 2.5 * e.forward_flops_per_seq()
    # params follow the 12*d^2-per-layer transformer convention
    d, L = 512, 9
    assert e.params() == L * (4 * d * d + 2 * 4 * d * d)
    assert e.embedding_params() == 262 * 512


def test_cross_attention_dropout_reduces_flops():
    full = _est(cross_attention_dropout=0.0).forward_flops_per_seq()
    half = _est(cross_attention_dropout=0.5).forward_flops_per_seq()
    assert half < full


def test_train_flops_is_3x_forward():
    e = _est()
    assert e.train_flops(10) == 3.0 * e.forward_flops_per_seq() * 10


def test_power_law_fit_recovers_exponent():
    rng = np.random.default_rng(0)
    comp = np.logspace(18, 22, 12)
    y = 40.0 * comp ** -0.076 + 1.5 + rng.normal(0, 1e-4, size=comp.shape)
    a, b, c = fit_power_law(comp, y)
    assert abs(b - 0.076) < 0.01
    assert abs(c - 1.5) < 0.2
