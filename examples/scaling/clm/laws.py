"""Chinchilla-style power-law fits over scaling-study validation losses (parity:
reference examples/scaling/laws.py): L(C) = a * C^-b + c fitted on (compute, loss)
pairs via log-space least squares with a grid search over the irreducible term."""
from __future__ import annotations

import numpy as np


def fit_power_law(compute: np.ndarray, loss: np.ndarray, c_grid=None):
    """Returns (a, b, c) for loss ~= a * compute^-b + c."""
    compute = np.asarray(compute, dtype=np.float64)
    loss = np.asarray(loss, dtype=np.float64)
    if c_grid is None:
        c_grid = np.linspace(0.0, loss.min() * 0.99, 200)
    best = None
    for c in c_grid:
        y = np.log(loss - c)
        x = np.log(compute)
        A = np.stack([x, np.ones_like(x)], axis=1)
        coef, res, *_ = np.linalg.lstsq(A, y, rcond=None)
        slope, intercept = coef
        pred = np.exp(intercept) * compute ** slope + c
        err = float(((pred - loss) ** 2).mean())
        if best is None or err < best[0]:
            best = (err, np.exp(intercept), -slope, c)
    _, a, b, c = best
    return a, b, c


def predict(a, b, c, compute):
    return a * np.asarray(compute, dtype=np.float64) ** -b + c
