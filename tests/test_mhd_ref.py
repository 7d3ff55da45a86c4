"""Sanity tests of the NumPy MHD reference (no GPU): operator accuracy on
analytic fields and physical invariants of the RHS."""
import math

import numpy as np

from stencil_amd.models import mhd_ref as M


def grid(n=32):
    z, y, x = np.meshgrid(
        np.arange(n, dtype=np.float64),
        np.arange(n, dtype=np.float64),
        np.arange(n, dtype=np.float64),
        indexing="ij",
    )
    return x, y, z


def test_d1_d2_accuracy():
    n = 32
    x, _, _ = grid(n)
    L = 2 * math.pi
    ds = L / n
    f = np.sin(2 * math.pi * x / n)
    k = 2 * math.pi / L
    d1 = M.d1(f, "x", ds)
    want = k * np.cos(k * x * ds)
    assert np.abs(d1 - want).max() < 1e-6  # 6th order at 32 points
    d2 = M.d2(f, "x", ds)
    assert np.abs(d2 + k * k * f).max() < 1e-6


def test_cross_derivative_matches_composition():
    n = 16
    x, y, _ = grid(n)
    f = np.sin(2 * math.pi * x / n) * np.cos(4 * math.pi * y / n)
    a = M.dcross(f, "x", 1.0, "y", 1.0)
    b = M.d1(M.d1(f, "x", 1.0), "y", 1.0)
    np.testing.assert_allclose(a, b, atol=1e-12)


def test_rhs_zero_on_uniform_state():
    n = 16
    F = [np.zeros((n, n, n)) for _ in range(8)]
    F[M.LNRHO] += 0.3
    F[M.SS] += 0.1
    cf = dict(dsx=0.1, dsy=0.1, dsz=0.1, cs2=1.0, cp_inv=1.0, nu=1e-2, eta=1e-2, chi=1e-3)
    out = M.rhs(F, cf)
    for q, r in enumerate(out):
        assert np.abs(r).max() < 1e-12, f"field {q} rhs not zero on uniform state"


def test_substep_swaps_and_updates():
    n = 16
    rng = np.random.default_rng(0)
    curr = [rng.normal(0, 0.01, (n, n, n)) for _ in range(8)]
    nxt = [np.zeros((n, n, n)) for _ in range(8)]
    cf = dict(dsx=0.1, dsy=0.1, dsz=0.1, cs2=1.0, cp_inv=1.0, nu=1e-2, eta=1e-2, chi=1e-3)
    c1, n1 = M.substep(curr, nxt, 0, 1e-4, cf)
    # after substep 0: new curr = f1, new next = f0
    for q in range(8):
        np.testing.assert_array_equal(n1[q], curr[q])
        assert not np.array_equal(c1[q], curr[q])
    # three substeps run without blowup on smooth data
    c2, n2 = M.substep(c1, n1, 1, 1e-4, cf)
    c3, _ = M.substep(c2, n2, 2, 1e-4, cf)
    for q in range(8):
        assert np.isfinite(c3[q]).all()
        assert np.abs(c3[q] - curr[q]).max() < 0.01
