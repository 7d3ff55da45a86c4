"""NumPy reference implementation of the MHD solver (periodic global grid).

Mirrors csrc/src/mhd.hip exactly (same 6th-order coefficients, same
equations, same Williamson RK3 two-buffer update); used to verify the HIP
kernel's numerics on small grids.
"""
from __future__ import annotations

import numpy as np

LNRHO, UUX, UUY, UUZ, AAX, AAY, AAZ, SS = range(8)
D1 = (3.0 / 4.0, -3.0 / 20.0, 1.0 / 60.0)
D2 = (-49.0 / 18.0, 3.0 / 2.0, -3.0 / 20.0, 1.0 / 90.0)
ALPHA = (0.0, -5.0 / 9.0, -153.0 / 128.0)
BETA = (1.0 / 3.0, 15.0 / 16.0, 8.0 / 15.0)

_AXIS = {"x": 2, "y": 1, "z": 0}


def _sh(f, d, axis):
    """f shifted by +d cells along axis (periodic)"""
    return np.roll(f, -d, axis=_AXIS[axis])


def d1(f, axis, ds):
    a = axis
    return (
        D1[0] * (_sh(f, 1, a) - _sh(f, -1, a))
        + D1[1] * (_sh(f, 2, a) - _sh(f, -2, a))
        + D1[2] * (_sh(f, 3, a) - _sh(f, -3, a))
    ) / ds


def d2(f, axis, ds):
    a = axis
    return (
        D2[0] * f
        + D2[1] * (_sh(f, 1, a) + _sh(f, -1, a))
        + D2[2] * (_sh(f, 2, a) + _sh(f, -2, a))
        + D2[3] * (_sh(f, 3, a) + _sh(f, -3, a))
    ) / (ds * ds)


def dcross(f, ax1, ds1, ax2, ds2):
    s = np.zeros_like(f)
    for i in (1, 2, 3):
        for j in (1, 2, 3):
            term = (
                _sh(_sh(f, i, ax1), j, ax2)
                - _sh(_sh(f, i, ax1), -j, ax2)
                - _sh(_sh(f, -i, ax1), j, ax2)
                + _sh(_sh(f, -i, ax1), -j, ax2)
            )
            s += D1[i - 1] * D1[j - 1] * term
    return s / (ds1 * ds2)


def rhs(F, cf):
    """F: list of 8 (z,y,x) fp64 arrays. cf: dict with dsx..chi."""
    dsx, dsy, dsz = cf["dsx"], cf["dsy"], cf["dsz"]
    lap = lambda f: d2(f, "x", dsx) + d2(f, "y", dsy) + d2(f, "z", dsz)
    gx = lambda f: d1(f, "x", dsx)
    gy = lambda f: d1(f, "y", dsy)
    gz = lambda f: d1(f, "z", dsz)

    u = [F[UUX], F[UUY], F[UUZ]]
    glnr = (gx(F[LNRHO]), gy(F[LNRHO]), gz(F[LNRHO]))
    gss = (gx(F[SS]), gy(F[SS]), gz(F[SS]))

    du = [[gx(c), gy(c), gz(c)] for c in u]  # du[i][j] = d u_i / d x_j
    # separable formulation (matches csrc/src/mhd.hip exactly): div fields
    # are computed, stored, and differentiated -- grad(div .) is three
    # 7-point first derivatives of one field
    divu = gx(u[0]) + gy(u[1]) + gz(u[2])
    lap_u = [lap(c) for c in u]
    graddiv_u = [gx(divu), gy(divu), gz(divu)]

    A = [F[AAX], F[AAY], F[AAZ]]
    B = (gy(A[2]) - gz(A[1]), gz(A[0]) - gx(A[2]), gx(A[1]) - gy(A[0]))
    lap_a = [lap(c) for c in A]
    divA = gx(A[0]) + gy(A[1]) + gz(A[2])
    j = [gx(divA) - lap_a[0], gy(divA) - lap_a[1], gz(divA) - lap_a[2]]
    rho_inv = np.exp(-F[LNRHO])

    jxB = (
        j[1] * B[2] - j[2] * B[1],
        j[2] * B[0] - j[0] * B[2],
        j[0] * B[1] - j[1] * B[0],
    )
    uxB = (
        u[1] * B[2] - u[2] * B[1],
        u[2] * B[0] - u[0] * B[2],
        u[0] * B[1] - u[1] * B[0],
    )
    ugradu = [u[0] * du[i][0] + u[1] * du[i][1] + u[2] * du[i][2] for i in range(3)]
    press = [glnr[i] + cf["cp_inv"] * gss[i] for i in range(3)]

    out = [None] * 8
    out[LNRHO] = -(u[0] * glnr[0] + u[1] * glnr[1] + u[2] * glnr[2]) - divu
    for i, q in enumerate((UUX, UUY, UUZ)):
        out[q] = (
            -ugradu[i]
            - cf["cs2"] * press[i]
            + rho_inv * jxB[i]
            + cf["nu"] * (lap_u[i] + graddiv_u[i] / 3.0)
        )
    for i, q in enumerate((AAX, AAY, AAZ)):
        out[q] = uxB[i] + cf["eta"] * lap_a[i]
    out[SS] = -(u[0] * gss[0] + u[1] * gss[1] + u[2] * gss[2]) + cf["chi"] * lap(F[SS])
    return out


def substep(curr, nxt, step, dt, cf):
    """Williamson two-buffer update on full periodic arrays.
    Returns (new_curr, new_next) after the post-substep swap."""
    r = rhs(curr, cf)
    aob = 0.0 if step == 0 else ALPHA[step] / BETA[step - 1]
    new_next = [
        c + BETA[step] * (aob * (c - p) + dt * rr) for c, p, rr in zip(curr, nxt, r)
    ]
    return new_next, curr  # swapped
