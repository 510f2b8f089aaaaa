"""Rational approximation of matrix powers + multishift application
(ref: the RHMC machinery behind invertMultiShiftQuda / MILC's rational
hybrid Monte Carlo — QUDA consumes externally-generated Remez
coefficients; here the partial-fraction coefficients are generated
in-tree by a weighted least-squares fit on a log-Chebyshev grid with
log-spaced poles, which is exponentially accurate for x^alpha on
[lo, hi] and validated against dense eigendecompositions in the tests).

    x^alpha  ~=  r0 + sum_l  res_l / (x + pole_l)

Application to an SPD operator A uses ONE multishift CG over all poles.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List

import numpy as np

from ..fields.spinor import SpinorField
from ..ops import blas
from .multishift import multishift_cg_solve


@dataclass
class RationalApprox:
    alpha: float
    lo: float
    hi: float
    r0: float
    res: List[float]
    poles: List[float]
    max_rel_err: float

    def evaluate(self, x):
        x = np.asarray(x, dtype=float)
        y = np.full_like(x, self.r0)
        for r, p in zip(self.res, self.poles):
            y = y + r / (x + p)
        return y


def rational_approx(alpha: float, lo: float, hi: float, n: int = 12,
                    grid: int = 2000) -> RationalApprox:
    """Fit x^alpha on [lo, hi] with n log-spaced poles; relative-error
    weighted least squares on a dense log grid."""
    assert 0 < lo < hi
    poles = np.geomspace(lo / 10.0, hi * 10.0, n)
    x = np.geomspace(lo, hi, grid)
    target = x ** alpha
    # design matrix: [1, 1/(x+p_1), ...] with relative weighting
    A = np.empty((grid, n + 1))
    A[:, 0] = 1.0
    for i, p in enumerate(poles):
        A[:, i + 1] = 1.0 / (x + p)
    # Lawson iteration: reweighting by |error| drives the weighted LSQ
    # toward the minimax (equioscillating) solution
    w = 1.0 / target
    lw = np.ones(grid)
    coef = None
    for _ in range(30):
        ww = w * np.sqrt(lw)
        coef, *_ = np.linalg.lstsq(A * ww[:, None], target * ww, rcond=None)
        err = np.abs((A @ coef) / target - 1.0)
        lw = lw * (err + 1e-14)
        lw = lw / lw.max()
    approx = RationalApprox(alpha, lo, hi, float(coef[0]),
                            [float(c) for c in coef[1:]],
                            [float(p) for p in poles], 0.0)
    approx.max_rel_err = float(np.max(np.abs(approx.evaluate(x) / target - 1)))
    return approx


def rational_apply(op, out: SpinorField, phi: SpinorField,
                   approx: RationalApprox, *, tol: float = 1e-10,
                   maxiter: int = 2000) -> SpinorField:
    """out = A^alpha phi with A = op.MdagM, via one multishift CG over the
    partial-fraction poles (negative residues allowed)."""
    shifts = list(approx.poles)
    # multishift requires ascending shifts with shifts[0] the smallest
    order = np.argsort(shifts)
    shifts_sorted = [shifts[i] for i in order]
    xs = [phi.clone_empty() for _ in shifts_sorted]
    st = multishift_cg_solve(op, xs, phi, shifts_sorted, tol=tol,
                             maxiter=maxiter)
    assert st.converged, "rational_apply multishift failed"
    blas.copy(out, phi)
    blas.scal(approx.r0, out)
    for i, oi in enumerate(order):
        blas.axpy(approx.res[oi], xs[i], out)
    return out


def rhmc_pseudofermion_action(op, phi: SpinorField, approx_inv: RationalApprox,
                              **kw) -> float:
    """S_f = phi^dag A^{-alpha} phi (e.g. alpha=1/4 for 2-flavor-rooted
    staggered: approx_inv generated for x^{-1/4} of A = MdagM)."""
    t = phi.clone_empty()
    rational_apply(op, t, phi, approx_inv, **kw)
    return blas.re_dot(phi, t)
