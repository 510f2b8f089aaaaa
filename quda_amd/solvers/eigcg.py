"""Incremental eigCG: CG that harvests low eigenpairs of the hermitian
operator while solving, and deflates subsequent right-hand sides with the
accumulated basis (ref: lib/inv_eigcg_quda.cpp — re-derived from the
Stathopoulos-Orginos algorithm; the thick restart here recomputes the
post-restart Lanczos coupling with ONE explicit matvec instead of the
closed-form boundary row, trading a rare extra apply for index-proof
correctness).

CG <-> Lanczos: with v_j = r_j/||r_j||, the projected operator is
tridiagonal with
    T[j,j]   = 1/alpha_j + beta_{j-1}/alpha_{j-1}
    T[j,j+1] = -sqrt(beta_j)/alpha_j
When the m-vector window fills: Rayleigh-Ritz on T_m AND T_{m-1}
(2*nev vectors total, the key eigCG trick that keeps convergence), the
window restarts with those Ritz vectors, and iteration continues.
"""

from __future__ import annotations

from math import sqrt
from typing import List, Optional

import numpy as np

from ..fields.spinor import SpinorField
from ..ops import blas
from .cg import SolverStats


def _new_like(x: SpinorField) -> SpinorField:
    return SpinorField(x.geo, x.precision, x.device, x.n_parity,
                       nspin=x.nspin, ls=x.ls)


class IncrementalDeflation:
    """A-orthonormal accumulated eigenspace across RHS: U, lam_i = u^d A u.
    Initial guess x0 = sum_i u_i <u_i, b>/lam_i (Galerkin with
    near-eigenvectors)."""

    def __init__(self, max_vec: int = 64):
        self.max_vec = max_vec
        self.U: List[SpinorField] = []
        self.lam: List[float] = []

    def add(self, op, vecs: List[SpinorField]) -> int:
        added = 0
        for v in vecs:
            if len(self.U) >= self.max_vec:
                break
            w = _new_like(v)
            blas.copy(w, v)
            for u in self.U:  # Gram-Schmidt against the accumulated basis
                blas.caxpy(-blas.c_dot(u, w), u, w)
            n2 = blas.norm2(w)
            if n2 < 1e-8:
                continue
            blas.scal(1.0 / sqrt(n2), w)
            t = _new_like(w)
            tmp = _new_like(w)
            op.MdagM(t, w, tmp)
            self.U.append(w)
            self.lam.append(blas.re_dot(w, t))
            added += 1
        return added

    def guess(self, x: SpinorField, b: SpinorField) -> SpinorField:
        x.zero_()
        for u, lam in zip(self.U, self.lam):
            if abs(lam) < 1e-30:
                continue
            blas.caxpy(blas.c_dot(u, b) / lam, u, x)
        return x


def eigcg_solve(op, x: SpinorField, b: SpinorField, *, nev: int = 4,
                m: int = 16, tol: float = 1e-8, maxiter: int = 1000,
                harvest: Optional[list] = None) -> SolverStats:
    """CG solve of op.MdagM x = b that appends up to `nev` (lam, vec)
    Ritz pairs to `harvest` (a list) as a side effect."""
    assert m > 2 * nev + 1, "window must exceed 2*nev+1"
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2
    r = _new_like(b)
    t = _new_like(b)
    tmp = _new_like(b)
    if blas.norm2(x) > 0:
        op.MdagM(t, x, tmp)
        blas.copy(r, b)
        blas.axpy(-1.0, t, r)
    else:
        blas.copy(r, b)
    p = _new_like(b)
    blas.copy(p, r)
    Ap = _new_like(b)
    r2 = blas.norm2(r)

    track = harvest is not None
    V: List[SpinorField] = []
    T = np.zeros((m + 1, m + 1))
    k = 0

    def push(rvec, rho):
        nonlocal k
        v = _new_like(rvec)
        blas.copy(v, rvec)
        blas.scal(1.0 / rho, v)
        V.append(v)
        k += 1

    def restart(rvec, rho):
        """Thick restart: Rayleigh-Ritz over T_m and T_{m-1} (the eigCG
        double-window), then one explicit matvec for the coupling of the
        incoming residual vector."""
        nonlocal k, T, V
        Tm = T[:m, :m].copy()
        w1, y1 = np.linalg.eigh(Tm)
        w2, y2 = np.linalg.eigh(Tm[:m - 1, :m - 1])
        Y = np.zeros((m, 2 * nev))
        Y[:, :nev] = y1[:, :nev]
        Y[:m - 1, nev:] = y2[:, :nev]
        Q, _ = np.linalg.qr(Y)
        hw, hz = np.linalg.eigh(Q.T @ Tm @ Q)
        M_ = Q @ hz
        newV = []
        for j in range(M_.shape[1]):
            vj = _new_like(rvec)
            vj.zero_()
            for i in range(m):
                blas.axpy(float(M_[i, j]), V[i], vj)
            newV.append(vj)
        V = newV
        k = len(V)
        T = np.zeros((m + 1, m + 1))
        T[:k, :k] = np.diag(hw)
        vnew = _new_like(rvec)
        blas.copy(vnew, rvec)
        blas.scal(1.0 / rho, vnew)
        av = _new_like(rvec)
        op.MdagM(av, vnew, tmp)
        for i in range(k):
            T[i, k] = T[k, i] = blas.re_dot(V[i], av)
        T[k, k] = blas.re_dot(vnew, av)
        V.append(vnew)
        k += 1

    if track:
        push(r, sqrt(r2))
    it = 0
    alpha_old, beta_old = None, 0.0
    explicit_diag = False  # restart() already set the new vector's diagonal
    while it < maxiter and r2 > stop:
        op.MdagM(Ap, p, tmp)
        alpha = r2 / blas.re_dot(p, Ap)
        if track and not explicit_diag:
            idx = k - 1
            T[idx, idx] = 1.0 / alpha + (beta_old / alpha_old
                                         if alpha_old else 0.0)
        blas.axpy(alpha, p, x)
        r2_old = r2
        r2 = blas.axpy_norm2(-alpha, Ap, r)
        beta = r2 / r2_old
        if track:
            idx = k - 1
            if k == m:
                restart(r, sqrt(r2) if r2 > 0 else 1.0)
                explicit_diag = True
            else:
                T[idx, idx + 1] = T[idx + 1, idx] = -sqrt(beta) / alpha
                push(r, sqrt(r2) if r2 > 0 else 1.0)
                explicit_diag = False
        blas.xpay(r, beta, p)
        alpha_old, beta_old = alpha, beta
        it += 1
    stats.iters = it
    stats.resid = sqrt(r2 / b2)
    stats.converged = r2 <= stop
    if track and k > 1:
        kk = k - (0 if explicit_diag else 1)  # drop the half-filled tail
        kk = max(kk, 1)
        w, z = np.linalg.eigh(T[:kk, :kk])
        for j in range(min(nev, kk)):
            u = _new_like(b)
            u.zero_()
            for i in range(kk):
                blas.axpy(float(z[i, j]), V[i], u)
            n2 = blas.norm2(u)
            if n2 > 1e-12:
                blas.scal(1.0 / sqrt(n2), u)
                harvest.append((float(w[j]), u))
    return stats


def inc_eigcg_solve(op, xs: List[SpinorField], bs: List[SpinorField], *,
                    nev: int = 4, m: int = 16, max_defl: int = 48,
                    tol: float = 1e-8, maxiter: int = 1000,
                    defl: Optional[IncrementalDeflation] = None):
    """Solve the sequence MdagM x_i = b_i, harvesting eigenpairs from each
    solve and deflating the next (ref: incremental eigCG driver,
    inv_eigcg_quda.cpp RestartVT/SearchSpaceUpdate roles). Returns
    (stats_list, deflation)."""
    defl = defl or IncrementalDeflation(max_defl)
    out = []
    for x, b in zip(xs, bs):
        if defl.U:
            defl.guess(x, b)
        harvest: list = []
        st = eigcg_solve(op, x, b, nev=nev, m=m, tol=tol, maxiter=maxiter,
                         harvest=harvest)
        defl.add(op, [u for _, u in harvest])
        out.append(st)
    return out, defl
