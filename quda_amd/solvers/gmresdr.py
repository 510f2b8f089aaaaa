"""GMRES-DR: GMRES with deflated restarting (ref: lib/inv_gmresdr_quda.cpp
— re-derived from Morgan's algorithm): each restart keeps k harmonic Ritz
vectors of the Arnoldi Hessenberg as the head of the next cycle's basis,
so restarted convergence on indefinite / wrapped spectra (coarse MG ops,
the Kahler-Dirac operator) does not stall the way plain restarted
GCR/GMRES does.
"""

from __future__ import annotations

from math import sqrt
from typing import List

import numpy as np

from ..fields.spinor import SpinorField
from ..ops import blas
from .cg import SolverStats


def _new_like(x: SpinorField) -> SpinorField:
    return SpinorField(x.geo, x.precision, x.device, x.n_parity,
                       nspin=x.nspin, ls=x.ls)


def gmresdr_solve(op, x: SpinorField, b: SpinorField, *, m: int = 20,
                  k: int = 8, tol: float = 1e-8,
                  maxiter: int = 2000) -> SolverStats:
    """Solve op.M x = b (general nonsymmetric). m = Krylov cycle length,
    k = deflation subspace kept across restarts (k < m)."""
    assert 0 < k < m
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * sqrt(b2)

    r = _new_like(b)
    t = _new_like(b)
    if blas.norm2(x) > 0:
        op.M(t, x)
        blas.copy(r, b)
        blas.axpy(-1.0, t, r)
    else:
        blas.copy(r, b)

    V: List[SpinorField] = []
    H = np.zeros((m + 1, m), dtype=complex)
    c = np.zeros(m + 1, dtype=complex)
    beta = sqrt(blas.norm2(r))
    v0 = _new_like(r)
    blas.copy(v0, r)
    blas.scal(1.0 / beta, v0)
    V = [v0]
    c[0] = beta
    j = 0            # current number of filled Hessenberg columns
    total = 0
    resid = beta

    def arnoldi_step():
        nonlocal j
        w = _new_like(b)
        op.M(w, V[j])
        for i in range(j + 1):
            h = blas.c_dot(V[i], w)
            H[i, j] = h
            blas.caxpy(-h, V[i], w)
        nrm = sqrt(blas.norm2(w))
        H[j + 1, j] = nrm
        if nrm > 1e-30:
            blas.scal(1.0 / nrm, w)
        V.append(w)
        j += 1

    while total < maxiter and resid > stop:
        while j < m and total < maxiter:
            arnoldi_step()
            total += 1
            y, res, *_ = np.linalg.lstsq(H[:j + 1, :j], c[:j + 1],
                                         rcond=None)
            resid = np.linalg.norm(c[:j + 1] - H[:j + 1, :j] @ y)
            if resid <= stop:
                break
        # update x with the current cycle's correction
        y, *_ = np.linalg.lstsq(H[:j + 1, :j], c[:j + 1], rcond=None)
        for i in range(j):
            blas.caxpy(complex(y[i]), V[i], x)
        ctilde = c[:j + 1] - H[:j + 1, :j] @ y      # residual in V basis
        resid = np.linalg.norm(ctilde)
        if resid <= stop or total >= maxiter:
            break
        # ---- deflated restart: harmonic Ritz vectors of H_j ----
        Hm = H[:j, :j]
        hlast = H[j, j - 1]
        em = np.zeros(j)
        em[j - 1] = 1.0
        f = np.linalg.solve(Hm.conj().T, em) * (abs(hlast) ** 2)
        theta, G = np.linalg.eig(Hm + np.outer(f, em))
        order = np.argsort(np.abs(theta))
        G = G[:, order[:k]]
        P = np.zeros((j + 1, k + 1), dtype=complex)
        P[:j, :k] = G
        P[:, k] = ctilde
        Q, _ = np.linalg.qr(P)
        # new basis V_{k+1} = V_{j+1} Q ; new Hbar = Q^H Hbar_j Q_top
        newV = []
        for col in range(k + 1):
            vq = _new_like(b)
            vq.zero_()
            for i in range(j + 1):
                blas.caxpy(complex(Q[i, col]), V[i], vq)
            newV.append(vq)
        Hnew = Q.conj().T @ H[:j + 1, :j] @ Q[:j, :k]
        cnew = Q.conj().T @ ctilde
        V = newV
        H = np.zeros((m + 1, m), dtype=complex)
        H[:k + 1, :k] = Hnew
        c = np.zeros(m + 1, dtype=complex)
        c[:k + 1] = cnew
        j = k

    stats.iters = total
    stats.resid = float(resid) / sqrt(b2)
    stats.converged = resid <= stop
    # refresh true residual into stats
    op.M(t, x)
    stats.true_resid = sqrt(blas.xmy_norm2(b, t) / b2)
    return stats
