"""BiCGStab and BiCGStab(l) (ref: lib/inv_bicgstab_quda.cpp,
lib/inv_bicgstabl_quda.cpp — same algorithms, re-derived; operate on the
non-hermitian M so even-odd systems converge without the normal op)."""

from __future__ import annotations

from math import sqrt

import numpy as np

from ..fields.spinor import SpinorField
from ..ops import blas
from .cg import SolverStats


def bicgstab_solve(op, x: SpinorField, b: SpinorField, *, tol: float = 1e-8,
                   maxiter: int = 1000, dagger: bool = False) -> SolverStats:
    """Solve M x = b (M = op.M; dagger=True solves Mdag x = b)."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r, r0, p, v, t = new(), new(), new(), new(), new()
    # r = b - M x
    x2 = blas.norm2(x)
    if x2 > 0.0:
        op.M(r, x, dagger=dagger)
        blas.xmy_norm2(b, r)
    else:
        blas.copy(r, b)
    blas.copy(r0, r)
    blas.copy(p, r)
    rho = blas.c_dot(r0, r)
    r2 = blas.norm2(r)
    k = 0
    while r2 > stop and k < maxiter:
        op.M(v, p, dagger=dagger)
        r0v = blas.c_dot(r0, v)
        if abs(r0v) == 0.0:
            break
        alpha = rho / r0v
        blas.caxpy(-alpha, v, r)            # s = r - alpha v (in place)
        op.M(t, r, dagger=dagger)
        t2 = blas.norm2(t)
        if t2 == 0.0:
            blas.caxpy(alpha, p, x)
            r2 = blas.norm2(r)
            k += 1
            break
        omega = blas.c_dot(t, r) / t2
        blas.caxpy(alpha, p, x)
        blas.caxpy(omega, r, x)             # x += alpha p + omega s
        blas.caxpy(-omega, t, r)            # r = s - omega t
        r2 = blas.norm2(r)
        rho_new = blas.c_dot(r0, r)
        if abs(rho) == 0.0 or abs(omega) == 0.0:
            break
        beta = (rho_new / rho) * (alpha / omega)
        rho = rho_new
        # p = r + beta (p - omega v)
        blas.caxpy(-omega, v, p)
        blas.caxpby(1.0, r, beta, p)
        k += 1

    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.true_resid = stats.resid
    stats.converged = r2 <= stop
    return stats


def bicgstabl_solve(op, x: SpinorField, b: SpinorField, *, L: int = 2,
                    tol: float = 1e-8, maxiter: int = 1000) -> SolverStats:
    """BiCGStab(L): L BiCG steps + an L-th degree MR polynomial per cycle
    (ref: lib/inv_bicgstabl_quda.cpp; the MR coefficients come from a dense
    least-squares on the small Gram matrix, solved on the host)."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r = [new() for _ in range(L + 1)]
    u = [new() for _ in range(L + 1)]
    r0 = new()
    x2 = blas.norm2(x)
    if x2 > 0.0:
        op.M(r[0], x)
        blas.xmy_norm2(b, r[0])
    else:
        blas.copy(r[0], b)
    blas.copy(r0, r[0])
    rho0, alpha, omega = 1.0 + 0j, 0.0 + 0j, 1.0 + 0j
    r2 = blas.norm2(r[0])
    k = 0
    while r2 > stop and k < maxiter:
        rho0 = -omega * rho0
        # ---- BiCG part ----
        breakdown = False
        for j in range(L):
            rho1 = blas.c_dot(r0, r[j])
            if abs(rho0) == 0.0:
                breakdown = True
                break
            beta = alpha * rho1 / rho0
            rho0 = rho1
            for i in range(j + 1):
                blas.caxpby(1.0, r[i], -beta, u[i])   # u_i = r_i - beta u_i
            op.M(u[j + 1], u[j])
            gamma = blas.c_dot(r0, u[j + 1])
            if abs(gamma) == 0.0:
                breakdown = True
                break
            alpha = rho0 / gamma
            for i in range(j + 1):
                blas.caxpy(-alpha, u[i + 1], r[i])
            op.M(r[j + 1], r[j])
            blas.caxpy(alpha, u[0], x)
            k += 1
        if breakdown:
            break
        # ---- MR part: minimize ||r_0 - sum_j g_j r_j|| over g ----
        G = np.empty((L, L), dtype=complex)
        c = np.empty(L, dtype=complex)
        for i in range(1, L + 1):
            for j in range(1, i + 1):
                G[i - 1, j - 1] = blas.c_dot(r[i], r[j])
                G[j - 1, i - 1] = np.conj(G[i - 1, j - 1])
            c[i - 1] = blas.c_dot(r[i], r[0])
        try:
            g = np.linalg.solve(G, c)
        except np.linalg.LinAlgError:
            break
        omega = complex(g[L - 1])
        if abs(omega) == 0.0:
            break
        # x += sum g_j r_{j-1};  r_0 -= sum g_j r_j;  u_0 -= sum g_j u_j
        for j in range(1, L + 1):
            blas.caxpy(complex(g[j - 1]), r[j - 1], x)
            blas.caxpy(-complex(g[j - 1]), r[j], r[0])
            blas.caxpy(-complex(g[j - 1]), u[j], u[0])
        r2 = blas.norm2(r[0])

    stats.iters = k
    stats.resid = sqrt(max(r2, 0.0) / b2)
    stats.true_resid = stats.resid
    stats.converged = r2 <= stop
    return stats
