"""Communication-avoiding solvers: CA-CG and CA-GCR
(ref: lib/inv_ca_cg.cpp, lib/inv_ca_gcr.cpp — s-step basis per cycle with
all global reductions batched into one Gram-matrix assembly per cycle,
which is the whole point on a latency-bound coarse grid)."""

from __future__ import annotations

from math import sqrt

import numpy as np

from ..fields.spinor import SpinorField
from ..ops import blas
from .cg import SolverStats


def _gram(vs):
    """Hermitian Gram matrix G_ij = <v_i, v_j> (host, n^2/2 c_dots)."""
    n = len(vs)
    G = np.empty((n, n), dtype=complex)
    for i in range(n):
        for j in range(i, n):
            G[i, j] = blas.c_dot(vs[i], vs[j])
            G[j, i] = np.conj(G[i, j])
    return G


def ca_cg_solve(op, x: SpinorField, b: SpinorField, *, tol: float = 1e-8,
                maxiter: int = 1000, basis_size: int = 4,
                basis: str = "power", lambda_min: float = 0.0,
                lambda_max: float = 10.0) -> SolverStats:
    """CA-CG on MdagM: per cycle build an s-step basis, Galerkin-project
    (V^dag A V) c = V^dag r, and update x += V c. One batched reduction
    set per s matrix applications. basis = "power"
    (V = [r, Ar, ..., A^{s-1} r]) or "chebyshev" (QudaCABasis
    enum_quda.h:207 — shifted Chebyshev T_k((A-c)/h) r, numerically
    stable at larger s; supply spectrum bounds lambda_min/max)."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2
    s = basis_size
    cheb = basis == "chebyshev"
    cc = 0.5 * (lambda_max + lambda_min)
    h = 0.5 * (lambda_max - lambda_min)

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r, tmp = new(), new()
    V = [new() for _ in range(s + 1)]
    W = [new() for _ in range(s)]  # W[j] = A V[j]
    x2 = blas.norm2(x)
    if x2 > 0.0:
        op.MdagM(r, x, tmp)
        r2 = blas.xmy_norm2(b, r)
    else:
        blas.copy(r, b)
        r2 = b2
    k = 0
    while r2 > stop and k < maxiter:
        blas.copy(V[0], r)
        for j in range(s):
            op.MdagM(W[j], V[j], tmp)
            k += 1
            if not cheb:
                blas.copy(V[j + 1], W[j])
            else:
                # V[j+1] = (2 - [j==0]) * (W[j] - c V[j])/h - V[j-1]
                fac = (1.0 if j == 0 else 2.0) / h
                blas.copy(V[j + 1], W[j])
                blas.axpy(-cc, V[j], V[j + 1])
                blas.scal(fac, V[j + 1])
                if j > 0:
                    blas.axpy(-1.0, V[j - 1], V[j + 1])
        G = np.empty((s, s), dtype=complex)
        rhs = np.empty(s, dtype=complex)
        for i in range(s):
            for j in range(s):
                G[i, j] = blas.c_dot(V[i], W[j])
            rhs[i] = blas.c_dot(V[i], r)
        try:
            c = np.linalg.solve(G, rhs)
        except np.linalg.LinAlgError:
            break
        for j in range(s):
            blas.caxpy(complex(c[j]), V[j], x)
            blas.caxpy(-complex(c[j]), W[j], r)
        r2 = blas.norm2(r)

    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.converged = r2 <= stop
    return stats


def ca_gcr_solve(op, x: SpinorField, b: SpinorField, *, tol: float = 1e-8,
                 maxiter: int = 1000, basis_size: int = 4,
                 dagger: bool = False) -> SolverStats:
    """CA-GCR on M: per cycle minimize ||r - (A V) c|| via the normal
    equations of the s-step power basis (ref: lib/inv_ca_gcr.cpp — the
    default MG smoother/coarse solver)."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2
    s = basis_size

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r = new()
    V = [new() for _ in range(s + 1)]
    x2 = blas.norm2(x)
    if x2 > 0.0:
        op.M(r, x, dagger=dagger)
        r2 = blas.xmy_norm2(b, r)
    else:
        blas.copy(r, b)
        r2 = b2
    k = 0
    while r2 > stop and k < maxiter:
        blas.copy(V[0], r)
        for j in range(s):
            op.M(V[j + 1], V[j], dagger=dagger)
            k += 1
        AV = V[1:]
        G = _gram(AV)
        rhs = np.array([blas.c_dot(av, r) for av in AV])
        try:
            c = np.linalg.lstsq(G, rhs, rcond=None)[0]
        except np.linalg.LinAlgError:
            break
        for j in range(s):
            blas.caxpy(complex(c[j]), V[j], x)
            blas.caxpy(-complex(c[j]), AV[j], r)
        r2 = blas.norm2(r)

    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.converged = r2 <= stop
    return stats
