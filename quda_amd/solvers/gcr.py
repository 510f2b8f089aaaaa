"""GCR (flexible, restarted) and MR (ref: lib/inv_gcr_quda.cpp,
lib/inv_mr_quda.cpp). GCR is the outer solver for multigrid-preconditioned
and domain-decomposed solves; MR is the standard MG smoother."""

from __future__ import annotations

from math import sqrt
from typing import Callable, Optional

from ..fields.spinor import SpinorField
from ..ops import blas
from .cg import SolverStats


def gcr_solve(op, x: SpinorField, b: SpinorField, *, tol: float = 1e-8,
              maxiter: int = 1000, nkrylov: int = 10,
              precond: Optional[Callable[[SpinorField, SpinorField], None]] = None
              ) -> SolverStats:
    """Flexible GCR with restart length `nkrylov`. `precond(z, r)` applies
    the (possibly nonlinear/changing) preconditioner K: z ~= M^-1 r."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r = new()
    x2 = blas.norm2(x)
    if x2 > 0.0:
        op.M(r, x)
        blas.xmy_norm2(b, r)
    else:
        blas.copy(r, b)
    r2 = blas.norm2(r)

    p = [new() for _ in range(nkrylov)]
    Ap = [new() for _ in range(nkrylov)]
    k = 0
    while r2 > stop and k < maxiter:
        # build one restart cycle
        alphas = []
        m = 0
        while m < nkrylov and r2 > stop and k < maxiter:
            if precond is not None:
                precond(p[m], r)
            else:
                blas.copy(p[m], r)
            op.M(Ap[m], p[m])
            # modified Gram-Schmidt against previous Ap's
            for i in range(m):
                beta = blas.c_dot(Ap[i], Ap[m])
                blas.caxpy(-beta, Ap[i], Ap[m])
                blas.caxpy(-beta, p[i], p[m])
            Ap2 = blas.norm2(Ap[m])
            if Ap2 == 0.0:
                break
            inv = 1.0 / sqrt(Ap2)
            blas.scal(inv, Ap[m])
            blas.scal(inv, p[m])
            alpha = blas.c_dot(Ap[m], r)
            blas.caxpy(alpha, p[m], x)
            blas.caxpy(-alpha, Ap[m], r)
            r2 = blas.norm2(r)
            alphas.append(alpha)
            m += 1
            k += 1
        if m == 0:
            break
        # recompute true residual at restart
        op.M(r, x)
        r2 = blas.xmy_norm2(b, r)

    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.true_resid = stats.resid
    stats.converged = r2 <= stop
    return stats


def mr_solve(op, x: SpinorField, b: SpinorField, *, tol: float = 1e-8,
             maxiter: int = 100, omega: float = 1.0,
             zero_init: bool = True) -> SolverStats:
    """Minimal residual iteration x += omega <Ar,r>/<Ar,Ar> r
    (ref: lib/inv_mr_quda.cpp; omega<1 under-relaxes for smoothing)."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r, Ar = new(), new()
    if zero_init:
        x.zero_()
        blas.copy(r, b)
        r2 = b2
    else:
        op.M(r, x)
        r2 = blas.xmy_norm2(b, r)
    k = 0
    while r2 > stop and k < maxiter:
        op.M(Ar, r)
        Ar2 = blas.norm2(Ar)
        if Ar2 == 0.0:
            break
        alpha = blas.c_dot(Ar, r) / Ar2
        blas.caxpy(omega * alpha, r, x)
        blas.caxpy(-omega * alpha, Ar, r)
        r2 = blas.norm2(r)
        k += 1

    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.true_resid = stats.resid
    stats.converged = r2 <= stop
    return stats
