"""CG-family variants: CGNE/CGNR wrappers, steepest descent, preconditioned
CG, CG3 (ref: lib/inv_cgne.cpp, inv_cgnr.cpp, inv_sd_quda.cpp,
inv_pcg_quda.cpp, inv_cg3_quda.cpp)."""

from __future__ import annotations

from math import sqrt
from typing import Callable, Optional

from ..fields.spinor import SpinorField
from ..ops import blas
from .cg import SolverStats, cg_solve


class _NormalOp:
    """MdagM (use_mmdag=False) or MMdag view of an op, exposed as .MdagM."""

    def __init__(self, op, mmdag: bool = False):
        self.op = op
        self.mmdag = mmdag

    def MdagM(self, out, inp, tmp):
        if self.mmdag:
            self.op.M(tmp, inp, dagger=True)
            self.op.M(out, tmp, dagger=False)
        else:
            self.op.M(tmp, inp, dagger=False)
            self.op.M(out, tmp, dagger=True)
        return out


def cgnr_solve(op, x: SpinorField, b: SpinorField, **kw) -> SolverStats:
    """CG on the normal equations MdagM x = Mdag b (ref: inv_cgnr.cpp)."""
    bp = SpinorField(b.geo, b.precision, b.device, b.n_parity, nspin=b.nspin, ls=b.ls)
    op.M(bp, b, dagger=True)
    return cg_solve(_NormalOp(op), x, bp, **kw)


def cgne_solve(op, x: SpinorField, b: SpinorField, **kw) -> SolverStats:
    """CG on M Mdag y = b, x = Mdag y (ref: inv_cgne.cpp)."""
    y = SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)
    stats = cg_solve(_NormalOp(op, mmdag=True), y, b, **kw)
    op.M(x, y, dagger=True)
    return stats


def sd_solve(op, x: SpinorField, b: SpinorField, *, tol: float = 1e-8,
             maxiter: int = 1000) -> SolverStats:
    """Steepest descent on the hermitian PSD MdagM (ref: inv_sd_quda.cpp)."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r, Ar, tmp = new(), new(), new()
    x2 = blas.norm2(x)
    if x2 > 0.0:
        op.MdagM(r, x, tmp)
        r2 = blas.xmy_norm2(b, r)
    else:
        blas.copy(r, b)
        r2 = b2
    k = 0
    while r2 > stop and k < maxiter:
        op.MdagM(Ar, r, tmp)
        rAr = blas.re_dot(r, Ar)
        if rAr <= 0.0:
            break
        alpha = r2 / rAr
        blas.axpy(alpha, r, x)
        r2 = blas.axpy_norm2(-alpha, Ar, r)
        k += 1
    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.converged = r2 <= stop
    return stats


def pcg_solve(op, x: SpinorField, b: SpinorField, *,
              precond: Callable[[SpinorField, SpinorField], None],
              tol: float = 1e-8, maxiter: int = 1000) -> SolverStats:
    """Preconditioned CG on MdagM with SPD preconditioner K
    (ref: lib/inv_pcg_quda.cpp): precond(z, r) applies z = K r."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r, z, p, Ap, tmp = new(), new(), new(), new(), new()
    x2 = blas.norm2(x)
    if x2 > 0.0:
        op.MdagM(r, x, tmp)
        r2 = blas.xmy_norm2(b, r)
    else:
        blas.copy(r, b)
        r2 = b2
    precond(z, r)
    blas.copy(p, z)
    rz = blas.re_dot(r, z)
    k = 0
    while r2 > stop and k < maxiter:
        op.MdagM(Ap, p, tmp)
        pAp = blas.re_dot(p, Ap)
        if pAp <= 0.0:
            break
        alpha = rz / pAp
        blas.axpy(alpha, p, x)
        r2 = blas.axpy_norm2(-alpha, Ap, r)
        k += 1
        if r2 <= stop:
            break
        precond(z, r)
        rz_new = blas.re_dot(r, z)
        beta = rz_new / rz
        rz = rz_new
        blas.xpay(z, beta, p)
    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.converged = r2 <= stop
    return stats


def cg3_solve(op, x: SpinorField, b: SpinorField, *, tol: float = 1e-8,
              maxiter: int = 1000) -> SolverStats:
    """Three-term recurrence CG (ref: lib/inv_cg3_quda.cpp)."""
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    def new():
        return SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)

    r, Ar, tmp = new(), new(), new()
    x_prev, r_prev = new(), new()
    x.zero_()
    blas.copy(r, b)
    r2 = b2
    rho = 1.0
    k = 0
    r2_prev = r2
    while r2 > stop and k < maxiter:
        op.MdagM(Ar, r, tmp)
        rAr = blas.re_dot(r, Ar)
        if rAr <= 0.0:
            break
        gamma = r2 / rAr
        if k == 0:
            rho_new = 1.0
        else:
            d = 1.0 - (gamma / gamma_prev) * (r2 / r2_prev) / rho
            if d == 0.0:
                break
            rho_new = 1.0 / d
        # x_{k+1} = rho(x + gamma r) + (1-rho) x_prev
        # r_{k+1} = rho(r - gamma Ar) + (1-rho) r_prev
        xs, rs = new(), new()
        blas.copy(xs, x)
        blas.copy(rs, r)
        blas.axpy(gamma, r, x)
        blas.axpy(-gamma, Ar, r)
        if k > 0:
            blas.axpby(1.0 - rho_new, x_prev, rho_new, x)
            blas.axpby(1.0 - rho_new, r_prev, rho_new, r)
        blas.copy(x_prev, xs)
        blas.copy(r_prev, rs)
        r2_prev = r2
        r2 = blas.norm2(r)
        gamma_prev = gamma
        rho = rho_new
        k += 1
    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.converged = r2 <= stop
    return stats
