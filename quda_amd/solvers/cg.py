"""Conjugate gradient with mixed precision + reliable updates
(ref: lib/inv_cg_quda.cpp:63 operator(), include/reliable_updates.h:33 —
same algorithm structure, re-implemented).

Solves MdagM x = b where `op` exposes MdagM(out, in, tmp) on single- or
full-parity SpinorFields. Outer (precise) residual bookkeeping runs at
`x.precision`; the Krylov iteration runs at `sloppy` precision. A reliable
update (true-residual recomputation + solution accumulation) triggers when
the iterated residual falls by `delta` relative to the max since the last
update.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from math import sqrt
from typing import Callable, Optional

from ..fields.spinor import SpinorField
from ..ops import blas


@dataclass
class SolverStats:
    iters: int = 0
    resid: float = 0.0
    true_resid: float = 0.0
    reliable_updates: int = 0
    converged: bool = False
    hq_resid: float = 0.0
    flops: float = 0.0
    seconds: float = 0.0


def hq_residual(x: SpinorField, r: SpinorField) -> float:
    """Fermilab heavy-quark residual: sqrt(mean_x |r(x)|^2/|x(x)|^2)
    (ref: the HQ-residual path of inv_cg_quda.cpp hqsolve:438 /
    QUDA_HEAVY_QUARK_RESIDUAL)."""
    import torch
    xv = x.to_complex()
    rv = r.to_complex()
    dims = tuple(range(2, xv.dim()))
    x2 = xv.abs().square().sum(dim=dims).clamp_min(1e-300)
    r2 = rv.abs().square().sum(dim=dims)
    from ..parallel import comms
    num = comms.allreduce_sum(float((r2 / x2).sum()))
    den = comms.allreduce_sum(float(x2.numel()))
    return sqrt(num / den)


def cg_solve(op, x: SpinorField, b: SpinorField, *,
             op_sloppy=None, sloppy: Optional[str] = None,
             tol: float = 1e-8, maxiter: int = 1000,
             delta: float = 0.1, hq_tol: float = 0.0) -> SolverStats:
    """CG on the (hermitian PSD) operator op.MdagM.

    x: initial guess (overwritten with solution), b: source — both at the
    "precise" precision. op_sloppy/sloppy select the inner precision
    (default: same operator, same precision => plain CG). hq_tol > 0
    ADDS the Fermilab heavy-quark residual criterion: convergence then
    requires BOTH the L2 and the HQ residual targets (checked at
    reliable updates / completion; stats.hq_resid reports the final
    value).
    """
    stats = SolverStats()
    prec_hi = x.precision
    sloppy = sloppy or prec_hi
    op_sloppy = op_sloppy or op
    geo, dev, npar, nsp, nls = x.geo, x.device, x.n_parity, x.nspin, x.ls

    def hi():
        return SpinorField(geo, prec_hi, dev, npar, nspin=nsp, ls=nls)

    def lo():
        return SpinorField(geo, sloppy, dev, npar, nspin=nsp, ls=nls)

    b2 = blas.norm2(b)
    if b2 == 0.0:
        x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    r = hi()           # precise residual
    tmp_hi = hi()
    # r = b - MdagM x   (skip apply if x == 0)
    x2 = blas.norm2(x)
    if x2 > 0.0:
        op.MdagM(r, x, tmp_hi)
        r2 = blas.xmy_norm2(b, r)
    else:
        blas.copy(r, b)
        r2 = b2

    mixed = (sloppy != prec_hi) or (op_sloppy is not op)
    r_s = lo()
    p = lo()
    Ap = lo()
    x_s = lo()         # sloppy solution accumulator since last update
    tmp_s = lo()
    blas.copy(r_s, r)
    blas.copy(p, r_s)
    x_s.zero_()

    maxr = sqrt(r2)    # max residual since last reliable update
    k = 0
    x_s_dirty = False  # x_s holds progress not yet folded into x
    while r2 > stop and k < maxiter:
        op_sloppy.MdagM(Ap, p, tmp_s)
        pAp = blas.re_dot(p, Ap)
        if pAp <= 0.0:
            break  # breakdown (ref: inv_cg_quda.cpp:265 checks)
        alpha = r2 / pAp
        x_s_dirty = True
        r2_old = r2
        # fused: x_s += alpha p; r_s -= alpha Ap; r2 = ||r_s||^2
        r2 = blas.triple_cg_update(alpha, p, Ap, x_s, r_s)
        k += 1
        rnorm = sqrt(r2)
        maxr = max(maxr, rnorm)

        need_reliable = mixed and (rnorm < delta * maxr)
        if need_reliable or r2 <= stop:
            # accumulate + recompute true residual at high precision
            blas.copy(tmp_hi, x_s)
            blas.axpy(1.0, tmp_hi, x)
            op.MdagM(r, x, tmp_hi)
            r2 = blas.xmy_norm2(b, r)
            x_s.zero_()
            x_s_dirty = False
            blas.copy(r_s, r)
            # restart direction with beta continuation:
            # p = r_s + beta p  with beta = r2_new / r2_old
            beta = r2 / r2_old
            blas.xpay(r_s, beta, p)
            maxr = sqrt(r2)
            stats.reliable_updates += 1
            if r2 <= stop:
                break
        else:
            beta = r2 / r2_old
            blas.xpay(r_s, beta, p)

    if x_s_dirty:
        # Fold the sloppy accumulator into x unconditionally — a maxiter
        # or pAp<=0 exit must not discard progress since the last
        # reliable update (ref: inv_cg_quda.cpp copies xSloppy back after
        # the loop) — and recompute the true residual.
        blas.copy(tmp_hi, x_s)
        blas.axpy(1.0, tmp_hi, x)
        op.MdagM(r, x, tmp_hi)
        r2 = blas.xmy_norm2(b, r)

    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.true_resid = stats.resid
    stats.converged = r2 <= stop
    if hq_tol > 0.0 and stats.converged:
        stats.hq_resid = hq_residual(x, r)
        if stats.hq_resid > hq_tol:
            # polish until the HQ criterion also holds (restarted CG on
            # the current residual; bounded by maxiter total)
            extra = 0
            while stats.hq_resid > hq_tol and k + extra < maxiter:
                st2 = cg_solve(op, x, b, op_sloppy=op_sloppy,
                               sloppy=sloppy, tol=stats.resid * 0.1,
                               maxiter=maxiter - k - extra, delta=delta)
                extra += st2.iters
                op.MdagM(r, x, tmp_hi)
                blas.xmy_norm2(b, r)
                stats.hq_resid = hq_residual(x, r)
                if st2.iters == 0:
                    break
            stats.iters = k + extra
            stats.converged = stats.hq_resid <= hq_tol
    return stats
