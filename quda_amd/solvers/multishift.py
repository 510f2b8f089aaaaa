"""Multi-shift CG: solve (MdagM + sigma_i) x_i = b for all shifts in one
Krylov space (ref: lib/inv_multi_cg_quda.cpp — zeta/beta/alpha shifted
recurrences, Jegerlehner hep-lat/9612014; up to QUDA_MAX_MULTI_SHIFT=32
shifts, quda_constants.h:31)."""

from __future__ import annotations

from math import sqrt
from typing import List, Sequence

from ..fields.spinor import SpinorField
from ..ops import blas
from .cg import SolverStats


def multishift_cg_solve(op, xs: List[SpinorField], b: SpinorField,
                        shifts: Sequence[float], *, tol: float = 1e-8,
                        maxiter: int = 1000) -> SolverStats:
    """Shifted CG: xs[i] <- (MdagM + shifts[i])^-1 b.

    shifts[0] is the base system (smallest shift, slowest convergence);
    the other systems ride the same Krylov space via zeta recurrences:
        zeta_{k+1} = zeta_k zeta_{k-1} a_{k-1} /
            ( a_k b_{k-1} (zeta_{k-1} - zeta_k)
              + zeta_{k-1} a_{k-1} (1 + ds a_k) )
        a^s_k = a_k zeta_{k+1}/zeta_k,  b^s_k = b_k (zeta_{k+1}/zeta_k)^2
    with a/b the base alpha/beta and ds = shift_i - shift_0."""
    n = len(shifts)
    assert len(xs) == n and n >= 1
    stats = SolverStats()
    b2 = blas.norm2(b)
    if b2 == 0.0:
        for x in xs:
            x.zero_()
        stats.converged = True
        return stats
    stop = tol * tol * b2

    def new():
        return SpinorField(b.geo, b.precision, b.device, b.n_parity, nspin=b.nspin, ls=b.ls)

    r, Ap, tmp = new(), new(), new()
    ps = [new() for _ in range(n)]
    blas.copy(r, b)
    for i in range(n):
        xs[i].zero_()
        blas.copy(ps[i], b)

    zeta = [1.0] * n       # zeta_k
    zeta_old = [1.0] * n   # zeta_{k-1}
    converged = [False] * n
    r2 = b2
    alpha_prev = 1.0       # base alpha_{k-1} (init per Jegerlehner)
    beta_prev = 0.0        # base beta_{k-1}
    k = 0
    # Shifted p-updates are DEFERRED into the next MdagM's comms window
    # (ref: inv_multi_cg_quda.cpp:115 ShiftUpdate via dslash::aux_worker):
    # they touch only ps[i>=1], which nothing reads until the following
    # x-update, so they legally overlap the halo wait.
    pending = [None]

    def _aux():
        if pending[0] is not None:
            fn, pending[0] = pending[0], None
            fn()

    from ..ops import dispatch as _dsp

    while r2 > stop and k < maxiter:
        _dsp.aux_worker = _aux
        try:
            op.MdagM(Ap, ps[0], tmp)
        finally:
            _dsp.aux_worker = None
        _aux()  # no comms window ran it (local / fused policy)
        if shifts[0] != 0.0:
            blas.axpy(shifts[0], ps[0], Ap)
        pAp = blas.re_dot(ps[0], Ap)
        if pAp <= 0.0:
            break
        alpha0 = r2 / pAp
        zeta_next = [1.0] * n
        alpha_s = [alpha0] * n
        for i in range(1, n):
            if converged[i]:
                continue
            ds = shifts[i] - shifts[0]
            den = (alpha0 * beta_prev * (zeta_old[i] - zeta[i])
                   + zeta_old[i] * alpha_prev * (1.0 + ds * alpha0))
            if den == 0.0:
                converged[i] = True
                continue
            zeta_next[i] = zeta[i] * zeta_old[i] * alpha_prev / den
            alpha_s[i] = alpha0 * zeta_next[i] / zeta[i]
        # x/r updates
        blas.axpy(alpha0, ps[0], xs[0])
        r2_old = r2
        r2 = blas.axpy_norm2(-alpha0, Ap, r)
        for i in range(1, n):
            if not converged[i]:
                blas.axpy(alpha_s[i], ps[i], xs[i])
        beta0 = r2 / r2_old
        # p updates: base now, shifted ones deferred to the next comms
        # window (captured state: this iteration's zeta/beta coefficients)
        blas.xpay(r, beta0, ps[0])
        updates = []
        for i in range(1, n):
            if converged[i]:
                continue
            ratio = zeta_next[i] / zeta[i]
            beta_s = beta0 * ratio * ratio
            updates.append((i, zeta_next[i], beta_s))
            if zeta_next[i] * zeta_next[i] * r2 < stop:
                converged[i] = True
            zeta_old[i], zeta[i] = zeta[i], zeta_next[i]

        def _shift_p_updates(us=updates):
            for i, zn, bs in us:
                blas.caxpby(zn, r, bs, ps[i])

        pending[0] = _shift_p_updates
        alpha_prev, beta_prev = alpha0, beta0
        k += 1

    pending[0] = None  # trailing p-update is never consumed
    stats.iters = k
    stats.resid = sqrt(r2 / b2)
    stats.converged = r2 <= stop
    return stats
