"""Block CG for multiple right-hand sides sharing one Krylov space
(ref: lib/inv_msrc_cg_quda.cpp / the block-Krylov program, README.md:270
arXiv:1710.09745 — O'Leary block CG with small dense block coefficients
solved on the host; the NxN reductions and block updates run as single
complex GEMMs via ops.multi_blas)."""

from __future__ import annotations

from math import sqrt
from typing import List

import numpy as np
import torch

from ..fields.spinor import SpinorField
from ..ops import blas
from ..ops.multi_blas import block_caxpy, block_cdot
from .cg import SolverStats


def block_cg_solve(op, xs: List[SpinorField], bs: List[SpinorField], *,
                   tol: float = 1e-8, maxiter: int = 1000) -> SolverStats:
    """Solve MdagM x_i = b_i for all i in one block-Krylov iteration."""
    N = len(bs)
    stats = SolverStats()
    b2 = [blas.norm2(b) for b in bs]
    stop = [tol * tol * x for x in b2]

    def new():
        b = bs[0]
        return SpinorField(b.geo, b.precision, b.device, b.n_parity,
                           nspin=b.nspin, ls=b.ls)

    R = [new() for _ in range(N)]
    P = [new() for _ in range(N)]
    AP = [new() for _ in range(N)]
    tmp = new()
    for i in range(N):
        xs[i].zero_()
        blas.copy(R[i], bs[i])
        blas.copy(P[i], bs[i])
    rho = block_cdot(R, R)          # [N,N]
    k = 0
    batched = hasattr(op, "MdagM_batch")
    while k < maxiter:
        if batched:
            # merged-halo multi-RHS path: one message per face for the
            # whole block (dispatch.dslash_wilson_batch underneath)
            op.MdagM_batch(AP, P)
        else:
            for i in range(N):
                op.MdagM(AP[i], P[i], tmp)
        gamma = block_cdot(P, AP).cpu().numpy()
        try:
            alpha = np.linalg.solve(gamma, rho.cpu().numpy())
        except np.linalg.LinAlgError:
            break
        block_caxpy(alpha.T, P, xs)             # X += P alpha
        block_caxpy(-alpha.T, AP, R)            # R -= AP alpha
        k += 1
        rho_new = block_cdot(R, R)
        diag = rho_new.diagonal().real.cpu().numpy()
        if all(diag[i] <= stop[i] for i in range(N)):
            rho = rho_new
            break
        try:
            beta = np.linalg.solve(rho.cpu().numpy(), rho_new.cpu().numpy())
        except np.linalg.LinAlgError:
            break
        # P = R + P beta
        Pold = [new() for _ in range(N)]
        for i in range(N):
            blas.copy(Pold[i], P[i])
            blas.copy(P[i], R[i])
        block_caxpy(beta.T, Pold, P)
        rho = rho_new
    stats.iters = k
    diag = rho.diagonal().real.cpu().numpy()
    stats.resid = sqrt(max(diag[i] / b2[i] for i in range(N)))
    stats.converged = all(diag[i] <= stop[i] for i in range(N))
    return stats
