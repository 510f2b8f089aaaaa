"""MSPCG-style domain-decomposed preconditioning (ref: the inner-Schwarz
PCG the reference pairs with its fused Moebius kernels, arXiv:2104.05615,
lib/inv_pcg_quda.cpp — re-designed for the xGMI topology: the
preconditioner is a FIXED small number of CG iterations on the
communication-free local operator (comms disabled via the communicator
stack; the pre-exchanged boundary links stay in the stencil layout, so
the local operator is the physical one with frozen neighbor boundaries).
A fixed inner iteration count keeps the preconditioner linear and SPD,
so plain PCG applies."""

from __future__ import annotations

from ..fields.spinor import SpinorField
from ..ops import blas
from .cg import SolverStats, cg_solve
from .variants import pcg_solve


def schwarz_precond(op, *, inner_iters: int = 6):
    """K r = (local MdagM)^{-1}-ish r via `inner_iters` CG iterations with
    collectives and halo exchange disabled (each rank solves its own
    domain)."""
    from ..parallel import comms

    def precond(z: SpinorField, r: SpinorField):
        with comms.solo_mode():
            z.zero_()
            cg_solve(op, z, r, tol=1e-30, maxiter=inner_iters)

    return precond


def mspcg_solve(op, x: SpinorField, b: SpinorField, *,
                inner_iters: int = 6, tol: float = 1e-8,
                maxiter: int = 1000) -> SolverStats:
    """PCG on MdagM with the Schwarz local-solve preconditioner. Trades
    outer iterations (each with halo exchanges + global reductions) for
    communication-free local work — the win grows with the
    communication/compute ratio of the machine."""
    return pcg_solve(op, x, b, precond=schwarz_precond(
        op, inner_iters=inner_iters), tol=tol, maxiter=maxiter)


def dd_gcr_solve(op, x: SpinorField, b: SpinorField, *,
                 inner_iters: int = 6, tol: float = 1e-8,
                 maxiter: int = 500, nkrylov: int = 10) -> SolverStats:
    """Domain-decomposed GCR: flexible GCR preconditioned by the same
    communication-free local solve (ref: the DD preconditioner option of
    lib/inv_gcr_quda.cpp — the >100-GPU strong-scaling configuration of
    the reference, arXiv:1109.2935). Nonsymmetric-friendly flavor of
    mspcg_solve."""
    from .gcr import gcr_solve

    def precond(z: SpinorField, r: SpinorField):
        from ..parallel import comms
        from .variants import cgnr_solve
        with comms.solo_mode():
            z.zero_()
            cgnr_solve(op, z, r, tol=1e-30, maxiter=inner_iters)

    return gcr_solve(op, x, b, tol=tol, maxiter=maxiter, nkrylov=nkrylov,
                     precond=precond)
