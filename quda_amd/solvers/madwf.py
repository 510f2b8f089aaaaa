"""MADWF: Moebius-accelerated domain-wall solver with a trainable
Ls -> Ls' transfer (ref: include/madwf_ml.h, lib/madwf_ml.cpp,
kernels/madwf_transfer.cuh — re-designed: the transfer is a pair of
per-chirality [Ls', Ls] complex matrices applied as an s-space
contraction; the preconditioner is z = T^dag M_small^-1 T r plus an
identity term on the complement (the PC operator is near-identity), used
inside flexible GCR; training minimizes the MADWF-ML chi^2
||M_big z(r) - r||^2 by autograd, with the inner solve differentiated
through its dagger solve).
"""

from __future__ import annotations

from typing import Optional

import torch

from ..fields.spinor import SpinorField
from ..ops import blas
from .gcr import gcr_solve
from .variants import cgnr_solve


class TransferLs:
    """Chirality-split s-space transfer: small(s') = sum_s T_pm(s',s)
    P_pm big(s) with P+ = upper DeGrand-Rossi spins (0,1), P- = lower."""

    def __init__(self, Ls: int, Lsp: int, dtype=torch.complex128,
                 device="cpu"):
        self.Ls, self.Lsp = Ls, Lsp
        # boundary-preserving truncation: keep the first ceil(Lsp/2) and
        # last floor(Lsp/2) slices (both domain walls survive)
        k = (Lsp + 1) // 2
        T = torch.zeros((Lsp, Ls), dtype=dtype, device=device)
        for i in range(k):
            T[i, i] = 1.0
        for i in range(Lsp - k):
            T[Lsp - 1 - i, Ls - 1 - i] = 1.0
        self.Tp = T.clone()
        self.Tm = T.clone()

    def _views(self, psi: SpinorField, ls: int):
        v = psi.to_complex()[0]
        V = v.shape[0] // ls
        return v.reshape(ls, V, 4, 3)

    def apply(self, out: SpinorField, psi: SpinorField) -> SpinorField:
        v = self._views(psi, self.Ls)
        dt = v.dtype
        up = torch.einsum("ts,svxc->tvxc", self.Tp.to(dt), v[:, :, 0:2, :])
        lo = torch.einsum("ts,svxc->tvxc", self.Tm.to(dt), v[:, :, 2:4, :])
        w = torch.cat([up, lo], dim=2)
        out.from_complex(w.reshape(-1, 4, 3).unsqueeze(0))
        return out

    def apply_dag(self, out: SpinorField, psi: SpinorField) -> SpinorField:
        v = self._views(psi, self.Lsp)
        dt = v.dtype
        up = torch.einsum("ts,tvxc->svxc", self.Tp.conj().to(dt),
                          v[:, :, 0:2, :])
        lo = torch.einsum("ts,tvxc->svxc", self.Tm.conj().to(dt),
                          v[:, :, 2:4, :])
        w = torch.cat([up, lo], dim=2)
        out.from_complex(w.reshape(-1, 4, 3).unsqueeze(0))
        return out


def madwf_precond(op_big, op_small, T: TransferLs, *, inner_tol: float = 1e-3,
                  inner_maxiter: int = 200):
    """Returns precond(z, r) for gcr_solve: z = T^dag M_s^-1 T r +
    (r - T^dag T r). The complement term keeps the preconditioner
    full-rank (the dropped middle slices pass through unchanged — the
    even-odd PC operator is 1 + O(Dhat^2) so identity is the right
    zeroth-order inverse there)."""

    def precond(z: SpinorField, r: SpinorField):
        rs = op_small.new_spinor(r.precision)
        T.apply(rs, r)
        xs = op_small.new_spinor(r.precision)
        cgnr_solve(op_small, xs, rs, tol=inner_tol, maxiter=inner_maxiter)
        T.apply_dag(z, xs)
        # + (r - T^dag T r)
        tt = op_big.new_spinor(r.precision)
        T.apply(rs, r)
        T.apply_dag(tt, rs)
        blas.axpy(1.0, r, z)
        blas.axpy(-1.0, tt, z)

    return precond


def madwf_solve(op_big, op_small, x: SpinorField, b: SpinorField, *,
                T: Optional[TransferLs] = None, tol: float = 1e-8,
                maxiter: int = 500, nkrylov: int = 10,
                inner_tol: float = 1e-3, inner_maxiter: int = 200):
    """Flexible-GCR MADWF solve of op_big.M x = b (even-odd PC fields)."""
    if T is None:
        T = TransferLs(op_big.Ls, op_small.Ls,
                       device=b.data.device)
    pre = madwf_precond(op_big, op_small, T, inner_tol=inner_tol,
                        inner_maxiter=inner_maxiter)
    return gcr_solve(op_big, x, b, tol=tol, maxiter=maxiter,
                     nkrylov=nkrylov, precond=pre)


# ---------------------------------------------------------------------------
# MADWF-ML transfer training (autograd; the inner solve differentiates
# through the dagger solve: d/dv [A^-1 v] pulls back as A^-dag grad)
# ---------------------------------------------------------------------------

class _SolveFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, v, op, tol, maxiter):
        ctx.op, ctx.tol, ctx.maxiter = op, tol, maxiter
        y = _solve_complex(op, v.detach(), False, tol, maxiter)
        return y

    @staticmethod
    def backward(ctx, grad):
        # y = M^-1 v is linear-holomorphic: pull grad back through M^-dag
        g = _solve_complex(ctx.op, grad, True, ctx.tol, ctx.maxiter)
        return g, None, None, None


class _ApplyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, v, op):
        ctx.op = op
        return _apply_complex(op, v.detach(), False)

    @staticmethod
    def backward(ctx, grad):
        return _apply_complex(ctx.op, grad, True), None


def _field_of(op, v):
    f = op.new_spinor("double")
    f.from_complex(v.unsqueeze(0))
    return f


def _apply_complex(op, v, dagger):
    f = _field_of(op, v)
    o = op.new_spinor("double")
    op.M(o, f, dagger=dagger)
    return o.to_complex()[0]


def _solve_complex(op, v, dagger, tol, maxiter):
    b = _field_of(op, v)
    x = op.new_spinor("double")
    if dagger:
        # solve M^dag y = v: M M^dag y = M v -> via normal equations on
        # the flipped order; use CGNR on the dagger operator
        class _Dag:
            Ls = op.Ls

            def new_spinor(self, precision="double", n_parity=1):
                return op.new_spinor(precision)

            def M(self, out, inp, dagger=False):
                return op.M(out, inp, dagger=not dagger)

        cgnr_solve(_Dag(), x, b, tol=tol, maxiter=maxiter)
    else:
        cgnr_solve(op, x, b, tol=tol, maxiter=maxiter)
    return x.to_complex()[0]


def train_transfer(op_big, op_small, T: TransferLs, *, n_samples: int = 4,
                   iters: int = 30, lr: float = 0.02, seed: int = 7,
                   inner_tol: float = 1e-6, inner_maxiter: int = 500,
                   verbose: bool = False):
    """Minimize chi^2 = E_r ||M_big T^dag M_s^-1 T r - r||^2 / ||r||^2
    over the transfer matrices (MADWF-ML, madwf_ml.cpp:train). Returns
    the chi^2 history; T is updated in place."""
    Ls, Lsp = T.Ls, T.Lsp
    samples = []
    for i in range(n_samples):
        r = op_big.new_spinor("double")
        r.gaussian_(seed=seed + 100 + i)
        rv = r.to_complex()[0]
        samples.append(rv / rv.norm())
    Tp = T.Tp.clone().requires_grad_(True)
    Tm = T.Tm.clone().requires_grad_(True)
    opt = torch.optim.Adam([Tp, Tm], lr=lr)
    hist = []
    V4 = samples[0].shape[0] // Ls
    for it in range(iters):
        opt.zero_grad()
        chi2 = torch.zeros((), dtype=torch.float64)
        for rv in samples:
            v = rv.reshape(Ls, V4, 4, 3)
            up = torch.einsum("ts,svxc->tvxc", Tp, v[:, :, 0:2, :])
            lo = torch.einsum("ts,svxc->tvxc", Tm, v[:, :, 2:4, :])
            small = torch.cat([up, lo], dim=2).reshape(-1, 4, 3)
            y = _SolveFn.apply(small, op_small, inner_tol, inner_maxiter)
            y = y.reshape(Lsp, V4, 4, 3)
            bu = torch.einsum("ts,tvxc->svxc", Tp.conj(), y[:, :, 0:2, :])
            bl = torch.einsum("ts,tvxc->svxc", Tm.conj(), y[:, :, 2:4, :])
            z = torch.cat([bu, bl], dim=2).reshape(-1, 4, 3)
            Mz = _ApplyFn.apply(z, op_big)
            chi2 = chi2 + (Mz - rv).abs().square().sum()
        chi2 = chi2 / n_samples
        chi2.backward()
        opt.step()
        hist.append(float(chi2.detach()))
        if verbose:
            print(f"madwf train it={it} chi2={float(chi2):.4e}")
    with torch.no_grad():
        T.Tp.copy_(Tp)
        T.Tm.copy_(Tm)
    return hist
