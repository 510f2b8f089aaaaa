"""Minimum-residual extrapolation chronological initial guess
(ref: lib/inv_mre.cpp MinResExt + the chrono_* QudaInvertParam fields):
project the new RHS onto the span of previous solutions, minimizing the
residual, before handing to the Krylov solver."""

from __future__ import annotations

from typing import List

import numpy as np

from ..fields.spinor import SpinorField
from ..ops import blas


class ChronoForecaster:
    """Keeps up to `max_dim` past solutions of an operator; forecast()
    produces the residual-minimizing initial guess for a new source."""

    def __init__(self, max_dim: int = 8):
        self.max_dim = max_dim
        self.basis: List[SpinorField] = []

    def clear(self):
        self.basis.clear()

    def append(self, x: SpinorField):
        keep = SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)
        blas.copy(keep, x)
        self.basis.append(keep)
        if len(self.basis) > self.max_dim:
            self.basis.pop(0)

    def forecast(self, op, x: SpinorField, b: SpinorField, *,
                 use_mdagm: bool = True):
        """x <- argmin_{x in span(basis)} ||b - A x|| (A = MdagM or M)."""
        n = len(self.basis)
        if n == 0:
            x.zero_()
            return x
        Aps = []
        tmp = SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)
        for p in self.basis:
            Ap = SpinorField(x.geo, x.precision, x.device, x.n_parity, nspin=x.nspin, ls=x.ls)
            if use_mdagm:
                op.MdagM(Ap, p, tmp)
            else:
                op.M(Ap, p)
            Aps.append(Ap)
        G = np.empty((n, n), dtype=complex)
        rhs = np.empty(n, dtype=complex)
        for i in range(n):
            for j in range(i, n):
                G[i, j] = blas.c_dot(Aps[i], Aps[j])
                G[j, i] = np.conj(G[i, j])
            rhs[i] = blas.c_dot(Aps[i], b)
        try:
            c = np.linalg.lstsq(G, rhs, rcond=None)[0]
        except np.linalg.LinAlgError:
            x.zero_()
            return x
        x.zero_()
        for i in range(n):
            blas.caxpy(complex(c[i]), self.basis[i], x)
        return x
