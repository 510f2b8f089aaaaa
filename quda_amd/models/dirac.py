"""Dirac operator hierarchy (ref: include/dirac_quda.h:156 + lib/dirac*.cpp
— same class roles, MI355X-first implementation over ops.dispatch).

Conventions (kappa normalization, matching the reference semantics):
  full:        M psi = psi - kappa * D psi                  (Wilson)
               M psi = A psi - kappa * D psi                (clover)
  even-odd PC (symmetric, QUDA_MATPC_EVEN_EVEN):
               M_pc chi_e = chi_e - kappa^2 A_ee^-1 D_eo A_oo^-1 D_oe chi_e
  (A == 1 for plain Wilson).

prepare()/reconstruct() map a full-lattice source/solution to/from the
preconditioned system (ref: dirac_quda.h:358).
"""

from __future__ import annotations

from typing import Optional

import torch

from ..fields.clover import CloverField
from ..fields.gauge import GaugeField
from ..fields.spinor import SpinorField
from ..ops import blas, dispatch


class Dirac:
    """Base: owns gauge (+clover) and scalar params; applies to parity or
    full SpinorFields. Subclasses define M / MdagM on their solve space."""

    def __init__(self, gauge: GaugeField, kappa: float):
        self.gauge = gauge
        self.kappa = float(kappa)
        self.geo = gauge.geo

    # single-parity hop: out(par) = D in(1-par)
    def dslash(self, out: SpinorField, inp: SpinorField, parity: int,
               dagger: bool = False, xpay: Optional[tuple] = None):
        return dispatch.dslash_wilson(out, inp, self.gauge, parity, dagger, xpay)

    def new_spinor(self, precision=None, n_parity=1) -> SpinorField:
        return SpinorField(self.geo, precision or self.gauge.precision,
                           self.gauge.device, n_parity)

    def flops_per_site(self) -> int:
        """Wilson dslash flop count (ref model: include/dslash.h:467 —
        1320 flops/site for Nc=3, Ns=4)."""
        return 1320


class DiracWilson(Dirac):
    """Full-lattice Wilson M = 1 - kappa D (ref: lib/dirac_wilson.cpp)."""

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        # out_e = in_e - kappa D_eo in_o ; out_o = in_o - kappa D_oe in_e
        for p in (0, 1):
            self.dslash(out.parity_view(p), inp.parity_view(1 - p), p, dagger,
                        xpay=(-1.0 / self.kappa, inp.parity_view(p)))
        blas.scal(-self.kappa, out)
        return out

    def MdagM(self, out: SpinorField, inp: SpinorField, tmp: SpinorField):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out


class DiracWilsonPC(Dirac):
    """Even-odd preconditioned Wilson: M_pc = 1 - kappa^2 D_eo D_oe acting on
    the even checkerboard (ref: lib/dirac_wilson.cpp DiracWilsonPC)."""

    def __init__(self, gauge: GaugeField, kappa: float):
        super().__init__(gauge, kappa)
        self._tmp_o: Optional[SpinorField] = None

    def _tmp(self, like: SpinorField) -> SpinorField:
        if (self._tmp_o is None or self._tmp_o.precision != like.precision
                or self._tmp_o.device != like.device):
            self._tmp_o = SpinorField(self.geo, like.precision, like.device, 1)
        return self._tmp_o

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        t = self._tmp(inp)
        # t_o = D_oe in_e ; out_e = in_e - kappa^2 D_eo t_o
        self.dslash(t, inp, 1, dagger)
        self.dslash(out, t, 0, dagger, xpay=(-1.0 / self.kappa ** 2, inp))
        blas.scal(-self.kappa ** 2, out)
        return out

    def MdagM(self, out: SpinorField, inp: SpinorField, tmp: SpinorField):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    # -- source prep / solution reconstruction (ref dirac_quda.h:358) ------
    def prepare(self, b_full: SpinorField) -> SpinorField:
        """b_e' = b_e + kappa D_eo b_o (kappa-normalized even source)."""
        be = SpinorField(self.geo, b_full.precision, b_full.device, 1)
        self.dslash(be, b_full.parity_view(1), 0,
                    xpay=(1.0 / self.kappa, b_full.parity_view(0)))
        blas.scal(self.kappa, be)
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        """x_o = kappa (b_o_ + D_oe x_e) -> writes both parities of x_full."""
        x_full.parity_view(0).copy_(x_e)
        xo = x_full.parity_view(1)
        self.dslash(xo, x_e, 1, xpay=(1.0 / self.kappa, b_full.parity_view(1)))
        blas.scal(self.kappa, xo)
        return x_full


class _CloverMixin:
    clover: CloverField

    def apply_A(self, out, inp, parity, inverse=False):
        return dispatch.apply_clover(out, inp, self.clover, parity, inverse)


class DiracClover(Dirac, _CloverMixin):
    """Full-lattice Wilson-clover M = A - kappa D
    (ref: lib/dirac_clover.cpp)."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float):
        super().__init__(gauge, kappa)
        self.clover = clover

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        t = SpinorField(self.geo, inp.precision, inp.device, 1)
        for p in (0, 1):
            self.apply_A(t, inp.parity_view(p), p)
            self.dslash(out.parity_view(p), inp.parity_view(1 - p), p, dagger,
                        xpay=(-1.0 / self.kappa, t))
        blas.scal(-self.kappa, out)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def flops_per_site(self) -> int:
        return 1320 + 504  # dslash + clover (ref dslash.h flop model)


class DiracCloverPC(Dirac, _CloverMixin):
    """Symmetric even-odd preconditioned clover:
    M_pc = 1 - kappa^2 A_ee^-1 D_eo A_oo^-1 D_oe
    (ref: lib/dirac_clover.cpp DiracCloverPC, QUDA_MATPC_EVEN_EVEN)."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float):
        super().__init__(gauge, kappa)
        self.clover = clover
        self._t1: Optional[SpinorField] = None
        self._t2: Optional[SpinorField] = None

    def _tmps(self, like: SpinorField):
        if (self._t1 is None or self._t1.precision != like.precision
                or self._t1.device != like.device):
            self._t1 = SpinorField(self.geo, like.precision, like.device, 1)
            self._t2 = SpinorField(self.geo, like.precision, like.device, 1)
        return self._t1, self._t2

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        t1, t2 = self._tmps(inp)
        if not dagger:
            self.dslash(t1, inp, 1)            # t1_o = D_oe in_e
            self.apply_A(t2, t1, 1, inverse=True)   # t2 = A_oo^-1 t1
            self.dslash(t1, t2, 0)             # t1_e = D_eo t2
            self.apply_A(t2, t1, 0, inverse=True)   # t2 = A_ee^-1 t1
        else:
            # (M_pc)^dag = 1 - kappa^2 D_oe^dag A_oo^-1 D_eo^dag A_ee^-1
            self.apply_A(t2, inp, 0, inverse=True)
            self.dslash(t1, t2, 1, dagger=True)
            self.apply_A(t2, t1, 1, inverse=True)
            self.dslash(t1, t2, 0, dagger=True)
            t2, t1 = t1, t2
        # out = in - kappa^2 t2
        blas.copy(out, inp)
        blas.axpy(-self.kappa ** 2, t2, out)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def prepare(self, b_full: SpinorField) -> SpinorField:
        """b_e' = kappa A_ee^-1 (b_e + kappa D_eo A_oo^-1 b_o)."""
        t = SpinorField(self.geo, b_full.precision, b_full.device, 1)
        be = SpinorField(self.geo, b_full.precision, b_full.device, 1)
        self.apply_A(t, b_full.parity_view(1), 1, inverse=True)
        self.dslash(be, t, 0, xpay=(1.0 / self.kappa, b_full.parity_view(0)))
        self.apply_A(t, be, 0, inverse=True)
        blas.copy(be, t)
        blas.scal(self.kappa, be)
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        """x_o = kappa A_oo^-1 (b_o + D_oe x_e)."""
        x_full.parity_view(0).copy_(x_e)
        t = SpinorField(self.geo, x_full.precision, x_full.device, 1)
        self.dslash(t, x_e, 1, xpay=(1.0 / self.kappa, b_full.parity_view(1)))
        xo = x_full.parity_view(1)
        self.apply_A(xo, t, 1, inverse=True)
        blas.scal(self.kappa, xo)
        return x_full

    def flops_per_site(self) -> int:
        return 1320 + 504
