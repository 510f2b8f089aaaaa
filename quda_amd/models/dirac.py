"""Dirac operator hierarchy (ref: include/dirac_quda.h:156 + lib/dirac*.cpp
— same class roles, MI355X-first implementation over the fused kernels in
ops.dispatch: every operator application is the minimum number of launches).

Conventions (kappa normalization, matching the reference semantics):
  full:        M psi = psi - kappa * D psi                  (Wilson)
               M psi = A psi - kappa * D psi                (clover)
  even-odd PC (symmetric, QUDA_MATPC_EVEN_EVEN):
               M_pc = 1 - kappa^2 A_ee^-1 D_eo A_oo^-1 D_oe
  source prep: b' = A_ee^-1 (b_e + kappa D_eo A_oo^-1 b_o)
  reconstruct: psi_o = A_oo^-1 (b_o + kappa D_oe psi_e)     (A == 1: Wilson)
"""

from __future__ import annotations

from typing import Optional

from ..fields.clover import CloverField
from ..fields.gauge import GaugeField
from ..fields.spinor import SpinorField
from ..ops import blas
from ..ops.dispatch import CLOV_POST, CLOV_X, PLAIN, apply_clover, dslash_wilson


class Dirac:
    """Base: owns gauge (+clover) and scalar params."""

    def __init__(self, gauge: GaugeField, kappa: float):
        self.gauge = gauge
        self.kappa = float(kappa)
        self.geo = gauge.geo
        self._tmps = {}

    def dslash(self, out, inp, parity, dagger=False, mode=PLAIN, a=1.0,
               x=None, clover=None, clover_inverse=False):
        return dslash_wilson(out, inp, self.gauge, parity, dagger, mode, a, x,
                             clover, clover_inverse)

    def tmp(self, name: str, like: SpinorField, n_parity=1) -> SpinorField:
        key = (name, like.precision, str(like.device), n_parity)
        t = self._tmps.get(key)
        if t is None:
            t = SpinorField(self.geo, like.precision, like.device, n_parity)
            self._tmps[key] = t
        return t

    def new_spinor(self, precision=None, n_parity=1) -> SpinorField:
        return SpinorField(self.geo, precision or self.gauge.precision,
                           self.gauge.device, n_parity)

    def flops_per_site(self) -> int:
        """Wilson dslash flop model (ref: include/dslash.h:467): 1320/site."""
        return 1320


class DiracWilson(Dirac):
    """Full-lattice Wilson M = 1 - kappa D (ref: lib/dirac_wilson.cpp)."""

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        for p in (0, 1):
            self.dslash(out.parity_view(p), inp.parity_view(1 - p), p, dagger,
                        a=-self.kappa, x=inp.parity_view(p))
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out


class DiracWilsonPC(Dirac):
    """Even-odd preconditioned Wilson: M_pc = 1 - kappa^2 D_eo D_oe on the
    even checkerboard (ref: lib/dirac_wilson.cpp DiracWilsonPC)."""

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        t = self.tmp("pc_odd", inp)
        self.dslash(t, inp, 1, dagger)
        self.dslash(out, t, 0, dagger, a=-self.kappa ** 2, x=inp)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def prepare(self, b_full: SpinorField) -> SpinorField:
        """b' = b_e + kappa D_eo b_o."""
        be = self.new_spinor(b_full.precision)
        self.dslash(be, b_full.parity_view(1), 0, a=self.kappa,
                    x=b_full.parity_view(0))
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        """x_o = b_o + kappa D_oe x_e."""
        blas.copy(x_full.parity_view(0), x_e)
        self.dslash(x_full.parity_view(1), x_e, 1, a=self.kappa,
                    x=b_full.parity_view(1))
        return x_full


class _CloverMixin:
    clover: CloverField

    def apply_A(self, out, inp, parity, inverse=False):
        return apply_clover(out, inp, self.clover, parity, inverse)


class DiracClover(Dirac, _CloverMixin):
    """Full-lattice Wilson-clover M = A - kappa D
    (ref: lib/dirac_clover.cpp): one fused CLOV_X launch per parity."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float):
        super().__init__(gauge, kappa)
        self.clover = clover

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        for p in (0, 1):
            self.dslash(out.parity_view(p), inp.parity_view(1 - p), p, dagger,
                        mode=CLOV_X, a=-self.kappa, x=inp.parity_view(p),
                        clover=self.clover)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def flops_per_site(self) -> int:
        return 1320 + 504


class DiracCloverPC(Dirac, _CloverMixin):
    """Symmetric even-odd preconditioned clover
    (ref: lib/dirac_clover.cpp DiracCloverPC, QUDA_MATPC_EVEN_EVEN):
    M_pc = 1 - kappa^2 A_ee^-1 D_eo A_oo^-1 D_oe  — two fused launches."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float):
        super().__init__(gauge, kappa)
        self.clover = clover

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        k2 = -self.kappa ** 2
        if not dagger:
            t = self.tmp("pc_odd", inp)
            self.dslash(t, inp, 1, mode=CLOV_POST, clover=self.clover,
                        clover_inverse=True)
            self.dslash(out, t, 0, mode=CLOV_POST, a=k2, x=inp,
                        clover=self.clover, clover_inverse=True)
        else:
            # M^dag = 1 - k^2 D_oe^dag A_oo^-1 D_eo^dag A_ee^-1
            t0 = self.tmp("pc_even", inp)
            t1 = self.tmp("pc_odd", inp)
            self.apply_A(t0, inp, 0, inverse=True)
            self.dslash(t1, t0, 1, dagger=True, mode=CLOV_POST,
                        clover=self.clover, clover_inverse=True)
            self.dslash(out, t1, 0, dagger=True, a=k2, x=inp)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def prepare(self, b_full: SpinorField) -> SpinorField:
        """b' = A_ee^-1 (b_e + kappa D_eo A_oo^-1 b_o)."""
        t = self.new_spinor(b_full.precision)
        be = self.new_spinor(b_full.precision)
        self.apply_A(t, b_full.parity_view(1), 1, inverse=True)
        self.dslash(be, t, 0, a=self.kappa, x=b_full.parity_view(0))
        self.apply_A(t, be, 0, inverse=True)
        blas.copy(be, t)
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        """x_o = A_oo^-1 (b_o + kappa D_oe x_e)."""
        blas.copy(x_full.parity_view(0), x_e)
        t = self.tmp("pc_odd", x_full)
        self.dslash(t, x_e, 1, a=self.kappa, x=b_full.parity_view(1))
        self.apply_A(x_full.parity_view(1), t, 1, inverse=True)
        return x_full

    def flops_per_site(self) -> int:
        return 1320 + 504
