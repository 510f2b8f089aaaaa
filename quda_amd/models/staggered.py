"""Staggered Dirac operators (ref: lib/dirac_staggered.cpp — naive
Kogut-Susskind; mass normalization M = 2m + D with D antihermitian, so
MdagM = 4m^2 - D^2 and the even-odd system is hermitian positive
definite)."""

from __future__ import annotations

from ..fields.gauge import GaugeField
from ..fields.spinor import SpinorField
from ..ops import blas
from ..ops.dispatch import dslash_staggered
from .dirac import Dirac


class DiracStaggered(Dirac):
    """Full-lattice staggered M = 2m + D."""

    def __init__(self, gauge: GaugeField, mass: float):
        super().__init__(gauge, kappa=0.0)
        self.mass = float(mass)

    def new_spinor(self, precision=None, n_parity=1) -> SpinorField:
        return SpinorField(self.geo, precision or self.gauge.precision,
                           self.gauge.device, n_parity, nspin=1)

    def dslash(self, out, inp, parity, dagger=False, **kw):
        """Parity-hopping KS term (the dslashQuda entry; D^dag = -D)."""
        return dslash_staggered(out, inp, self.gauge, parity,
                                b=-1.0 if dagger else 1.0)

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        b = -1.0 if dagger else 1.0
        for p in (0, 1):
            dslash_staggered(out.parity_view(p), inp.parity_view(1 - p), self.gauge,
                             p, a=2.0 * self.mass, b=b, x=inp.parity_view(p))
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def flops_per_site(self) -> int:
        """ref include/dslash.h staggered flop model: 570/site + mass."""
        return 570 + 12


class DiracStaggeredPC(Dirac):
    """Even-odd preconditioned staggered: the hermitian PD operator
    M_pc = 4m^2 - D_eo D_oe on even sites (ref: lib/dirac_staggered.cpp
    DiracStaggeredPC; D_ee = 0 for staggered)."""

    def __init__(self, gauge: GaugeField, mass: float):
        super().__init__(gauge, kappa=0.0)
        self.mass = float(mass)

    def new_spinor(self, precision=None, n_parity=1) -> SpinorField:
        return SpinorField(self.geo, precision or self.gauge.precision,
                           self.gauge.device, n_parity, nspin=1)

    def tmp(self, name, like, n_parity=1):
        key = (name, like.precision, str(like.device), n_parity, 1)
        t = self._tmps.get(key)
        if t is None:
            t = SpinorField(self.geo, like.precision, like.device, n_parity,
                            nspin=1)
            self._tmps[key] = t
        return t

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        """M_pc is hermitian: dagger is a no-op."""
        t = self.tmp("pc_odd", inp)
        dslash_staggered(t, inp, self.gauge, 1)            # t = D_oe in
        dslash_staggered(out, t, self.gauge, 0, a=4.0 * self.mass ** 2,
                         b=-1.0, x=inp)                    # 4m^2 in - D_eo t
        return out

    def MdagM(self, out, inp, tmp):
        # M_pc itself is hermitian PD: CG applies it once per iteration
        # (ref: staggered solves use QUDA_SOLVE_DIRECT_PC the same way)
        self.M(out, inp)
        return out

    def prepare(self, b_full: SpinorField) -> SpinorField:
        """From M x = b: (4m^2 - D_eo D_oe) x_e = 2m b_e - D_eo b_o."""
        be = self.new_spinor(b_full.precision)
        dslash_staggered(be, b_full.parity_view(1), self.gauge, 0,
                         a=2.0 * self.mass, b=-1.0, x=b_full.parity_view(0))
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        """x_o = (b_o - D_oe x_e) / (2m)."""
        blas.copy(x_full.parity_view(0), x_e)
        xo = x_full.parity_view(1)
        dslash_staggered(xo, x_e, self.gauge, 1,
                         a=1.0 / (2.0 * self.mass), b=-1.0 / (2.0 * self.mass),
                         x=b_full.parity_view(1))
        return x_full

    def flops_per_site(self) -> int:
        return 570 + 12


class DiracImprovedStaggered(DiracStaggered):
    """Asqtad/HISQ-style improved staggered: fat 1-hop + Naik 3-hop links
    (ref: lib/dirac_improved_staggered.cpp). `fat`/`lng` are GaugeFields
    (lng built with shift=3)."""

    def __init__(self, fat: GaugeField, lng: GaugeField, mass: float):
        super().__init__(fat, mass)
        self.lng = lng

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        b = -1.0 if dagger else 1.0
        for p in (0, 1):
            dslash_staggered(out.parity_view(p), inp.parity_view(1 - p),
                             self.gauge, p, a=2.0 * self.mass, b=b,
                             x=inp.parity_view(p), long_gauge=self.lng)
        return out

    def flops_per_site(self) -> int:
        return 1146 + 12  # ref dslash.h improved staggered flop model


class DiracImprovedStaggeredPC(DiracStaggeredPC):
    """Even-odd PC improved staggered: 4m^2 - D_eo D_oe with the fat+long
    D (ref: lib/dirac_improved_staggered.cpp)."""

    def __init__(self, fat: GaugeField, lng: GaugeField, mass: float):
        super().__init__(fat, mass)
        self.lng = lng

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        t = self.tmp("pc_odd", inp)
        dslash_staggered(t, inp, self.gauge, 1, long_gauge=self.lng)
        dslash_staggered(out, t, self.gauge, 0, a=4.0 * self.mass ** 2,
                         b=-1.0, x=inp, long_gauge=self.lng)
        return out

    def prepare(self, b_full: SpinorField) -> SpinorField:
        be = self.new_spinor(b_full.precision)
        dslash_staggered(be, b_full.parity_view(1), self.gauge, 0,
                         a=2.0 * self.mass, b=-1.0, x=b_full.parity_view(0),
                         long_gauge=self.lng)
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        blas.copy(x_full.parity_view(0), x_e)
        xo = x_full.parity_view(1)
        dslash_staggered(xo, x_e, self.gauge, 1,
                         a=1.0 / (2.0 * self.mass), b=-1.0 / (2.0 * self.mass),
                         x=b_full.parity_view(1), long_gauge=self.lng)
        return x_full
