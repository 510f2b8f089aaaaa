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

import torch

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
               x=None, clover=None, clover_inverse=False, twist=(0.0, 0.0)):
        return dslash_wilson(out, inp, self.gauge, parity, dagger, mode, a, x,
                             clover, clover_inverse, twist)

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


class DiracTwistedMass(Dirac):
    """Degenerate twisted-mass Wilson, twisted basis
    (ref: lib/dirac_twisted_mass.cpp): M = T(1, 2 kappa mu) - kappa D with
    T(b) = b_re + i b_im g5. One fused TWIST_X launch per parity."""

    def __init__(self, gauge: GaugeField, kappa: float, mu: float):
        super().__init__(gauge, kappa)
        self.mu = float(mu)

    @property
    def eps(self) -> float:
        return 2.0 * self.kappa * self.mu

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        sgn = -1.0 if dagger else 1.0
        from ..ops.dispatch import TWIST_X
        for p in (0, 1):
            self.dslash(out.parity_view(p), inp.parity_view(1 - p), p, dagger,
                        mode=TWIST_X, a=-self.kappa, x=inp.parity_view(p),
                        twist=(1.0, sgn * self.eps))
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def flops_per_site(self) -> int:
        return 1320 + 48


class DiracTwistedMassPC(Dirac):
    """Symmetric even-odd preconditioned twisted mass
    (ref: lib/dirac_twisted_mass.cpp DiracTwistedMassPC):
    M_pc = 1 - kappa^2 T^-1 D_eo T^-1 D_oe, T^-1 = T(c, -eps c),
    c = 1/(1+eps^2). Two fused TWIST_POST launches."""

    def __init__(self, gauge: GaugeField, kappa: float, mu: float):
        super().__init__(gauge, kappa)
        self.mu = float(mu)

    @property
    def eps(self) -> float:
        return 2.0 * self.kappa * self.mu

    def _tinv(self, dagger: bool = False):
        c = 1.0 / (1.0 + self.eps ** 2)
        s = 1.0 if dagger else -1.0
        return (c, s * self.eps * c)

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        from ..ops.dispatch import TWIST_POST, apply_twist_field
        k2 = -self.kappa ** 2
        if not dagger:
            t = self.tmp("pc_odd", inp)
            self.dslash(t, inp, 1, mode=TWIST_POST, twist=self._tinv())
            self.dslash(out, t, 0, mode=TWIST_POST, a=k2, x=inp,
                        twist=self._tinv())
        else:
            # M^dag = 1 - k^2 D^dag T^-dag D^dag T^-dag
            t0 = self.tmp("pc_even", inp)
            t1 = self.tmp("pc_odd", inp)
            apply_twist_field(t0, inp, *self._tinv(dagger=True))
            self.dslash(t1, t0, 1, dagger=True, mode=TWIST_POST,
                        twist=self._tinv(dagger=True))
            self.dslash(out, t1, 0, dagger=True, a=k2, x=inp)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def prepare(self, b_full: SpinorField) -> SpinorField:
        """b' = T_ee^-1 (b_e + kappa D_eo T_oo^-1 b_o)."""
        from ..ops.dispatch import apply_twist_field
        t = self.new_spinor(b_full.precision)
        be = self.new_spinor(b_full.precision)
        apply_twist_field(t, b_full.parity_view(1), *self._tinv())
        self.dslash(be, t, 0, a=self.kappa, x=b_full.parity_view(0))
        apply_twist_field(t, be, *self._tinv())
        from ..ops import blas
        blas.copy(be, t)
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        """x_o = T_oo^-1 (b_o + kappa D_oe x_e)."""
        from ..ops import blas
        from ..ops.dispatch import apply_twist_field
        blas.copy(x_full.parity_view(0), x_e)
        t = self.tmp("pc_odd", x_full)
        self.dslash(t, x_e, 1, a=self.kappa, x=b_full.parity_view(1))
        apply_twist_field(x_full.parity_view(1), t, *self._tinv())
        return x_full

    def flops_per_site(self) -> int:
        return 1320 + 48


class DiracTwistedClover(Dirac, _CloverMixin):
    """Full-lattice twisted clover (ref: lib/dirac_twisted_clover.cpp):
    M = (A + i 2 kappa mu g5) - kappa D — one fused CLOVTW_X launch per
    parity."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float,
                 mu: float):
        super().__init__(gauge, kappa)
        self.clover = clover
        self.mu = float(mu)

    @property
    def eps(self) -> float:
        return 2.0 * self.kappa * self.mu

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        from ..ops.dispatch import CLOVTW_X
        sgn = -1.0 if dagger else 1.0
        for p in (0, 1):
            self.dslash(out.parity_view(p), inp.parity_view(1 - p), p, dagger,
                        mode=CLOVTW_X, a=-self.kappa, x=inp.parity_view(p),
                        clover=self.clover, twist=(1.0, sgn * self.eps))
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def flops_per_site(self) -> int:
        return 1320 + 504 + 48


class DiracNdegTwistedMass(Dirac):
    """Non-degenerate twisted-mass doublet (ref: lib/dirac_twisted_mass.cpp
    two-flavor branch + kernels/dslash_ndeg_twisted_mass.cuh):
    M = (1 + i 2 kappa mu g5 tau3 - 2 kappa eps tau1) - kappa D on a
    flavor doublet, represented as an ls=2 field (slice = flavor; tau1 is
    the Ls=2 s-hop with mf=-1, g5 tau3 the per-flavor-signed twist)."""

    def __init__(self, gauge: GaugeField, kappa: float, mu: float,
                 epsilon: float):
        super().__init__(gauge, kappa)
        self.mu = float(mu)
        self.epsilon = float(epsilon)

    @property
    def a_t(self) -> float:
        return 2.0 * self.kappa * self.mu

    @property
    def b_t(self) -> float:
        return -2.0 * self.kappa * self.epsilon

    def new_spinor(self, precision=None, n_parity=1) -> SpinorField:
        return SpinorField(self.geo, precision or self.gauge.precision,
                           self.gauge.device, n_parity, ls=2)

    def _apply_A(self, out, inp, dagger=False):
        """out = (1 + i a g5 tau3 + b tau1) in  (per parity view)."""
        from ..ops.dispatch import apply_twist_field, dwf5_op
        sgn = -1.0 if dagger else 1.0
        # out = in + b tau1 in  (Ls=2 s-hop with mf=-1 gives psi(1-s))
        dwf5_op(out, inp, 1.0, self.b_t, -1.0, kind=0)
        # out += i a g5 tau3 in
        apply_twist_field(out, inp, 0.0, sgn * self.a_t, tau3=True, acc=True)
        return out

    def _apply_Ainv(self, out, inp, dagger=False):
        """A^-1 = (1 - i a g5 tau3 - b tau1)/(1 + a^2 - b^2)."""
        from ..ops import blas
        from ..ops.dispatch import apply_twist_field, dwf5_op
        sgn = -1.0 if dagger else 1.0
        den = 1.0 + self.a_t ** 2 - self.b_t ** 2
        dwf5_op(out, inp, 1.0, -self.b_t, -1.0, kind=0)
        apply_twist_field(out, inp, 0.0, -sgn * self.a_t, tau3=True, acc=True)
        blas.scal(1.0 / den, out)
        return out

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        from ..ops.dispatch import dslash_wilson_slice, dwf_halo_exchange
        assert inp.n_parity == 2
        for p in (0, 1):
            op = out.parity_view(p)
            self._apply_A(op, inp.parity_view(p), dagger)
            io = inp.parity_view(1 - p)
            h = dwf_halo_exchange(io, 1 - p, dagger)
            for s in (0, 1):
                dslash_wilson_slice(op, io, self.gauge, p, s, dagger,
                                    a=-self.kappa, x=op, halo=h)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def flops_per_site(self) -> int:
        return 2 * 1320 + 2 * 96  # per 4-d site (two flavors)


class DiracNdegTwistedMassPC(DiracNdegTwistedMass):
    """Symmetric even-odd PC doublet:
    M_pc = 1 - kappa^2 Ainv D Ainv D (A flavor-structured, x-local;
    A and the 4-d hops commute since A is spin... A contains g5: it does
    NOT commute with D — apply in operator order like the Moebius PC)."""

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        from ..ops.dispatch import dslash_wilson_slice, dwf_halo_exchange
        assert inp.n_parity == 1
        t = self.tmp("ndeg_t", inp)
        u = self.tmp("ndeg_u", inp)
        k2 = -self.kappa ** 2

        def dhat(dst, src, parity, acc_into=None, a=1.0):
            h = dwf_halo_exchange(src, 1 - parity, dagger)
            if acc_into is None:
                dst.zero_()
            for s in (0, 1):
                dslash_wilson_slice(dst, src, self.gauge, parity, s, dagger,
                                    a=a, x=acc_into if acc_into is not None
                                    else dst, halo=h)
            return dst

        if not dagger:
            dhat(t, inp, 1)                    # t_o = D_oe in
            self._apply_Ainv(u, t)             # u = Ainv t
            dhat(t, u, 0)                      # t_e = D_eo u
            self._apply_Ainv(u, t)             # u = Ainv t
            from ..ops import blas
            blas.copy(out, inp)
            blas.axpy(k2, u, out)
        else:
            # M^dag = 1 - k^2 D^d Ainv^d D^d Ainv^d
            self._apply_Ainv(u, inp, dagger=True)
            dhat(t, u, 1)                      # (D_eo)^d -> odd
            self._apply_Ainv(u, t, dagger=True)
            dhat(t, u, 0)                      # (D_oe)^d -> even
            from ..ops import blas
            blas.copy(out, inp)
            blas.axpy(k2, t, out)
        return out

    def tmp(self, name, like):
        key = (name, like.precision, str(like.device), like.n_parity, 2)
        t = self._tmps.get(key)
        if t is None:
            t = SpinorField(self.geo, like.precision, like.device,
                            like.n_parity, ls=2)
            self._tmps[key] = t
        return t


class DiracCloverHasenbuschTwist(DiracClover):
    """Clover + Hasenbusch twist (ref: lib/dirac_clover_hasenbusch_twist.cpp):
    M' = M_clover + i mu_h g5 — used for Hasenbusch mass splitting in HMC.
    Composed as the fused clover op plus one accumulate-twist launch."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float,
                 mu_h: float):
        super().__init__(gauge, clover, kappa)
        self.mu_h = float(mu_h)

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        from ..ops.dispatch import apply_twist_field
        super().M(out, inp, dagger)
        sgn = -1.0 if dagger else 1.0
        apply_twist_field(out, inp, 0.0, sgn * self.mu_h, acc=True)
        return out


class DiracCloverHasenbuschTwistPC(DiracCloverPC):
    """PC version: M'_pc = M_pc + i mu_h g5
    (ref: dirac_quda.h DiracCloverHasenbuschTwistPC)."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float,
                 mu_h: float):
        super().__init__(gauge, clover, kappa)
        self.mu_h = float(mu_h)

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        from ..ops.dispatch import apply_twist_field
        super().M(out, inp, dagger)
        sgn = -1.0 if dagger else 1.0
        apply_twist_field(out, inp, 0.0, sgn * self.mu_h, acc=True)
        return out


class DiracTwistedCloverPC(Dirac, _CloverMixin):
    """Symmetric even-odd preconditioned twisted clover
    (ref: lib/dirac_twisted_clover.cpp DiracTwistedCloverPC):
    M_pc = 1 - kappa^2 Atc^-1 D Atc^-1 D with Atc = A + i eps g5.
    Atc^-1 = (A - i eps g5)(A^2 + eps^2)^-1 — both factors hermitian and
    chirality-block-diagonal (they all commute), so the inverse is two
    packed-clover applies + one twist accumulate per application.
    `b_inv` holds the precomputed (A^2 + eps^2)^-1 as a CloverField."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float,
                 mu: float):
        super().__init__(gauge, kappa)
        self.clover = clover
        self.mu = float(mu)
        self.eps = 2.0 * kappa * mu
        # build (A^2 + eps^2)^{-1} once (setup; ref "dynamic clover" role)
        import torch
        A = clover.to_complex()
        eye = torch.eye(12, dtype=A.dtype, device=A.device)
        B = torch.linalg.inv(A @ A + self.eps ** 2 * eye)
        self.b_inv = CloverField(gauge.geo, clover.precision,
                                 gauge.device).from_matrices(B)

    def _apply_Atc_inv(self, out, inp, parity, dagger=False):
        from ..ops.dispatch import apply_clover, apply_twist_field
        t = self.tmp("tc_t", inp)
        apply_clover(t, inp, self.b_inv, parity)        # t = B in
        apply_clover(out, t, self.clover, parity)       # out = A t
        sgn = 1.0 if dagger else -1.0
        apply_twist_field(out, t, 0.0, sgn * self.eps, acc=True)
        return out

    def M(self, out, inp, dagger: bool = False):
        k2 = -self.kappa ** 2
        t = self.tmp("tc_o", inp)
        u = self.tmp("tc_e", inp)
        if not dagger:
            self.dslash(t, inp, 1)
            self._apply_Atc_inv(u, t, 1)
            self.dslash(t, u, 0)
            self._apply_Atc_inv(u, t, 0)
            blas.copy(out, inp)
            blas.axpy(k2, u, out)
        else:
            # M^dag = 1 - k^2 D^d Atc^-dag D^d Atc^-dag
            self._apply_Atc_inv(u, inp, 0, dagger=True)
            self.dslash(t, u, 1, dagger=True)
            self._apply_Atc_inv(u, t, 1, dagger=True)
            self.dslash(t, u, 0, dagger=True)
            blas.copy(out, inp)
            blas.axpy(k2, t, out)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def prepare(self, b_full: SpinorField) -> SpinorField:
        """b' = Atc_ee^-1 (b_e + kappa D_eo Atc_oo^-1 b_o)."""
        t = self.new_spinor(b_full.precision)
        be = self.new_spinor(b_full.precision)
        self._apply_Atc_inv(t, b_full.parity_view(1), 1)
        self.dslash(be, t, 0, a=self.kappa, x=b_full.parity_view(0))
        self._apply_Atc_inv(t, be, 0)
        blas.copy(be, t)
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        """x_o = Atc_oo^-1 (b_o + kappa D_oe x_e)."""
        blas.copy(x_full.parity_view(0), x_e)
        t = self.tmp("tc_o", x_full)
        self.dslash(t, x_e, 1, a=self.kappa, x=b_full.parity_view(1))
        self._apply_Atc_inv(x_full.parity_view(1), t, 1)
        return x_full

    def flops_per_site(self) -> int:
        return 1320 + 2 * 504 + 48


def apply_gamma5(out: "SpinorField", inp: "SpinorField") -> "SpinorField":
    """out = g5 in (DeGrand-Rossi diag(1,1,-1,-1)); device path rides the
    twist kernel (i*g5 then a -i rescale), CPU flips the lower spins."""
    from ..ops import blas
    from ..ops.dispatch import apply_twist_field, on_gpu
    if on_gpu(out, inp):
        out.zero_()
        apply_twist_field(out, inp, 0.0, 1.0)  # out = i g5 in
        blas.caxpby(complex(0.0, -1.0), out, 0.0, out)
        return out
    v = inp.to_complex().clone()
    v[..., 2:4, :] = -v[..., 2:4, :]
    out.from_complex(v)
    return out


class DiracG5M:
    """gamma5-wrapped operator functor (ref: DiracG5M dirac_quda.h:2441):
    hermitian (indefinite) for gamma5-hermitian actions, so hermitian
    eigensolvers and MINRES-type methods apply directly."""

    def __init__(self, op):
        self.op = op
        self.geo = op.geo

    def new_spinor(self, precision="double", n_parity=2):
        return self.op.new_spinor(precision, n_parity)

    def M(self, out, inp, dagger=False):
        # for g5-hermitian M, (g5 M)^dag = M^dag g5 = g5 M: self-adjoint
        t = self.op.new_spinor(inp.precision, inp.n_parity)
        self.op.M(t, inp, dagger=False)
        return apply_gamma5(out, t)

    def MdagM(self, out, inp, tmp):
        # (g5 M)^2 = g5 M g5 M = M^dag M for g5-hermitian M
        self.M(tmp, inp)
        return self.M(out, tmp)


class DiracMdagMLocal:
    """MdagM with communications disabled (ref: DiracMdagMLocal
    dirac_quda.h:2510 — the MSPCG inner operator: each rank applies its
    local operator with frozen boundaries)."""

    def __init__(self, op):
        self.op = op
        self.geo = op.geo

    def MdagM(self, out, inp, tmp):
        from ..parallel import comms
        with comms.solo_mode():
            return self.op.MdagM(out, inp, tmp)


class DiracNdegTwistedClover(DiracNdegTwistedMass):
    """Non-degenerate twisted-clover doublet (ref: the
    QUDA_TWISTED_CLOVER ndeg branch of lib/dirac_twisted_clover.cpp /
    kernels/dslash_ndeg_twisted_clover.cuh):
      M = (C + i 2 kappa mu g5 tau3 - 2 kappa eps tau1) - kappa D
    on an ls=2 flavor doublet. With X = i a g5 tau3 + b tau1 and
    [C, X] = 0, the flavor-local inverse is
      A^-1 = (C - X) (C^2 + a^2 - b^2)^-1
    with the site-local G^-1 = (C^2 + a^2 - b^2)^-1 precomputed as a
    CloverField (same dynamic-inversion role as the degenerate PC)."""

    def __init__(self, gauge: GaugeField, clover: CloverField, kappa: float,
                 mu: float, epsilon: float):
        super().__init__(gauge, kappa, mu, epsilon)
        self.clover = clover
        import torch
        A = clover.to_complex()
        eye = torch.eye(12, dtype=A.dtype, device=A.device)
        G = A @ A + (self.a_t ** 2 - self.b_t ** 2) * eye
        self.g_inv = CloverField(gauge.geo, clover.precision,
                                 gauge.device).from_matrices(
                                     torch.linalg.inv(G))

    def tmp(self, name, like):
        key = (name, like.precision, str(like.device), like.n_parity, 2)
        t = self._tmps.get(key)
        if t is None:
            t = SpinorField(self.geo, like.precision, like.device,
                            like.n_parity, ls=2)
            self._tmps[key] = t
        return t

    def _clov5(self, out: SpinorField, inp: SpinorField, parity: int,
               field: CloverField) -> SpinorField:
        """out = field * in on BOTH flavor slices of a single-parity
        ls=2 field."""
        from ..ops.dispatch import apply_clover, on_gpu
        V = self.geo.volume_cb
        if on_gpu(out, inp):
            for s in (0, 1):
                apply_clover(out, inp, field, parity, v_stride=2 * V,
                             s_offset=s * V)
            return out
        from ..ops import reference as ref
        A = field.to_complex()[parity]
        psi = inp.to_complex()[0].reshape(2, V, 4, 3)
        res = torch.stack([ref.apply_clover(A, psi[0]),
                           ref.apply_clover(A, psi[1])])
        out.from_complex(res.reshape(2 * V, 4, 3).unsqueeze(0))
        return out

    # parity-aware A applications (clover needs the parity; the TM base
    # class applications are x-local and parity-blind)
    def _apply_A_p(self, out, inp, parity, dagger=False):
        from ..ops.dispatch import apply_twist_field
        sgn = -1.0 if dagger else 1.0
        self._clov5(out, inp, parity, self.clover)       # out = C in
        from ..ops.dispatch import dwf5_op
        dwf5_op(out, inp, 0.0, self.b_t, -1.0, kind=0, a=1.0, x=out)
        apply_twist_field(out, inp, 0.0, sgn * self.a_t, tau3=True,
                          acc=True)
        return out

    def _apply_Ainv_p(self, out, inp, parity, dagger=False):
        """out = (C - X) G^-1 in."""
        from ..ops import blas
        from ..ops.dispatch import apply_twist_field, dwf5_op
        sgn = -1.0 if dagger else 1.0
        t = self.tmp("ntc_g", inp)
        self._clov5(t, inp, parity, self.g_inv)          # t = G^-1 in
        self._clov5(out, t, parity, self.clover)         # out = C t
        # out -= X t  (X = i a g5 tau3 + b tau1, dagger flips a)
        dwf5_op(out, t, 0.0, -self.b_t, -1.0, kind=0, a=1.0, x=out)
        apply_twist_field(out, t, 0.0, -sgn * self.a_t, tau3=True,
                          acc=True)
        return out

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        from ..ops.dispatch import dslash_wilson_slice, dwf_halo_exchange
        assert inp.n_parity == 2
        for p in (0, 1):
            op = out.parity_view(p)
            self._apply_A_p(op, inp.parity_view(p), p, dagger)
            io = inp.parity_view(1 - p)
            h = dwf_halo_exchange(io, 1 - p, dagger)
            for s in (0, 1):
                dslash_wilson_slice(op, io, self.gauge, p, s, dagger,
                                    a=-self.kappa, x=op, halo=h)
        return out


class DiracNdegTwistedCloverPC(DiracNdegTwistedClover):
    """Symmetric even-odd PC doublet with clover:
    M_pc = 1 - kappa^2 Ainv D Ainv D (operator order as in the TM PC)."""

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        from ..ops import blas
        from ..ops.dispatch import dslash_wilson_slice, dwf_halo_exchange
        assert inp.n_parity == 1
        t = self.tmp("ntc_t", inp)
        u = self.tmp("ntc_u", inp)
        k2 = -self.kappa ** 2

        def dhat(dst, src, parity):
            h = dwf_halo_exchange(src, 1 - parity, dagger)
            dst.zero_()
            for s in (0, 1):
                dslash_wilson_slice(dst, src, self.gauge, parity, s, dagger,
                                    a=1.0, x=dst, halo=h)
            return dst

        if not dagger:
            dhat(t, inp, 1)
            self._apply_Ainv_p(u, t, 1)
            dhat(t, u, 0)
            self._apply_Ainv_p(u, t, 0)
            blas.copy(out, inp)
            blas.axpy(k2, u, out)
        else:
            self._apply_Ainv_p(u, inp, 0, dagger=True)
            dhat(t, u, 1)
            self._apply_Ainv_p(u, t, 1, dagger=True)
            dhat(t, u, 0)
            blas.copy(out, inp)
            blas.axpy(k2, t, out)
        return out


def _wilson_pc_mdagm_batch(self, outs, inps):
    """Batched MdagM for the multi-RHS solvers (attached to
    DiracWilsonPC): all four dslash applies ride the merged-halo batch
    path — one message per face for the whole block."""
    from ..ops.dispatch import dslash_wilson_batch
    n = len(inps)
    cache = self.__dict__.setdefault("_batch_tmps", {})
    key = (inps[0].precision, str(inps[0].device), n)
    ts = cache.get(key)
    if ts is None:
        mk = lambda: [SpinorField(self.geo, inps[0].precision,
                                  inps[0].device, inps[0].n_parity,
                                  nspin=4) for _ in range(n)]
        ts = (mk(), mk())
        cache[key] = ts
    t_odd, t_even = ts
    k2 = self.kappa * self.kappa
    dslash_wilson_batch(t_odd, inps, self.gauge, 1)
    dslash_wilson_batch(outs, t_odd, self.gauge, 0, a=-k2, xs=inps)
    dslash_wilson_batch(t_odd, outs, self.gauge, 1, dagger=True)
    dslash_wilson_batch(t_even, t_odd, self.gauge, 0, dagger=True,
                        a=-k2, xs=outs)
    for i in range(n):
        blas.copy(outs[i], t_even[i])
    return outs


DiracWilsonPC.MdagM_batch = _wilson_pc_mdagm_batch
