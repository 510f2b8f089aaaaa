"""Domain-wall (Shamir) and Moebius Dirac operators
(ref: lib/dirac_domain_wall.cpp, lib/dirac_mobius.cpp — re-derived).

Operator decomposition used throughout (DeGrand-Rossi, spin-diagonal P±):

  M = A - (1/2) Dhat B
    A = alpha + beta Ds     alpha = 1 + b5 (4 - M5),  beta = c5 (4 - M5) - 1
    B = b5 + c5 Ds
    Dhat = the 4-d hopping sum (per s-slice; parity-hopping)
    Ds   = P+ shift(s-1) + P- shift(s+1) with -mf boundary wraps

  Shamir DWF = Moebius(b5=1, c5=0):  M = (1 + (4-M5)) - Ds - (1/2) Dhat.

Symmetric even-odd PC (note Dhat does NOT commute with the s-structure:
its spin projectors mix chirality, so operator order matters):
  M_pc = 1 - (1/4) Ainv Dhat_eo B Ainv Dhat_oe B     on even sites.
"""

from __future__ import annotations

from typing import Optional

from ..fields.gauge import GaugeField
from ..fields.spinor import SpinorField
from ..ops import blas
from ..ops.dispatch import dslash_wilson_slice, dwf5_op


class DiracMobius:
    """Full 5-d Moebius operator."""

    def __init__(self, gauge: GaugeField, mf: float, m5: float, Ls: int,
                 b5: float = 1.5, c5: float = 0.5):
        self.gauge = gauge
        self.geo = gauge.geo
        self.mf = float(mf)
        self.m5 = float(m5)
        self.Ls = int(Ls)
        self.b5 = float(b5)
        self.c5 = float(c5)
        d4 = 4.0 - self.m5
        self.alpha = 1.0 + self.b5 * d4
        self.beta = self.c5 * d4 - 1.0

    # -- field helpers ------------------------------------------------------
    def new_spinor(self, precision=None, n_parity=1) -> SpinorField:
        return SpinorField(self.geo, precision or self.gauge.precision,
                           self.gauge.device, n_parity, ls=self.Ls)

    def _tmp(self, name, like):
        key = (name, like.precision, str(like.device), like.n_parity)
        cache = self.__dict__.setdefault("_tmps", {})
        t = cache.get(key)
        if t is None:
            t = SpinorField(self.geo, like.precision, like.device,
                            like.n_parity, ls=self.Ls)
            cache[key] = t
        return t

    # -- s-structure applications ------------------------------------------
    def apply_A(self, out, inp, dagger=False):
        return dwf5_op(out, inp, self.alpha, self.beta, self.mf, kind=0,
                       dagger=dagger)

    def apply_Ainv(self, out, inp, dagger=False, a=1.0, x=None):
        return dwf5_op(out, inp, self.alpha, self.beta, self.mf, kind=1,
                       dagger=dagger, a=a, x=x)

    def apply_B(self, out, inp, dagger=False):
        if self.c5 == 0.0 and self.b5 == 1.0:
            return blas.copy(out, inp)
        return dwf5_op(out, inp, self.b5, self.c5, self.mf, kind=0,
                       dagger=dagger)

    def _dhat(self, out, inp, parity, dagger=False, a=1.0, xpay=True):
        """out[s] = [out[s] +] a * Dhat in[s] for every slice: one halo
        exchange for all slices + s-batched multi-RHS kernels sharing the
        gauge loads (dispatch.dslash_wilson_slices)."""
        from ..ops.dispatch import dslash_wilson_slices, dwf_halo_exchange
        h = dwf_halo_exchange(inp, 1 - parity, dagger)
        dslash_wilson_slices(out, inp, self.gauge, parity, dagger, a=a,
                             x=out if xpay else None, halo=h)
        return out

    # -- full operator ------------------------------------------------------
    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        """out_p = A in_p - (1/2) Dhat_p,1-p (B in_{1-p})  (both parities)."""
        assert inp.n_parity == 2
        t = self._tmp("mob_t", inp.parity_view(0))
        for p in (0, 1):
            op = out.parity_view(p)
            ip = inp.parity_view(p)
            io = inp.parity_view(1 - p)
            if not dagger:
                self.apply_B(t, io)
                self.apply_A(op, ip)
                self._dhat(op, t, p, a=-0.5)
            else:
                # M^dag = A^dag - (1/2) B^dag Dhat^dag
                t.zero_()
                self._dhat(t, io, p, dagger=True, a=1.0, xpay=False)
                u = self._tmp("mob_u", t)
                self.apply_B(u, t, dagger=True)
                self.apply_A(op, ip, dagger=True)
                blas.axpy(-0.5, u, op)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def flops_per_site(self) -> int:
        return 1320 + 96  # per 4-d site per slice: Wilson + s-structure


class DiracMobiusPC(DiracMobius):
    """Symmetric even-even PC:
    M_pc = 1 - (1/4) Ainv Dhat_eo B Ainv Dhat_oe B
    (ref: lib/dirac_mobius.cpp DiracMobiusPC; right-to-left application
    order is B, Dhat, Ainv, B, Dhat, Ainv)."""

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        assert inp.n_parity == 1
        t = self._tmp("pc_t", inp)
        u = self._tmp("pc_u", inp)
        if not dagger:
            self.apply_B(t, inp)                    # t = B in  (even)
            u.zero_()
            self._dhat(u, t, 1, xpay=False)         # u = Dhat_oe t  (odd)
            self.apply_Ainv(t, u)                   # t = Ainv u
            self.apply_B(u, t)                      # u = B t
            t.zero_()
            self._dhat(t, u, 0, xpay=False)         # t = Dhat_eo u (even)
            self.apply_Ainv(out, t, a=-0.25, x=inp)
        else:
            # M^dag = 1 - (1/4) B^d Dhat_oe^d Ainv^d B^d Dhat_eo^d Ainv^d
            self.apply_Ainv(t, inp, dagger=True)    # even
            u.zero_()
            self._dhat(u, t, 1, dagger=True, xpay=False)  # (Dhat_eo)^d -> odd
            self.apply_B(t, u, dagger=True)
            self.apply_Ainv(u, t, dagger=True)      # odd
            t.zero_()
            self._dhat(t, u, 0, dagger=True, xpay=False)  # (Dhat_oe)^d -> even
            v = self._tmp("pc_v", inp)
            self.apply_B(v, t, dagger=True)
            blas.copy(out, inp)
            blas.axpy(-0.25, v, out)
        return out

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp, dagger=False)
        self.M(out, tmp, dagger=True)
        return out

    def prepare(self, b_full: SpinorField) -> SpinorField:
        """b' = Ainv [b_e + (1/2) Dhat_eo B Ainv b_o]."""
        be = self.new_spinor(b_full.precision)
        t = self._tmp("prep_t", be)
        u = self._tmp("prep_u", be)
        self.apply_Ainv(t, b_full.parity_view(1))
        self.apply_B(u, t)
        t.zero_()
        self._dhat(t, u, 0, xpay=False)
        blas.scal(0.5, t)
        blas.axpy(1.0, b_full.parity_view(0), t)
        self.apply_Ainv(be, t)
        return be

    def reconstruct(self, x_full: SpinorField, x_e: SpinorField,
                    b_full: SpinorField):
        """x_o = Ainv [b_o + (1/2) Dhat_oe B x_e]."""
        blas.copy(x_full.parity_view(0), x_e)
        t = self._tmp("rec_t", x_e)
        u = self._tmp("rec_u", x_e)
        self.apply_B(u, x_e)
        t.zero_()
        self._dhat(t, u, 1, xpay=False)
        blas.scal(0.5, t)
        blas.axpy(1.0, b_full.parity_view(1), t)
        self.apply_Ainv(x_full.parity_view(1), t)
        return x_full


class DiracDomainWall(DiracMobius):
    """Shamir DWF = Moebius(b5=1, c5=0) (ref: lib/dirac_domain_wall.cpp)."""

    def __init__(self, gauge: GaugeField, mf: float, m5: float, Ls: int):
        super().__init__(gauge, mf, m5, Ls, b5=1.0, c5=0.0)


class DiracDomainWallPC(DiracMobiusPC):
    def __init__(self, gauge: GaugeField, mf: float, m5: float, Ls: int):
        super().__init__(gauge, mf, m5, Ls, b5=1.0, c5=0.0)


class DiracZMobius(DiracMobius):
    """zMobius: complex per-slice b5[s], c5[s] (ref: lib/dirac_mobius.cpp
    zMobius branch + dslash5_domain_wall.cu M5_ZMOBIUS — re-derived; the
    per-slice tables are assembled host-side, see ops.dispatch._ztables).

      A_s = 1 + b5[s](4 - M5),  B hop = c5[s];  M = A - (1/2) Dhat B
    """

    def __init__(self, gauge: GaugeField, mf: float, m5: float, Ls: int,
                 b5, c5):
        b5 = [complex(v) for v in b5]
        c5 = [complex(v) for v in c5]
        assert len(b5) == Ls and len(c5) == Ls
        self.gauge = gauge
        self.geo = gauge.geo
        self.mf = float(mf)
        self.m5 = float(m5)
        self.Ls = int(Ls)
        self.b5v = b5
        self.c5v = c5
        d4 = 4.0 - self.m5
        self.alpha5 = [1.0 + b * d4 for b in b5]
        self.beta5 = [c * d4 - 1.0 for c in c5]

    def apply_A(self, out, inp, dagger=False):
        from ..ops.dispatch import zdwf5_op
        return zdwf5_op(out, inp, self.alpha5, self.beta5, self.mf, 0,
                        dagger=dagger)

    def apply_Ainv(self, out, inp, dagger=False, a=1.0, x=None):
        from ..ops.dispatch import zdwf5_op
        return zdwf5_op(out, inp, self.alpha5, self.beta5, self.mf, 1,
                        dagger=dagger, a=a, x=x)

    def apply_B(self, out, inp, dagger=False):
        from ..ops.dispatch import zdwf5_op
        return zdwf5_op(out, inp, self.b5v, self.c5v, self.mf, 0,
                        dagger=dagger)


class DiracZMobiusPC(DiracZMobius, DiracMobiusPC):
    """Symmetric even-even PC zMobius (M/prepare/reconstruct inherited from
    DiracMobiusPC; the s-operators from DiracZMobius)."""
    pass


class DiracMobiusEofa(DiracMobius):
    """Moebius EOFA operator (ref: lib/dirac_mobius_eofa.cpp
    DiracMobiusEofa + dslash5_mobius_eofa.cu — re-derived): the 5th-dim
    diagonal block A gains a rank-1 chiral term

        A_eofa = A + eofa_shift * P_pm |u><u|,   u_s ~ kappa^s

    where kappa = -beta/alpha is the bulk decay rate of the M5 recursion
    (|u> is the exponential surface-mode profile at the pm wall; pm=+1
    profiles decay from s=0, pm=-1 from s=Ls-1). mq1 is the base quark
    mass; mq2/mq3 are stored for the one-flavor action bookkeeping (the
    det-ratio pseudofermion wiring is layered above these operators).
    The inverse applies one host-folded Sherman-Morrison correction on
    top of the O(Ls) M5 inverse."""

    def __init__(self, gauge: GaugeField, m5: float, Ls: int,
                 b5: float = 1.5, c5: float = 0.5, *, mq1: float = 0.01,
                 mq2: float = 0.08, mq3: float = 0.08, eofa_pm: int = 1,
                 eofa_shift: float = -0.1):
        super().__init__(gauge, mq1, m5, Ls, b5, c5)
        self.mq1, self.mq2, self.mq3 = float(mq1), float(mq2), float(mq3)
        self.eofa_pm = int(eofa_pm)
        self.eofa_shift = float(eofa_shift)
        kap = -self.beta / self.alpha
        prof = [kap ** s for s in range(Ls)]
        if self.eofa_pm < 0:
            prof = prof[::-1]
        import math
        nrm = math.sqrt(sum(p * p for p in prof))
        self.eofa_u = [p / nrm for p in prof]

    def apply_A(self, out, inp, dagger=False):
        from ..ops.dispatch import eofa5_op
        return eofa5_op(out, inp, self.alpha, self.beta, self.mf, 0,
                        self.eofa_shift, self.eofa_pm, self.eofa_u,
                        self.eofa_u, dagger=dagger)

    def apply_Ainv(self, out, inp, dagger=False, a=1.0, x=None):
        from ..ops.dispatch import eofa5_op
        return eofa5_op(out, inp, self.alpha, self.beta, self.mf, 1,
                        self.eofa_shift, self.eofa_pm, self.eofa_u,
                        self.eofa_u, dagger=dagger, a=a, x=x)


class DiracMobiusEofaPC(DiracMobiusEofa, DiracMobiusPC):
    """Symmetric even-even PC EOFA (structure from DiracMobiusPC, the
    rank-1-extended s-ops from DiracMobiusEofa)."""
    pass


class DiracDomainWall4D(DiracDomainWall):
    """4-d-preconditioning flavor of Shamir DWF (ref:
    lib/dirac_domain_wall_4d.cpp). In this engine's decomposition
    M = A - (1/2) Dhat B with B = 1 for Shamir, the 4-d even-odd Schur
    complement IS the preconditioned system DiracDomainWallPC builds
    (M_pc = 1 - 1/4 Ainv Dhat Ainv Dhat), so the 4-d-PC operator
    coincides with the symmetric PC class; the alias keeps the
    reference's operator taxonomy addressable."""
    pass


class DiracDomainWall4DPC(DiracDomainWallPC):
    """See DiracDomainWall4D."""
    pass
