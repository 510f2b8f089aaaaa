"""Distance-preconditioned Wilson / Wilson-clover operators
(ref: lib/dslash_wilson_distance.cu + dslash_wilson.cuh:96-101 and the
weight of kernels/spinor_reweight.cuh:29, arXiv:1006.4028).

The reference scales the t-hops of the dslash by w(t±1)/w(t) with
  w(t) = cosh(alpha0 * ((t - t0 + nt) % nt - nt/2))      (alpha0 > 0)
       = 1/cosh(...)                                     (alpha0 < 0)
which is exactly the similarity transform M_dist = P^-1 M P with
P = diag(w(t)) (the site-diagonal 1/clover terms commute with P). We
apply that transform directly: scale in, fused stencil, scale out — two
site-diagonal device passes (the DistanceReweightSpinor kernel role; for
half fields the scale rides the per-site norm, zero extra data traffic).
t is the GLOBAL time coordinate and nt the global T extent, so the
weights are rank-aware on a T-partitioned grid.

Even-odd algebra (why prepare/reconstruct stay one-liners): with
b" = P^-1 b the transformed PC source is P_e^-1 prepare(b), and the
reconstruction of x = P x' reduces to the PLAIN reconstruct applied to
x_e = P_e x'_e — both derived in the class docstrings below."""

from __future__ import annotations

import torch

from ..fields.spinor import SpinorField
from .dirac import DiracClover, DiracCloverPC, DiracWilson, DiracWilsonPC


class DistanceWeight:
    """Per-site w / 1/w on the checkerboard layout."""

    def __init__(self, geo, alpha0: float, t0: int, device,
                 dtype=torch.float64):
        from ..parallel import comms
        self.alpha0 = float(alpha0)
        self.t0 = int(t0)
        nt = comms.grid_dims()[3] * geo.dims[3]
        tg0 = comms.grid_coords()[3] * geo.dims[3]
        lex_t = geo.coords[:, 3].to(torch.int64)
        t_cb = torch.stack([lex_t[geo.lex_of_cb[0]],
                            lex_t[geo.lex_of_cb[1]]]) + tg0   # [2, Vcb]
        arg = self.alpha0 * (((t_cb - self.t0 + nt) % nt).to(dtype)
                             - nt // 2)
        w = torch.cosh(arg) if self.alpha0 > 0 else 1.0 / torch.cosh(arg)
        self.w = w.to(device)
        self.winv = (1.0 / w).to(device)

    def apply(self, f: SpinorField, inverse: bool = False) -> SpinorField:
        """f *= w (or 1/w) in place. Single-parity fields are the EVEN
        checkerboard (the MATPC_EVEN_EVEN convention of models.dirac)."""
        w2 = self.winv if inverse else self.w
        w = w2 if f.n_parity == 2 else w2[0:1]
        if f.ls != 1:
            w = w.repeat(1, f.ls)  # site index = s*Vcb + x
        if f.norm is not None:
            f.norm.mul_(w.to(f.norm.dtype))
        else:
            f.data.mul_(w.to(f.data.dtype).unsqueeze(1).unsqueeze(-1))
        return f


def _distance_full(base):
    class _Distance(base):
        def __init__(self, *args, alpha0: float = 0.0, t0: int = 0,
                     **kwargs):
            super().__init__(*args, **kwargs)
            self.distance = DistanceWeight(self.geo, alpha0, t0,
                                           self.gauge.device)

        def M(self, out, inp, dagger: bool = False):
            # M_dist = P^-1 M P; M_dist^dag = P M^dag P^-1
            self.distance.apply(inp, inverse=dagger)
            try:
                super().M(out, inp, dagger=dagger)
            finally:
                self.distance.apply(inp, inverse=not dagger)  # restore
            self.distance.apply(out, inverse=not dagger)
            return out

        def prepare(self, b_full):
            """Transformed-system PC source: with b" = P^-1 b,
            b"_e + kappa D_dist,eo b"_o = P_e^-1 (b_e + kappa D_eo b_o)
            = P_e^-1 prepare(b) (A^-1 factors commute with P)."""
            be = super().prepare(b_full)
            return self.distance.apply(be, inverse=True)

        def reconstruct(self, x_full, x_e, b_full):
            """The physical solution is x = P x': x_e = P_e x'_e and
            x_o = b_o + kappa D_oe(P_e x'_e) (+ A^-1 factors) — i.e. the
            PLAIN reconstruct applied to P_e x'_e."""
            self.distance.apply(x_e)
            return super().reconstruct(x_full, x_e, b_full)

    _Distance.__name__ = base.__name__ + "Distance"
    return _Distance


DiracWilsonDistance = _distance_full(DiracWilson)
DiracWilsonDistancePC = _distance_full(DiracWilsonPC)
DiracCloverDistance = _distance_full(DiracClover)
DiracCloverDistancePC = _distance_full(DiracCloverPC)
