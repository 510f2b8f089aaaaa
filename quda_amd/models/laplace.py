"""Gauge Laplacian, covariant derivative, and Wuppertal quark smearing
(ref: lib/gauge_laplace.cpp GaugeLaplace, lib/gauge_covdev.cpp
GaugeCovDev, kernels/laplace.cuh + covariant_derivative.cuh,
performFermionSmearQuda — torch implementations over the oracle layout;
spin-diagonal, so they apply per spin component)."""

from __future__ import annotations

import torch

from ..fields.gauge import GaugeField
from ..fields.geometry import LatticeGeometry, checkerboard_join, checkerboard_split
from ..fields.spinor import SpinorField
from ..ops import blas


def _lex_fields(gauge: GaugeField):
    geo = gauge.geo
    u = gauge.to_complex()
    lo = geo.lex_of_cb.to(u.device)
    U = torch.empty((4, geo.volume, 3, 3), dtype=u.dtype, device=u.device)
    U[:, lo[0]] = u[:, 0]
    U[:, lo[1]] = u[:, 1]
    return U


def covdev_apply(gauge: GaugeField, psi: SpinorField, mu: int,
                 forward: bool = True) -> SpinorField:
    """Covariant derivative hop (ref: GaugeCovDev::MCD):
    forward: out(x) = U_mu(x) psi(x+mu); backward: U_mu(x-mu)^d psi(x-mu)."""
    geo = gauge.geo
    U = _lex_fields(gauge)
    p = checkerboard_join(psi.to_complex(), geo)
    if forward:
        idx = geo.neighbor_lex(mu, +1).to(p.device)
        out = torch.einsum("vij,vsj->vsi", U[mu], p[idx])
    else:
        idx = geo.neighbor_lex(mu, -1).to(p.device)
        out = torch.einsum("vji,vsj->vsi", U[mu][idx].conj(), p[idx])
    r = psi.clone_empty()
    r.from_complex(checkerboard_split(out, geo))
    return r


def laplace_apply(gauge: GaugeField, psi: SpinorField, out: SpinorField,
                  *, ndim: int = 3, a: float = 1.0, b: float = 0.0
                  ) -> SpinorField:
    """out = a * Lap psi + b * psi with
    Lap psi(x) = sum_{mu<ndim} [U psi(x+mu) + U^d psi(x-mu)] - 2 ndim psi
    (ref: kernels/laplace.cuh; ndim=3 spatial for LapH/smearing)."""
    geo = gauge.geo
    U = _lex_fields(gauge)
    p = checkerboard_join(psi.to_complex(), geo)
    shp = p.shape
    p = p.reshape(geo.volume, -1, 3)  # spin dim = 1 for nspin=1 fields
    acc = -2.0 * ndim * p
    for mu in range(ndim):
        fwd = geo.neighbor_lex(mu, +1).to(p.device)
        bwd = geo.neighbor_lex(mu, -1).to(p.device)
        acc = acc + torch.einsum("vij,vsj->vsi", U[mu], p[fwd])
        acc = acc + torch.einsum("vji,vsj->vsi", U[mu][bwd].conj(), p[bwd])
    res = (a * acc + b * p).reshape(shp)
    out.from_complex(checkerboard_split(res, geo))
    return out


class GaugeLaplace:
    """Laplace operator in Dirac-operator clothing (usable with the
    solver/eigensolver stack; hermitian negative semi-definite -> M uses
    -Lap + m2 which is HPD for m2>0)."""

    def __init__(self, gauge: GaugeField, m2: float = 0.0, ndim: int = 3):
        self.gauge = gauge
        self.geo = gauge.geo
        self.m2 = float(m2)
        self.ndim = ndim

    def new_spinor(self, precision=None, n_parity=2, nspin=4) -> SpinorField:
        return SpinorField(self.geo, precision or self.gauge.precision,
                           self.gauge.device, n_parity, nspin=nspin)

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        # hermitian: dagger is a no-op
        return laplace_apply(self.gauge, inp, out, ndim=self.ndim, a=-1.0,
                             b=self.m2)

    def MdagM(self, out, inp, tmp):
        self.M(tmp, inp)
        self.M(out, tmp)
        return out


def wuppertal_smear(gauge: GaugeField, psi: SpinorField, *, alpha: float,
                    n_steps: int) -> SpinorField:
    """Wuppertal smearing psi <- (1 + alpha Lap_3d / (1+6alpha))^n psi
    (ref: performFermionSmearQuda / wuppertalStep)."""
    # psi' = (psi + alpha * sum_hops) / (1 + 6 alpha)
    #      = a * Lap psi + psi   with a = alpha/(1+6alpha)
    #   (Lap = hops - 6: a*(-6) + 1 = (1+6alpha-6alpha)/(1+6alpha) checks out)
    a = alpha / (1.0 + 6.0 * alpha)
    cur = psi
    for _ in range(n_steps):
        nxt = psi.clone_empty()
        laplace_apply(gauge, cur, nxt, ndim=3, a=a, b=1.0)
        cur = nxt
    return cur
