"""Staggered spin-taste operators + two-link Gaussian quark smearing
(ref: lib/spin_taste.cu + kernels/spin_taste.cuh applySpinTaste,
lib/staggered_two_link_quda.cu + staggered_quark_smearing.cu
performTwoLinkGaussianSmearNStep — re-derived)."""

from __future__ import annotations

import torch

from ..fields.gauge import GaugeField
from ..fields.geometry import LatticeGeometry, checkerboard_join, checkerboard_split
from ..fields.spinor import SpinorField


def spin_taste_phase(geo: LatticeGeometry, kind: str) -> torch.Tensor:
    """[V_lex] +-1 phase for the common local spin-taste structures:
    gamma5 x gamma5 ('g5-g5' = epsilon(x) = (-1)^(x+y+z+t)), gamma_mu
    ('gX','gY','gZ','gT': (-1)^{sum of coords before mu}... standard MILC
    local phases), '1' (identity)."""
    c = geo.coords.to(torch.int64)
    if kind == "1":
        e = torch.zeros(geo.volume, dtype=torch.int64)
    elif kind == "g5-g5":
        e = c.sum(dim=1)
    elif kind in ("gX", "gY", "gZ", "gT"):
        mu = "XYZT".index(kind[1])
        e = c[:, :mu].sum(dim=1) if mu > 0 else torch.zeros(geo.volume,
                                                            dtype=torch.int64)
    elif kind == "g5gT":  # epsilon * eta_T-style
        e = c[:, :3].sum(dim=1)
    else:
        raise ValueError(kind)
    return torch.where(e % 2 == 0, 1.0, -1.0).to(torch.float64)


def apply_spin_taste(psi: SpinorField, kind: str) -> SpinorField:
    """Multiply a staggered field by the local spin-taste phase."""
    geo = psi.geo
    ph = spin_taste_phase(geo, kind).to(psi.device)
    lex = checkerboard_join(psi.to_complex(), geo)
    lex = lex * ph.view(-1, *([1] * (lex.dim() - 1))).to(lex.dtype)
    out = psi.clone_empty()
    out.from_complex(checkerboard_split(lex, geo))
    return out


def two_link_laplace(two: GaugeField, psi: SpinorField, out: SpinorField,
                     *, a: float, b: float) -> SpinorField:
    """out = a * Lap2 psi + b * psi with the TWO-LINK spatial Laplacian
    Lap2 psi(x) = sum_{i<3}[N_i(x) psi(x+2i) + N_i(x-2i)^d psi(x-2i)]
    - 6 psi, N_i = U_i(x) U_i(x+i) (pass `two` as a shift=2-style field
    built by two_links)."""
    geo = psi.geo
    u = two.to_complex()
    lo = geo.lex_of_cb.to(psi.device)
    U = torch.empty((4, geo.volume, 3, 3), dtype=u.dtype, device=u.device)
    U[:, lo[0]] = u[:, 0]
    U[:, lo[1]] = u[:, 1]
    p = checkerboard_join(psi.to_complex(), geo)
    shp = p.shape
    p = p.reshape(geo.volume, -1, 3)
    acc = -6.0 * p

    def nbr(mu, disp):
        c = geo.coords.to(torch.int64).clone()
        c[:, mu] = (c[:, mu] + disp) % geo.dims[mu]
        X, Y, Z, _ = geo.dims
        return (((c[:, 3] * Z + c[:, 2]) * Y + c[:, 1]) * X + c[:, 0]).to(psi.device)

    for i in range(3):
        f = nbr(i, +2)
        bwd = nbr(i, -2)
        acc = acc + torch.einsum("vij,vsj->vsi", U[i], p[f])
        acc = acc + torch.einsum("vji,vsj->vsi", U[i][bwd].conj(), p[bwd])
    out.from_complex(checkerboard_split((a * acc + b * p).reshape(shp), geo))
    return out


def two_links(u: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """N_mu(x) = U_mu(x) U_mu(x+mu) (ref: staggered_two_link.cuh)."""
    lo = geo.lex_of_cb.to(u.device)
    U = torch.empty((4, geo.volume, 3, 3), dtype=u.dtype, device=u.device)
    U[:, lo[0]] = u[:, 0]
    U[:, lo[1]] = u[:, 1]
    N = torch.empty_like(U)
    for mu in range(4):
        c = geo.coords.to(torch.int64).clone()
        c[:, mu] = (c[:, mu] + 1) % geo.dims[mu]
        X, Y, Z, _ = geo.dims
        idx = (((c[:, 3] * Z + c[:, 2]) * Y + c[:, 1]) * X + c[:, 0]).to(u.device)
        N[mu] = U[mu] @ U[mu][idx]
    return torch.stack([N[:, lo[0]], N[:, lo[1]]], dim=1)


def gaussian_smear_two_link(u: torch.Tensor, geo: LatticeGeometry,
                            psi: SpinorField, *, width: float,
                            n_steps: int) -> SpinorField:
    """Staggered Gaussian smearing with two-link transport
    (ref: performTwoLinkGaussianSmearNStep):
    psi <- [(1 + w Lap2/(4 n)) ]^n psi with w = width^2."""
    N = two_links(u, geo)

    class _Wrap:  # two_link_laplace only reads .to_complex()
        def to_complex(self_inner):
            return N

    two = _Wrap()
    coef = width * width / (4.0 * n_steps)
    cur = psi
    for _ in range(n_steps):
        nxt = psi.clone_empty()
        two_link_laplace(two, cur, nxt, a=coef, b=1.0)
        cur = nxt
    return cur
