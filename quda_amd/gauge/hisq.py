"""Staggered link construction: fat7/asqtad fattening + Naik long links
(ref: lib/llfat_quda.cu + kernels/llfat.cuh computeKSLinkQuda — re-derived
from the staple recursion; path coefficients are parameters exactly like
the interface's path_coeff array, with u0=1 defaults)."""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..fields.geometry import LatticeGeometry
from .ops import _from_lex, _to_lex, _shift


@dataclass
class KSLinkCoeffs:
    """path_coeff analogue (computeKSLinkQuda quda.h): (1-link, 3-staple,
    5-staple, 7-staple, Lepage, Naik)."""
    c1: float = 1.0 / 8.0
    c3: float = 1.0 / 16.0
    c5: float = 1.0 / 64.0
    c7: float = 1.0 / 384.0
    lepage: float = 0.0
    naik: float = 0.0


def fat7_coefficients() -> KSLinkCoeffs:
    """Fat7 smearing (weights sum to 1 at unit gauge)."""
    return KSLinkCoeffs()


def asqtad_coefficients() -> KSLinkCoeffs:
    """Asqtad (u0=1): fat7 + Lepage + Naik, 1-link carries the 9/8-style
    correction (users tune their own tadpole-improved sets in practice)."""
    return KSLinkCoeffs(c1=5.0 / 8.0, c3=1.0 / 16.0, c5=1.0 / 64.0,
                        c7=1.0 / 384.0, lepage=-1.0 / 8.0, naik=-1.0 / 24.0)


def _staple_of(U, W_mu, geo, mu, nu):
    """Covariant staple of a mu-oriented path product W (x -> x+mu):
    S(x) = U_nu(x) W(x+nu) U_nu(x+mu)^d + U_nu(x-nu)^d W(x-nu) U_nu(x+mu-nu)."""
    Unu = U[nu]
    W_xnu = _shift(W_mu, geo, nu, +1)
    Unu_xmu = _shift(Unu, geo, mu, +1)
    S = Unu @ W_xnu @ Unu_xmu.conj().mT
    Unu_mnu = _shift(Unu, geo, nu, -1)
    W_mnu = _shift(W_mu, geo, nu, -1)
    Unu_xmu_mnu = _shift(Unu_mnu, geo, mu, +1)
    S = S + Unu_mnu.conj().mT @ W_mnu @ Unu_xmu_mnu
    return S


def fat_links(u: torch.Tensor, geo: LatticeGeometry,
              coeffs: KSLinkCoeffs) -> torch.Tensor:
    """[4,2,V,3,3] -> fattened links (same layout)."""
    U = _to_lex(u, geo)
    F = torch.empty_like(U)
    for mu in range(4):
        acc = coeffs.c1 * U[mu]
        for nu in range(4):
            if nu == mu:
                continue
            S3 = _staple_of(U, U[mu], geo, mu, nu)
            acc = acc + coeffs.c3 * S3
            if coeffs.lepage != 0.0:
                acc = acc + coeffs.lepage * _staple_of(U, S3, geo, mu, nu)
            for rho in range(4):
                if rho in (mu, nu):
                    continue
                S5 = _staple_of(U, S3, geo, mu, rho)
                acc = acc + coeffs.c5 * S5
                for sig in range(4):
                    if sig in (mu, nu, rho):
                        continue
                    acc = acc + coeffs.c7 * _staple_of(U, S5, geo, mu, sig)
        F[mu] = acc
    return _from_lex(F, geo)


def naik_links(u: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """Long (3-hop straight) links N_mu(x) = U(x) U(x+mu) U(x+2mu)."""
    U = _to_lex(u, geo)
    N = torch.empty_like(U)
    for mu in range(4):
        U1 = _shift(U[mu], geo, mu, +1)
        U2 = _shift(U1, geo, mu, +1)
        N[mu] = U[mu] @ U1 @ U2
    return _from_lex(N, geo)


def unitarize_links(f: torch.Tensor, *, svd_rel_error: float = 1e-6,
                    return_failures: bool = False):
    """Project fattened links to U(3) via the symmetric polar factor
    W = F (F^d F)^{-1/2} (ref: lib/unitarize_links_quda.cu — SVD route).
    Counts links whose smallest singular value is below
    svd_rel_error * largest (the reference's `num_failures` device
    counter for ill-conditioned fattened links, interface_quda.cpp:119);
    pass return_failures=True to receive (W, n_failures)."""
    U_, S_, Vh = torch.linalg.svd(f)
    W = U_ @ Vh
    fails = int((S_[..., -1] < svd_rel_error * S_[..., 0]).sum().item())
    if return_failures:
        return W, fails
    if fails:
        import warnings
        warnings.warn(f"unitarize_links: {fails} ill-conditioned links")
    return W


def hisq_two_level_links(u: torch.Tensor, geo, *, coeffs1=None,
                         coeffs2=None):
    """Full HISQ link chain (ref: the computeKSLinkQuda double-fattening
    path, llfat + unitarize_links + second llfat): level-1 fat7 smear,
    PROJECT to U(3) (SVD polar), level-2 asqtad reweight on the
    unitarized links; Naik long links built from the unitarized field.
    Everything stays torch-differentiable (the unitarization included,
    via the SVD backward) so the HISQ force can chain through it."""
    c1 = coeffs1 or fat7_coefficients()
    c2 = coeffs2 or asqtad_coefficients()
    V = fat_links(u, geo, c1)
    U_, S_, Vh = torch.linalg.svd(V)
    W = U_ @ Vh
    X = fat_links(W, geo, c2)
    N = naik_links(W, geo)
    return X, N


def hisq_full_force(u: torch.Tensor, geo, mass: float, phi, *,
                    cg_tol: float = 1e-10, cg_maxiter: int = 3000):
    """HISQ force with the unitarization INSIDE the differentiated chain
    (ref: lib/unitarize_force_quda.cu role — the reference hand-derives
    the SVD derivative; here torch's SVD backward supplies it).
    Returns (S_f, F)."""
    from ..fields.gauge import GaugeField
    from ..models import DiracImprovedStaggered
    from ..ops import blas
    from ..ops.reference import (dslash_staggered_naik_parity,
                                 dslash_staggered_parity)
    from ..solvers.variants import _NormalOp
    from .fermion_force import autograd_fermion_force
    from ..solvers import cg_solve
    fat, lng = hisq_two_level_links(u, geo)
    gf = GaugeField(geo, "double", phi.device).from_complex(fat)
    gl = GaugeField(geo, "double", phi.device, shift=3).from_complex(lng)
    d = DiracImprovedStaggered(gf, gl, mass)
    X = d.new_spinor(n_parity=2)
    st = cg_solve(_NormalOp(d, mmdag=True), X, phi, tol=cg_tol,
                  maxiter=cg_maxiter)
    assert st.converged, "hisq full force CG failed"
    Y = d.new_spinor(n_parity=2)
    d.M(Y, X, dagger=True)
    S_f = blas.re_dot(phi, X)

    def apply_M(u_t, psi):
        f_t, n_t = hisq_two_level_links(u_t, geo)
        out = 2.0 * mass * psi.clone()
        for p in (0, 1):
            out[p] = out[p] + dslash_staggered_parity(f_t, psi[1 - p],
                                                      geo, p)
            out[p] = out[p] + dslash_staggered_naik_parity(n_t, psi[1 - p],
                                                           geo, p)
        return out

    F = autograd_fermion_force(u, geo, apply_M, X.to_complex(),
                               Y.to_complex())
    return S_f, F
