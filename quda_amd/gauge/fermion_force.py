"""Two-flavor Wilson pseudofermion force + fermionic HMC
(ref: lib/clover_force.cpp / wilson outer-product pieces,
kernels/clover_outer_product.cuh — re-derived for OUR conventions:
H = -sum tr P^2 + S_g + S_f,  Udot = P U,
S_f = phi^dag (M M^dag)^{-1} phi,  M = 1 - kappa D.

With X = (M M^dag)^{-1} phi and Y = M^dag X:
  dS_f/dt = -2 Re[X^dag dM/dt Y] = 2 kappa Re sum_{x,mu} tr[ Pdot_mu(x)
            ( U_mu(x) G1_mu(x) - G2_mu(x) U_mu(x)^dag ) ]
  G1(x,mu)_{c,a} = sum_s [P(-mu) Y](x+mu)_{s,c} conj(X(x))_{s,a}
  G2(x,mu)_{c,b} = sum_s [P(+mu) Y](x)_{s,c} conj(X(x+mu))_{s,b}
so energy conservation fixes  F_f = kappa TA[U G1 - G2 U^dag]
(TA = project_ta; validated by the finite-difference test)."""

from __future__ import annotations

from typing import Tuple

import torch

from ..fields.geometry import LatticeGeometry, checkerboard_join, checkerboard_split
from ..fields.gauge import GaugeField
from ..fields.spinor import SpinorField
from ..models import DiracWilson
from ..ops import blas
from ..ops.reference import _gamma_tensors
from ..solvers.cg import cg_solve
from ..solvers.variants import _NormalOp
from .ops import _from_lex, _to_lex, exp_su3, gauge_action, gauge_force, project_ta
from .hmc import mom_action, random_momentum


def wilson_fermion_force(u: torch.Tensor, geo: LatticeGeometry, kappa: float,
                         X: torch.Tensor, Y: torch.Tensor) -> torch.Tensor:
    """F_mu(x) = kappa TA[U_mu(x) G1 - G2 U_mu(x)^dag]; X/Y are full-parity
    oracle fields [2,Vcb,4,3]."""
    from ..parallel.halo import shift_lex
    dev, dt = u.device, u.dtype
    P = _gamma_tensors(dev, dt)  # [mu, 0:minus/1:plus, 4, 4]
    U = _to_lex(u, geo)
    Xl = checkerboard_join(X, geo)
    Yl = checkerboard_join(Y, geo)
    F = torch.empty_like(U)
    for mu in range(4):
        Y_xmu = shift_lex(Yl, geo, mu, +1)   # neighbor-rank slab when split
        X_xmu = shift_lex(Xl, geo, mu, +1)
        PmY_xmu = torch.einsum("st,vtc->vsc", P[mu, 0], Y_xmu)
        PpY = torch.einsum("st,vtc->vsc", P[mu, 1], Yl)
        G1 = torch.einsum("vsc,vsa->vca", PmY_xmu, Xl.conj())
        G2 = torch.einsum("vsc,vsb->vcb", PpY, X_xmu.conj())
        F[mu] = kappa * project_ta(U[mu] @ G1 - G2 @ U[mu].conj().mT)
    return _from_lex(F, geo)


def pseudofermion_refresh(d: DiracWilson, seed: int) -> SpinorField:
    """phi = M eta with Gaussian eta (exact heatbath: S_f = |eta|^2)."""
    eta = d.new_spinor(n_parity=2)
    eta.gaussian_(seed=seed)
    phi = d.new_spinor(n_parity=2)
    d.M(phi, eta)
    return phi


def fermion_action_and_force(u: torch.Tensor, geo: LatticeGeometry,
                             kappa: float, phi: SpinorField, *,
                             cg_tol: float = 1e-10, cg_maxiter: int = 2000
                             ) -> Tuple[float, torch.Tensor]:
    """S_f = phi^dag (M Mdag)^{-1} phi (global) and its MD force
    (multi-rank: the analytic Wilson force uses distributed shifts)."""
    g = GaugeField(geo, "double", phi.device).from_complex(u)
    d = DiracWilson(g, kappa)
    X = d.new_spinor(n_parity=2)
    st = cg_solve(_NormalOp(d, mmdag=True), X, phi, tol=cg_tol,
                  maxiter=cg_maxiter)
    assert st.converged, "fermion force CG failed"
    Y = d.new_spinor(n_parity=2)
    d.M(Y, X, dagger=True)
    S_f = blas.re_dot(phi, X)
    F = wilson_fermion_force(u, geo, kappa, X.to_complex(), Y.to_complex())
    return S_f, F


def hmc_trajectory_2f(u: torch.Tensor, geo: LatticeGeometry, beta: float,
                      kappa: float, *, n_md: int = 20, tau: float = 0.5,
                      seed: int = 0, cg_tol: float = 1e-10):
    """One two-flavor Wilson HMC trajectory (leapfrog, Metropolis).
    Returns (u', accepted, dH)."""
    P = random_momentum(geo, u.device, seed, u.dtype)
    gph = GaugeField(geo, "double", u.device).from_complex(u)
    phi = pseudofermion_refresh(DiracWilson(gph, kappa), seed + 7)

    def total_force(uc):
        Sf, Ff = fermion_action_and_force(uc, geo, kappa, phi, cg_tol=cg_tol)
        return gauge_force(uc, geo, beta) + Ff, Sf

    def hamiltonian(uc, Pc):
        Sf, _ = fermion_action_and_force(uc, geo, kappa, phi, cg_tol=cg_tol)
        return mom_action(Pc) + gauge_action(uc, geo, beta) + Sf

    dt = tau / n_md
    H0 = hamiltonian(u, P)
    uc = u.clone()
    F, _ = total_force(uc)
    P = P + 0.5 * dt * F
    for k in range(n_md):
        U = _to_lex(uc, geo)
        U = exp_su3(_to_lex(P, geo), dt) @ U
        uc = _from_lex(U, geo)
        F, _ = total_force(uc)
        P = P + (0.5 if k == n_md - 1 else 1.0) * dt * F
    H1 = hamiltonian(uc, P)
    dH = H1 - H0
    g = torch.Generator().manual_seed(seed + 13)
    accept = torch.rand(1, generator=g).item() < min(
        1.0, float(torch.exp(torch.tensor(-dH))))
    return (uc if accept else u), accept, dH


# ---------------------------------------------------------------------------
# generic autograd force (clover & friends)
# ---------------------------------------------------------------------------

def autograd_fermion_force(u: torch.Tensor, geo: LatticeGeometry,
                           apply_M, X: torch.Tensor, Y: torch.Tensor
                           ) -> torch.Tensor:
    """Exact MD force for S_f = phi^dag (M Mdag)^{-1} phi via reverse-mode
    differentiation of the (torch-differentiable) oracle operator
    (role of ref lib/clover_force.cpp computeCloverForceQuda — instead of
    hand-deriving the sigma-outer-product chain rule, the whole
    dS = -2 Re<X, dM Y> contraction is backpropagated through the
    clover/field-strength construction; validated against finite
    differences and the analytic Wilson force in the tests).

    apply_M(u, psi) must build M(u) psi with torch ops ([2,V,4,3] fields).
    Returns F with Pdot = F under H = -sum tr P^2 + S, Udot = P U:
    dS/dt = 2 Re tr(P U g^dag) (g = torch grad) and the K = -tr P^2
    convention give F = (1/2) TA[U g^dag] (verified exactly against the
    hand-derived Wilson force and by finite differences)."""
    from ..parallel import comms
    assert comms.comm_mask() == 0, \
        "fermion forces are single-rank this round (multi-rank HMC: round 2)"
    u_req = u.detach().clone().requires_grad_(True)
    MY = apply_M(u_req, Y)
    s = -2.0 * (X.conj() * MY).sum().real
    s.backward()
    g = u_req.grad  # dS/d(conj u) per torch's convention for real scalars
    from .ops import project_ta
    F = torch.empty_like(u)
    for mu in range(4):
        for p in (0, 1):
            F[mu, p] = 0.5 * project_ta(u[mu, p] @ g[mu, p].conj().mT)
    return F


def clover_fermion_force(u: torch.Tensor, geo: LatticeGeometry, kappa: float,
                         csw: float, phi: SpinorField, *,
                         cg_tol: float = 1e-10, cg_maxiter: int = 2000):
    """S_f and force for the two-flavor Wilson-CLOVER action
    (M = A(U) - kappa D with the clover term differentiated through its
    field-strength construction)."""
    from ..fields.clover import CloverField
    from ..models import DiracClover
    from ..ops.reference import clover_matrix, apply_clover, dslash_wilson_full

    def apply_M(u_t, psi):
        A = clover_matrix(u_t, geo, kappa, csw)
        return (apply_clover(A, psi)
                - kappa * dslash_wilson_full(u_t, psi, geo))

    # solve X = (M Mdag)^-1 phi with the fast path (kernels), grad with the
    # oracle path
    g = GaugeField(geo, "double", phi.device).from_complex(u)
    A = clover_matrix(u, geo, kappa, csw)
    cl = CloverField(geo, "double", phi.device).from_matrices(A)
    d = DiracClover(g, cl, kappa)
    X = d.new_spinor(n_parity=2)
    st = cg_solve(_NormalOp(d, mmdag=True), X, phi, tol=cg_tol,
                  maxiter=cg_maxiter)
    assert st.converged, "clover force CG failed"
    Yf = d.new_spinor(n_parity=2)
    d.M(Yf, X, dagger=True)
    S_f = blas.re_dot(phi, X)
    F = autograd_fermion_force(u, geo, apply_M, X.to_complex(),
                               Yf.to_complex())
    return S_f, F


def hisq_fermion_force(u: torch.Tensor, geo: LatticeGeometry, mass: float,
                       phi: SpinorField, coeffs=None, *,
                       cg_tol: float = 1e-10, cg_maxiter: int = 3000):
    """S_f and MD force for the improved-staggered (asqtad/HISQ-style)
    action: S_f = phi^dag (M Mdag)^{-1} phi with M = 2m + D[fat(U), naik(U)]
    (ref: lib/hisq_paths_force_quda.cu + kernels/hisq_paths_force.cuh —
    the one/three/five/seven-link + Lepage + Naik chain rule is obtained
    by backpropagating through the differentiable link fattening instead
    of the hand-derived path recursion; finite-difference validated)."""
    from ..fields.gauge import GaugeField
    from ..models import DiracImprovedStaggered
    from ..ops.reference import (dslash_staggered_naik_parity,
                                 dslash_staggered_parity)
    from .hisq import asqtad_coefficients, fat_links, naik_links

    c = coeffs or asqtad_coefficients()
    fat = fat_links(u, geo, c)
    lng = naik_links(u, geo)
    gf = GaugeField(geo, "double", phi.device).from_complex(fat)
    gl = GaugeField(geo, "double", phi.device, shift=3).from_complex(lng)
    d = DiracImprovedStaggered(gf, gl, mass)
    X = d.new_spinor(n_parity=2)
    st = cg_solve(_NormalOp(d, mmdag=True), X, phi, tol=cg_tol,
                  maxiter=cg_maxiter)
    assert st.converged, "hisq force CG failed"
    Y = d.new_spinor(n_parity=2)
    d.M(Y, X, dagger=True)
    S_f = blas.re_dot(phi, X)

    def apply_M(u_t, psi):
        f_t = fat_links(u_t, geo, c)
        n_t = naik_links(u_t, geo)
        out = 2.0 * mass * psi.clone()
        for p in (0, 1):
            out[p] = out[p] + dslash_staggered_parity(f_t, psi[1 - p], geo, p)
            out[p] = out[p] + dslash_staggered_naik_parity(n_t, psi[1 - p],
                                                           geo, p)
        return out

    F = autograd_fermion_force(u, geo, apply_M, X.to_complex(),
                               Y.to_complex())
    return S_f, F


def _oracle_cg(apply_A, b: torch.Tensor, tol: float, maxiter: int
               ) -> torch.Tensor:
    """Plain CG on a hermitian-PD complex-tensor operator."""
    x = torch.zeros_like(b)
    r = b.clone()
    p = r.clone()
    r2 = (r.conj() * r).sum().real
    b2 = r2.clone()
    for _ in range(maxiter):
        Ap = apply_A(p)
        alpha = r2 / (p.conj() * Ap).sum().real
        x = x + alpha * p
        r = r - alpha * Ap
        r2n = (r.conj() * r).sum().real
        if r2n < tol * tol * b2:
            break
        p = r + (r2n / r2) * p
        r2 = r2n
    return x


def hasenbusch_action_and_force(u: torch.Tensor, geo: LatticeGeometry,
                                kappa: float, mu_h: float,
                                phi1: torch.Tensor, phi2: torch.Tensor, *,
                                cg_tol: float = 1e-10,
                                cg_maxiter: int = 2000):
    """Hasenbusch mass-preconditioned two-flavor Wilson action
    (ref: the DiracCloverHasenbuschTwist consumers; W = M + i mu_h g5):

        S = phi1^d (W W^d)^-1 phi1  +  phi2^d W (M M^d)^-1 W^d phi2

    (det(MM^d) = det(WW^d) * det ratio). Forces by the adjoint trick:
    solve once with u detached, then differentiate the surrogate
    2 Re<w(u), y> - <y, A(u) y> whose u-gradient equals the true dS.
    Returns (S, F) with the validated F = 1/2 TA[U g^d] convention."""
    from ..ops.reference import dslash_wilson_full
    from .ops import project_ta

    def apply_M(ut, psi):
        return psi - kappa * dslash_wilson_full(ut, psi, geo)

    def apply_Mdag(ut, psi):
        g5psi = psi.clone()
        g5psi[..., 2:4, :] = -g5psi[..., 2:4, :]
        t = apply_M(ut, g5psi)
        t[..., 2:4, :] = -t[..., 2:4, :]
        return t

    def g5(psi):
        o = psi.clone()
        o[..., 2:4, :] = -o[..., 2:4, :]
        return o

    def apply_W(ut, psi):
        return apply_M(ut, psi) + 1j * mu_h * g5(psi)

    def apply_Wdag(ut, psi):
        return apply_Mdag(ut, psi) - 1j * mu_h * g5(psi)

    ud = u.detach()
    # solves at detached u
    X1 = _oracle_cg(lambda v: apply_W(ud, apply_Wdag(ud, v)), phi1,
                    cg_tol, cg_maxiter)
    w2 = apply_Wdag(ud, phi2)
    Y2 = _oracle_cg(lambda v: apply_M(ud, apply_Mdag(ud, v)), w2,
                    cg_tol, cg_maxiter)

    u_req = u.detach().clone().requires_grad_(True)
    # S1 surrogate: 2Re<phi1, X1> - <X1, W W^d X1>
    s1 = (2.0 * (phi1.conj() * X1).sum().real
          - (X1.conj() * apply_W(u_req, apply_Wdag(u_req, X1))).sum().real)
    # S2 surrogate: 2Re<W^d(u) phi2, Y2> - <Y2, M M^d Y2>
    s2 = (2.0 * (apply_Wdag(u_req, phi2).conj() * Y2).sum().real
          - (Y2.conj() * apply_M(u_req, apply_Mdag(u_req, Y2))).sum().real)
    s = s1 + s2
    s.backward()
    g = u_req.grad
    F = torch.empty_like(u)
    for mu in range(4):
        for p in (0, 1):
            F[mu, p] = 0.5 * project_ta(u[mu, p] @ g[mu, p].conj().mT)
    S = float((phi1.conj() * X1).sum().real
              + (w2.conj() * Y2).sum().real)
    return S, F


def hasenbusch_refresh(u: torch.Tensor, geo: LatticeGeometry, kappa: float,
                       mu_h: float, seed: int, *, cg_tol: float = 1e-10,
                       cg_maxiter: int = 2000):
    """(phi1, phi2) heatbath: phi1 = W eta1 and phi2 = W^-dag M eta2 make
    both action terms eta^d eta distributed."""
    from ..ops.reference import dslash_wilson_full
    gen = torch.Generator().manual_seed(seed)

    def gauss():
        return torch.view_as_complex(
            torch.randn((2, geo.volume_cb, 4, 3, 2), generator=gen,
                        dtype=torch.float64)) / (2.0 ** 0.5)

    def apply_M(psi):
        return psi - kappa * dslash_wilson_full(u, psi, geo)

    def g5(psi):
        o = psi.clone()
        o[..., 2:4, :] = -o[..., 2:4, :]
        return o

    def apply_Mdag(psi):
        return g5(apply_M(g5(psi)))

    def apply_W(psi):
        return apply_M(psi) + 1j * mu_h * g5(psi)

    def apply_Wdag(psi):
        return apply_Mdag(psi) - 1j * mu_h * g5(psi)

    eta1, eta2 = gauss(), gauss()
    phi1 = apply_W(eta1)
    b = apply_M(eta2)
    y = _oracle_cg(lambda v: apply_W(apply_Wdag(v)), apply_W(b),
                   cg_tol, cg_maxiter)
    # y = (W W^d)^-1 W b = W^-dag b
    return phi1, y


def _stag_pc_apply(ut: torch.Tensor, psi_e: torch.Tensor,
                   geo: LatticeGeometry, mass: float) -> torch.Tensor:
    """Hermitian-PD even-site staggered PC operator
    A = 4 m^2 - D_eo D_oe (torch-differentiable oracle)."""
    from ..ops.reference import dslash_staggered_parity
    t = dslash_staggered_parity(ut, psi_e, geo, 1)   # odd <- even
    s = dslash_staggered_parity(ut, t, geo, 0)       # even <- odd
    return 4.0 * mass * mass * psi_e - s


def rhmc_force(u: torch.Tensor, geo: LatticeGeometry, mass: float,
               phi_e: torch.Tensor, approx, *, cg_tol: float = 1e-10,
               cg_maxiter: int = 2000):
    """Exact rational-HMC force for S = phi^d r(A) phi with
    r(x) = r0 + sum_l res_l/(x + p_l) and A the staggered PC operator
    (ref: the MILC-side RHMC force the reference's multishift solver
    feeds): ONE multishift CG gives X_l = (A+p_l)^-1 phi; the force is
    the autograd of the surrogate -sum_l res_l <X_l, A(u) X_l>
    (adjoint trick; d/du of the constant terms vanishes). Returns (S, F)
    in the validated F = 1/2 TA[U g^d] convention."""
    from .ops import project_ta
    ud = u.detach()
    Xs = []
    # shifted solves, smallest shift first via plain per-shift CG on the
    # oracle (the production path uses solvers.multishift on fields;
    # force evaluation happens at oracle level for autograd)
    for p in approx.poles:
        Xs.append(_oracle_cg(
            lambda v, _p=p: _stag_pc_apply(ud, v, geo, mass) + _p * v,
            phi_e, cg_tol, cg_maxiter))
    S = approx.r0 * float((phi_e.conj() * phi_e).sum().real)
    for r, X in zip(approx.res, Xs):
        S += r * float((phi_e.conj() * X).sum().real)
    u_req = u.detach().clone().requires_grad_(True)
    s = None
    for r, X in zip(approx.res, Xs):
        term = -r * (X.conj() * _stag_pc_apply(u_req, X, geo, mass)
                     ).sum().real
        s = term if s is None else s + term
    s.backward()
    g = u_req.grad
    F = torch.empty_like(u)
    for mu in range(4):
        for p in (0, 1):
            F[mu, p] = 0.5 * project_ta(u[mu, p] @ g[mu, p].conj().mT)
    return S, F


def rhmc_refresh(u: torch.Tensor, geo: LatticeGeometry, mass: float,
                 approx_half, seed: int, *, cg_tol: float = 1e-10,
                 cg_maxiter: int = 2000) -> torch.Tensor:
    """phi = r_half(A) eta with r_half ~ x^{+alpha/2} so that
    S = phi^d r(A) phi (r ~ x^{-alpha}) is eta^d eta distributed."""
    gen = torch.Generator().manual_seed(seed)
    eta = torch.view_as_complex(
        torch.randn((geo.volume_cb, 3, 2), generator=gen,
                    dtype=torch.float64)) / (2.0 ** 0.5)
    phi = approx_half.r0 * eta
    for r, p in zip(approx_half.res, approx_half.poles):
        phi = phi + r * _oracle_cg(
            lambda v, _p=p: _stag_pc_apply(u, v, geo, mass) + _p * v,
            eta, cg_tol, cg_maxiter)
    return phi
