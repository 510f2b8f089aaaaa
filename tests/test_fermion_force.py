"""Pseudofermion force tests: finite-difference consistency + fermionic
HMC energy conservation (role of the reference's clover_force_test)."""
import math

import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.gauge import gauge_action, mom_action, random_momentum
from quda_amd.gauge.fermion_force import (fermion_action_and_force,
                                          hmc_trajectory_2f,
                                          pseudofermion_refresh)
from quda_amd.gauge.ops import _from_lex, _to_lex
from quda_amd.models import DiracWilson

KAPPA, BETA = 0.115, 5.5


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 4, 4, 4))
    from quda_amd.fields.gauge import project_su3
    gen = torch.Generator().manual_seed(161)
    eye = torch.eye(3, dtype=torch.complex128)
    m = eye + 0.3 * torch.view_as_complex(
        torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64))
    u = project_su3(m)
    g = GaugeField(geo, "double").from_complex(u)
    phi = pseudofermion_refresh(DiracWilson(g, KAPPA), seed=162)
    return geo, u, phi


def test_force_matches_finite_difference(setup):
    """dS_f/dt along Udot = P U must equal -2 tr(P F_f)."""
    geo, u, phi = setup
    P = random_momentum(geo, seed=163)
    eps = 1e-6
    U = _to_lex(u, geo)
    Pl = _to_lex(P, geo)
    up = _from_lex(torch.matrix_exp(eps * Pl) @ U, geo)
    um = _from_lex(torch.matrix_exp(-eps * Pl) @ U, geo)
    Sp, _ = fermion_action_and_force(up, geo, KAPPA, phi, cg_tol=1e-12)
    Sm, _ = fermion_action_and_force(um, geo, KAPPA, phi, cg_tol=1e-12)
    dSdt = (Sp - Sm) / (2 * eps)
    _, F = fermion_action_and_force(u, geo, KAPPA, phi, cg_tol=1e-12)
    trPF = torch.einsum("dpvij,dpvji->", P, F).real.item()
    # conservation: d/dt(-tr P^2) = -2 tr(P F) must cancel dS/dt
    assert abs(-2 * trPF + dSdt) < 1e-4 * max(abs(dSdt), 1.0), \
        (dSdt, -2 * trPF)


def test_hmc_2f_energy_conservation(setup):
    geo, u, phi = setup
    dHs = []
    for n in (8, 16):
        _, _, dH = hmc_trajectory_2f(u, geo, BETA, KAPPA, n_md=n, tau=0.25,
                                     seed=164, cg_tol=1e-11)
        dHs.append(abs(dH))
    assert dHs[1] < dHs[0]  # second-order integrator
    assert dHs[1] < 1.0


def test_autograd_force_matches_analytic_wilson(setup):
    """Autograd force == the hand-derived Wilson force."""
    from quda_amd.gauge.fermion_force import (autograd_fermion_force,
                                              fermion_action_and_force,
                                              wilson_fermion_force)
    from quda_amd.ops.reference import mat_wilson
    from quda_amd.solvers.variants import _NormalOp
    from quda_amd.solvers import cg_solve
    from quda_amd import GaugeField, SpinorField
    geo, u, phi = setup
    g = GaugeField(geo, "double").from_complex(u)
    d = DiracWilson(g, KAPPA)
    X = d.new_spinor(n_parity=2)
    cg_solve(_NormalOp(d, mmdag=True), X, phi, tol=1e-12, maxiter=2000)
    Y = d.new_spinor(n_parity=2)
    d.M(Y, X, dagger=True)
    F_analytic = wilson_fermion_force(u, geo, KAPPA, X.to_complex(),
                                      Y.to_complex())

    def apply_M(u_t, psi):
        return mat_wilson(u_t, psi, geo, KAPPA)

    F_auto = autograd_fermion_force(u, geo, apply_M, X.to_complex(),
                                    Y.to_complex())
    err = (F_auto - F_analytic).abs().max().item()
    scale = F_analytic.abs().max().item()
    assert err < 1e-10 * max(scale, 1.0), (err, scale)


def test_clover_force_finite_difference(setup):
    """Clover-term force (autograd through the field-strength chain) vs
    finite differences — the computeCloverForceQuda check."""
    from quda_amd.gauge.fermion_force import clover_fermion_force
    geo, u, _ = setup
    from quda_amd.fields.clover import CloverField
    from quda_amd.models import DiracClover
    from quda_amd.ops.reference import clover_matrix
    from quda_amd import GaugeField, SpinorField
    csw = 1.0
    # fresh pseudofermion under the clover operator
    g = GaugeField(geo, "double").from_complex(u)
    A = clover_matrix(u, geo, KAPPA, csw)
    cl = CloverField(geo, "double").from_matrices(A)
    d = DiracClover(g, cl, KAPPA)
    eta = d.new_spinor(n_parity=2)
    eta.gaussian_(seed=171)
    phi = d.new_spinor(n_parity=2)
    d.M(phi, eta)

    P = random_momentum(geo, seed=172)
    eps = 1e-6
    U = _to_lex(u, geo)
    Pl = _to_lex(P, geo)
    up = _from_lex(torch.matrix_exp(eps * Pl) @ U, geo)
    um = _from_lex(torch.matrix_exp(-eps * Pl) @ U, geo)
    Sp, _ = clover_fermion_force(up, geo, KAPPA, csw, phi, cg_tol=1e-12)
    Sm, _ = clover_fermion_force(um, geo, KAPPA, csw, phi, cg_tol=1e-12)
    dSdt = (Sp - Sm) / (2 * eps)
    _, F = clover_fermion_force(u, geo, KAPPA, csw, phi, cg_tol=1e-12)
    trPF = torch.einsum("dpvij,dpvji->", P, F).real.item()
    assert abs(-2 * trPF + dSdt) < 1e-4 * max(abs(dSdt), 1.0), \
        (dSdt, -2 * trPF)


def test_hisq_force_finite_difference():
    """The hisq_paths_force check: backprop through fat7+Lepage+Naik
    matches finite differences of the improved-staggered action."""
    from quda_amd.gauge.fermion_force import hisq_fermion_force
    from quda_amd.gauge.hisq import asqtad_coefficients, fat_links, naik_links
    from quda_amd.fields.gauge import GaugeField, project_su3
    from quda_amd.models import DiracImprovedStaggered
    from quda_amd import SpinorField
    geo = LatticeGeometry((4, 4, 4, 4))
    gen = torch.Generator().manual_seed(176)
    eye = torch.eye(3, dtype=torch.complex128)
    m = eye + 0.3 * torch.view_as_complex(
        torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64))
    u = project_su3(m)
    mass = 0.2
    # pseudofermion under the improved operator
    c = asqtad_coefficients()
    gf = GaugeField(geo, "double").from_complex(fat_links(u, geo, c))
    gl = GaugeField(geo, "double", shift=3).from_complex(naik_links(u, geo))
    d = DiracImprovedStaggered(gf, gl, mass)
    eta = d.new_spinor(n_parity=2)
    eta.gaussian_(seed=177)
    phi = d.new_spinor(n_parity=2)
    d.M(phi, eta)

    P = random_momentum(geo, seed=178)
    eps = 1e-6
    U = _to_lex(u, geo)
    Pl = _to_lex(P, geo)
    up = _from_lex(torch.matrix_exp(eps * Pl) @ U, geo)
    um = _from_lex(torch.matrix_exp(-eps * Pl) @ U, geo)
    Sp, _ = hisq_fermion_force(up, geo, mass, phi, cg_tol=1e-12)
    Sm, _ = hisq_fermion_force(um, geo, mass, phi, cg_tol=1e-12)
    dSdt = (Sp - Sm) / (2 * eps)
    _, F = hisq_fermion_force(u, geo, mass, phi, cg_tol=1e-12)
    trPF = torch.einsum("dpvij,dpvji->", P, F).real.item()
    assert abs(-2 * trPF + dSdt) < 1e-4 * max(abs(dSdt), 1.0), \
        (dSdt, -2 * trPF)


def test_hasenbusch_force_finite_difference(setup):
    """Mass-preconditioned (Hasenbusch) two-term action: combined force
    matches finite differences."""
    from quda_amd.gauge.fermion_force import (hasenbusch_action_and_force,
                                              hasenbusch_refresh)
    geo, u, _ = setup
    mu_h = 0.15
    phi1, phi2 = hasenbusch_refresh(u, geo, KAPPA, mu_h, seed=166)
    P = random_momentum(geo, seed=167)
    eps = 1e-6
    U = _to_lex(u, geo)
    Pl = _to_lex(P, geo)
    up = _from_lex(torch.matrix_exp(eps * Pl) @ U, geo)
    um = _from_lex(torch.matrix_exp(-eps * Pl) @ U, geo)
    Sp, _ = hasenbusch_action_and_force(up, geo, KAPPA, mu_h, phi1, phi2,
                                        cg_tol=1e-12)
    Sm, _ = hasenbusch_action_and_force(um, geo, KAPPA, mu_h, phi1, phi2,
                                        cg_tol=1e-12)
    dSdt = (Sp - Sm) / (2 * eps)
    _, F = hasenbusch_action_and_force(u, geo, KAPPA, mu_h, phi1, phi2,
                                       cg_tol=1e-12)
    trPF = torch.einsum("dpvij,dpvji->", P, F).real.item()
    assert abs(-2 * trPF + dSdt) < 1e-4 * max(abs(dSdt), 1.0), \
        (dSdt, -2 * trPF)


def test_hasenbusch_hmc_energy_conservation(setup):
    """Leapfrog with gauge + both Hasenbusch terms conserves H at
    O(dt^2)."""
    from quda_amd.gauge import gauge_action, gauge_force, leapfrog
    from quda_amd.gauge.fermion_force import (hasenbusch_action_and_force,
                                              hasenbusch_refresh)
    from quda_amd.gauge.hmc import _evolve_u, mom_action
    geo, u, _ = setup
    mu_h = 0.2
    phi1, phi2 = hasenbusch_refresh(u, geo, KAPPA, mu_h, seed=168)

    def total_force(uc):
        _, Ff = hasenbusch_action_and_force(uc, geo, KAPPA, mu_h, phi1,
                                            phi2, cg_tol=1e-11)
        return gauge_force(uc, geo, BETA) + Ff

    def H(uc, Pc):
        Sf, _ = hasenbusch_action_and_force(uc, geo, KAPPA, mu_h, phi1,
                                            phi2, cg_tol=1e-11)
        return mom_action(Pc) + gauge_action(uc, geo, BETA) + Sf

    P0 = random_momentum(geo, seed=169)
    dHs = []
    for n in (8, 16):
        dt = 0.25 / n
        uu, PP = u.clone(), P0 + 0.5 * dt * total_force(u)
        for k in range(n):
            uu = _evolve_u(uu, PP, geo, dt)
            PP = PP + (0.5 if k == n - 1 else 1.0) * dt * total_force(uu)
        dHs.append(abs(H(uu, PP) - H(u, P0)))
    assert dHs[1] < dHs[0]
    assert dHs[1] < 1.0, dHs


def test_rhmc_force_finite_difference():
    """Rational-HMC (rooted staggered) force from one multishift solve +
    adjoint surrogate matches finite differences."""
    from quda_amd.gauge.fermion_force import rhmc_force, rhmc_refresh
    from quda_amd.solvers import rational_approx
    geo = LatticeGeometry((4, 4, 4, 4))
    u = GaugeField(geo, "double").random_su3_(seed=81).to_complex()
    mass = 0.1
    lo, hi = 4 * mass * mass, 30.0
    r_inv = rational_approx(-0.25, lo, hi, n=10)
    r_half = rational_approx(0.125, lo, hi, n=10)
    phi = rhmc_refresh(u, geo, mass, r_half, seed=33)
    P = random_momentum(geo, seed=34)
    eps = 1e-6
    U = _to_lex(u, geo)
    Pl = _to_lex(P, geo)
    up = _from_lex(torch.matrix_exp(eps * Pl) @ U, geo)
    um = _from_lex(torch.matrix_exp(-eps * Pl) @ U, geo)
    Sp, _ = rhmc_force(up, geo, mass, phi, r_inv, cg_tol=1e-12)
    Sm, _ = rhmc_force(um, geo, mass, phi, r_inv, cg_tol=1e-12)
    dSdt = (Sp - Sm) / (2 * eps)
    _, F = rhmc_force(u, geo, mass, phi, r_inv, cg_tol=1e-12)
    trPF = torch.einsum("dpvij,dpvji->", P, F).real.item()
    assert abs(-2 * trPF + dSdt) < 1e-5 * max(abs(dSdt), 1.0), \
        (dSdt, -2 * trPF)


def test_rhmc_energy_conservation():
    """One-flavor-rooted staggered RHMC leapfrog: dH shrinks O(dt^2)."""
    from quda_amd.gauge import gauge_action, gauge_force
    from quda_amd.gauge.fermion_force import rhmc_force, rhmc_refresh
    from quda_amd.gauge.hmc import _evolve_u, mom_action
    from quda_amd.solvers import rational_approx
    geo = LatticeGeometry((4, 4, 4, 4))
    u = GaugeField(geo, "double").random_su3_(seed=82).to_complex()
    mass, beta = 0.1, 5.5
    lo, hi = 4 * mass * mass, 30.0
    r_inv = rational_approx(-0.25, lo, hi, n=10)
    r_half = rational_approx(0.125, lo, hi, n=10)
    phi = rhmc_refresh(u, geo, mass, r_half, seed=35)

    def force(uc):
        _, Ff = rhmc_force(uc, geo, mass, phi, r_inv, cg_tol=1e-11)
        return gauge_force(uc, geo, beta) + Ff

    def H(uc, Pc):
        Sf, _ = rhmc_force(uc, geo, mass, phi, r_inv, cg_tol=1e-11)
        return mom_action(Pc) + gauge_action(uc, geo, beta) + Sf

    P0 = random_momentum(geo, seed=36)
    dHs = []
    for n in (8, 16):
        dt = 0.2 / n
        uu, PP = u.clone(), P0 + 0.5 * dt * force(u)
        for k in range(n):
            uu = _evolve_u(uu, PP, geo, dt)
            PP = PP + (0.5 if k == n - 1 else 1.0) * dt * force(uu)
        dHs.append(abs(H(uu, PP) - H(u, P0)))
    assert dHs[1] < dHs[0]
    assert dHs[1] < 1.0, dHs


def test_hisq_full_force_through_unitarization():
    """HISQ force with the SVD unitarization inside the differentiated
    chain (unitarize_force role) matches finite differences."""
    from quda_amd.gauge.hisq import hisq_full_force, hisq_two_level_links
    from quda_amd.models import DiracImprovedStaggered
    from quda_amd.fields.gauge import GaugeField
    from quda_amd import SpinorField
    geo = LatticeGeometry((4, 4, 4, 4))
    u = GaugeField(geo, "double").random_su3_(seed=83).to_complex()
    mass = 0.2
    # pseudofermion refresh: phi = M eta on the two-level links
    fat, lng = hisq_two_level_links(u, geo)
    gf = GaugeField(geo, "double").from_complex(fat)
    gl = GaugeField(geo, "double", shift=3).from_complex(lng)
    d = DiracImprovedStaggered(gf, gl, mass)
    eta = SpinorField(geo, "double", nspin=1).gaussian_(seed=84)
    phi = SpinorField(geo, "double", nspin=1)
    d.M(phi, eta)
    P = random_momentum(geo, seed=85)
    eps = 1e-6
    U = _to_lex(u, geo)
    Pl = _to_lex(P, geo)
    up = _from_lex(torch.matrix_exp(eps * Pl) @ U, geo)
    um = _from_lex(torch.matrix_exp(-eps * Pl) @ U, geo)
    Sp, _ = hisq_full_force(up, geo, mass, phi, cg_tol=1e-12)
    Sm, _ = hisq_full_force(um, geo, mass, phi, cg_tol=1e-12)
    dSdt = (Sp - Sm) / (2 * eps)
    _, F = hisq_full_force(u, geo, mass, phi, cg_tol=1e-12)
    trPF = torch.einsum("dpvij,dpvji->", P, F).real.item()
    assert abs(-2 * trPF + dSdt) < 1e-4 * max(abs(dSdt), 1.0), \
        (dSdt, -2 * trPF)
