"""Twisted-mass operator tests (role of the reference's twisted-mass
dslash/invert test coverage; ref lib/dirac_twisted_mass.cpp semantics)."""
import math

import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.fields.clover import CloverField
from quda_amd.models import (DiracTwistedClover, DiracTwistedMass,
                             DiracTwistedMassPC)
from quda_amd.ops import blas
from quda_amd.ops import reference as ref
from quda_amd.solvers import cg_solve

KAPPA, MU = 0.12, 0.08


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=23)
    return geo, g


def test_tm_oracle_consistency(setup):
    """M psi == (1 + i eps g5) psi - kappa D psi via oracle pieces."""
    geo, g = setup
    d = DiracTwistedMass(g, KAPPA, MU)
    psi = SpinorField(geo, "double").gaussian_(seed=24)
    out = SpinorField(geo, "double")
    d.M(out, psi)
    u = g.to_complex()
    pc = psi.to_complex()
    expect = (ref.apply_twist(pc, 1.0, 2 * KAPPA * MU)
              - KAPPA * ref.dslash_wilson_full(u, pc, geo))
    err = (out.to_complex() - expect).abs().max().item()
    assert err < 1e-12


def test_tm_gamma5_hermiticity(setup):
    """g5 M(mu) g5 = M(mu)^dag (twisted basis relation)."""
    geo, g = setup
    d = DiracTwistedMass(g, KAPPA, MU)
    psi = SpinorField(geo, "double").gaussian_(seed=25)
    chi = SpinorField(geo, "double").gaussian_(seed=26)
    Mpsi = SpinorField(geo, "double")
    Mdchi = SpinorField(geo, "double")
    d.M(Mpsi, psi)
    d.M(Mdchi, chi, dagger=True)
    lhs = (chi.to_complex().conj() * Mpsi.to_complex()).sum()
    rhs = (Mdchi.to_complex().conj() * psi.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)


def test_tm_pc_prepare_reconstruct_matches_full(setup):
    """PC solve + reconstruct == full-operator solve (ref solve-type
    matrix, lib/solve.cpp)."""
    geo, g = setup
    full = DiracTwistedMass(g, KAPPA, MU)
    pc = DiracTwistedMassPC(g, KAPPA, MU)
    b = SpinorField(geo, "double").gaussian_(seed=27)
    # full solve via CGNR on M
    x_full = SpinorField(geo, "double")
    from quda_amd.solvers import cgnr_solve
    st = cgnr_solve(full, x_full, b, tol=1e-10, maxiter=800)
    assert st.converged
    # PC solve
    be = pc.prepare(b)
    xe = SpinorField(geo, "double", n_parity=1)
    from quda_amd.solvers import cgnr_solve as s2
    st2 = s2(pc, xe, be, tol=1e-11, maxiter=800)
    assert st2.converged
    x_rec = SpinorField(geo, "double")
    pc.reconstruct(x_rec, xe, b)
    err = (x_rec.to_complex() - x_full.to_complex()).abs().max().item()
    assert err < 1e-7


def test_tm_pc_cg(setup):
    geo, g = setup
    pc = DiracTwistedMassPC(g, KAPPA, MU)
    b = SpinorField(geo, "double", n_parity=1).gaussian_(seed=28)
    x = SpinorField(geo, "double", n_parity=1)
    st = cg_solve(pc, x, b, tol=1e-10, maxiter=400)
    assert st.converged


def test_twisted_clover_oracle(setup):
    geo, g = setup
    u = g.to_complex()
    A = ref.clover_matrix(u, geo, KAPPA, 1.0)
    cl = CloverField(geo, "double").from_matrices(A)
    d = DiracTwistedClover(g, cl, KAPPA, MU)
    psi = SpinorField(geo, "double").gaussian_(seed=29)
    out = SpinorField(geo, "double")
    d.M(out, psi)
    pc = psi.to_complex()
    expect = (ref.apply_clover(A, pc)
              + ref.apply_twist(pc, 0.0, 2 * KAPPA * MU)
              - KAPPA * ref.dslash_wilson_full(u, pc, geo))
    err = (out.to_complex() - expect).abs().max().item()
    assert err < 1e-12


def test_twisted_clover_cgnr(setup):
    geo, g = setup
    u = g.to_complex()
    A = ref.clover_matrix(u, geo, KAPPA, 1.0)
    cl = CloverField(geo, "double").from_matrices(A)
    d = DiracTwistedClover(g, cl, KAPPA, MU)
    b = SpinorField(geo, "double").gaussian_(seed=30)
    x = SpinorField(geo, "double")
    from quda_amd.solvers import cgnr_solve
    st = cgnr_solve(d, x, b, tol=1e-10, maxiter=800)
    assert st.converged
    r = SpinorField(geo, "double")
    d.M(r, x)
    tr = math.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b))
    assert tr < 1e-7


@pytest.mark.gpu
@pytest.mark.parametrize("prec", ["double", "single", "half"])
def test_tm_gpu_vs_oracle(setup, prec):
    geo, _ = setup
    gen = torch.Generator().manual_seed(51)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m)).cuda()
    g = GaugeField(geo, prec, "cuda").from_complex(u)
    d = DiracTwistedMass(g, KAPPA, MU)
    psi = SpinorField(geo, prec, "cuda").gaussian_(seed=52)
    out = SpinorField(geo, prec, "cuda")
    for dagger in (False, True):
        d.M(out, psi, dagger=dagger)
        sgn = -1 if dagger else 1
        pc = psi.to_complex()
        expect = (ref.apply_twist(pc, 1.0, sgn * 2 * KAPPA * MU)
                  - KAPPA * ref.dslash_wilson_full(u, pc, geo, dagger=dagger))
        err = (out.to_complex() - expect).abs().max().item()
        tol = {"double": 1e-12, "single": 1e-5, "half": 5e-3}[prec]
        assert err < tol, f"{prec} dag={dagger}: {err}"


@pytest.mark.gpu
def test_tm_pc_cg_gpu(setup):
    geo, _ = setup
    geo = LatticeGeometry((8, 8, 8, 8))
    gen = torch.Generator().manual_seed(53)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m)).cuda()
    g = GaugeField(geo, "double", "cuda").from_complex(u)
    pc = DiracTwistedMassPC(g, KAPPA, MU)
    b = SpinorField(geo, "double", "cuda", n_parity=1).gaussian_(seed=54)
    x = SpinorField(geo, "double", "cuda", n_parity=1)
    st = cg_solve(pc, x, b, tol=1e-9, maxiter=400)
    assert st.converged


def test_hasenbusch_twist(setup):
    """M' = M_clover + i mu g5; adjointness + CGNR solve."""
    from quda_amd.models import (DiracCloverHasenbuschTwist,
                                 DiracCloverHasenbuschTwistPC)
    geo, g = setup
    u = g.to_complex()
    A = ref.clover_matrix(u, geo, KAPPA, 1.0)
    cl = CloverField(geo, "double").from_matrices(A)
    d = DiracCloverHasenbuschTwist(g, cl, KAPPA, 0.05)
    a = SpinorField(geo, "double").gaussian_(seed=251)
    b = SpinorField(geo, "double").gaussian_(seed=252)
    Ma = SpinorField(geo, "double")
    Mdb = SpinorField(geo, "double")
    d.M(Ma, a)
    d.M(Mdb, b, dagger=True)
    lhs = (b.to_complex().conj() * Ma.to_complex()).sum()
    rhs = (Mdb.to_complex().conj() * a.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)
    # explicit check: M' psi = M psi + i mu g5 psi
    from quda_amd.models import DiracClover
    M0 = SpinorField(geo, "double")
    DiracClover(g, cl, KAPPA).M(M0, a)
    expect = M0.to_complex() + ref.apply_twist(a.to_complex(), 0.0, 0.05)
    assert (Ma.to_complex() - expect).abs().max().item() < 1e-12
    # PC solve
    pc = DiracCloverHasenbuschTwistPC(g, cl, KAPPA, 0.05)
    be = SpinorField(geo, "double", n_parity=1).gaussian_(seed=253)
    x = SpinorField(geo, "double", n_parity=1)
    from quda_amd.solvers import cgnr_solve
    st = cgnr_solve(pc, x, be, tol=1e-10, maxiter=600)
    assert st.converged


def test_twisted_clover_pc_vs_full(setup):
    """PC prepare/solve/reconstruct == full twisted-clover solve."""
    from quda_amd.models import DiracTwistedCloverPC
    from quda_amd.solvers import cgnr_solve
    geo, g = setup
    u = g.to_complex()
    A = ref.clover_matrix(u, geo, KAPPA, 1.0)
    cl = CloverField(geo, "double").from_matrices(A)
    full = DiracTwistedClover(g, cl, KAPPA, MU)
    pc = DiracTwistedCloverPC(g, cl, KAPPA, MU)
    # Atc^-1 inverts Atc = A + i eps g5
    psi = SpinorField(geo, "double", n_parity=1).gaussian_(seed=260)
    t = SpinorField(geo, "double", n_parity=1)
    pc._apply_Atc_inv(t, psi, 0)
    back = (ref.apply_clover(A[0], t.to_complex()[0])
            + ref.apply_twist(t.to_complex()[0], 0.0, 2 * KAPPA * MU))
    assert (back - psi.to_complex()[0]).abs().max().item() < 1e-10
    b = SpinorField(geo, "double").gaussian_(seed=261)
    x_full = SpinorField(geo, "double")
    st = cgnr_solve(full, x_full, b, tol=1e-11, maxiter=1500)
    assert st.converged
    be = pc.prepare(b)
    xe = SpinorField(geo, "double", n_parity=1)
    st2 = cgnr_solve(pc, xe, be, tol=1e-12, maxiter=1500)
    assert st2.converged
    x_rec = SpinorField(geo, "double")
    pc.reconstruct(x_rec, xe, b)
    err = (x_rec.to_complex() - x_full.to_complex()).abs().max().item()
    assert err < 1e-6, err
