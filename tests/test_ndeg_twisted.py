"""Non-degenerate twisted-mass doublet tests (ref coverage:
dslash_ndeg_twisted_mass*)."""
import math

import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.models import DiracNdegTwistedMass, DiracNdegTwistedMassPC
from quda_amd.ops import blas
from quda_amd.ops import reference as ref
from quda_amd.solvers import cgnr_solve

KAPPA, MU, EPS = 0.12, 0.07, 0.05


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=231)
    return geo, g


def doublet(geo, seed, n_parity=2):
    return SpinorField(geo, "double", n_parity=n_parity, ls=2).gaussian_(seed=seed)


def explicit_M(u, geo, psi2, dagger=False):
    """Oracle: per-flavor Wilson hops + i 2k mu g5 tau3 - 2k eps tau1."""
    V = geo.volume_cb
    sgn = -1 if dagger else 1
    out = torch.empty_like(psi2)
    a, b = 2 * KAPPA * MU, -2 * KAPPA * EPS
    for p in (0, 1):
        for f in (0, 1):
            sl = slice(f * V, (f + 1) * V)
            d4 = ref.dslash_wilson_parity(u, psi2[1 - p][sl], geo, p, dagger)
            tw = ref.apply_twist(psi2[p][sl], 0.0, sgn * a * (1 if f == 0 else -1))
            other = psi2[p][(1 - f) * V:(2 - f) * V]
            out[p][sl] = psi2[p][sl] + tw + b * other - KAPPA * d4
    return out


def test_ndeg_vs_explicit(setup):
    geo, g = setup
    d = DiracNdegTwistedMass(g, KAPPA, MU, EPS)
    psi = doublet(geo, 232)
    out = SpinorField(geo, "double", ls=2)
    for dagger in (False, True):
        d.M(out, psi, dagger=dagger)
        expect = explicit_M(g.to_complex(), geo, psi.to_complex(), dagger)
        err = (out.to_complex() - expect).abs().max().item()
        assert err < 1e-12, (dagger, err)


def test_ndeg_adjointness(setup):
    geo, g = setup
    d = DiracNdegTwistedMass(g, KAPPA, MU, EPS)
    a = doublet(geo, 233)
    b = doublet(geo, 234)
    Ma = SpinorField(geo, "double", ls=2)
    Mdb = SpinorField(geo, "double", ls=2)
    d.M(Ma, a)
    d.M(Mdb, b, dagger=True)
    lhs = (b.to_complex().conj() * Ma.to_complex()).sum()
    rhs = (Mdb.to_complex().conj() * a.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)


def test_ndeg_Ainv(setup):
    geo, g = setup
    d = DiracNdegTwistedMassPC(g, KAPPA, MU, EPS)
    psi = doublet(geo, 235, n_parity=1)
    t = SpinorField(geo, "double", n_parity=1, ls=2)
    u = SpinorField(geo, "double", n_parity=1, ls=2)
    d._apply_A(t, psi)
    d._apply_Ainv(u, t)
    err = (u.to_complex() - psi.to_complex()).abs().max().item()
    assert err < 1e-12, err


def test_ndeg_pc_solve(setup):
    geo, g = setup
    pc = DiracNdegTwistedMassPC(g, KAPPA, MU, EPS)
    b = doublet(geo, 236, n_parity=1)
    x = SpinorField(geo, "double", n_parity=1, ls=2)
    st = cgnr_solve(pc, x, b, tol=1e-10, maxiter=1000)
    assert st.converged
    r = SpinorField(geo, "double", n_parity=1, ls=2)
    pc.M(r, x)
    tr = math.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b))
    assert tr < 1e-8


@pytest.mark.gpu
def test_ndeg_gpu_vs_cpu(setup):
    geo, _ = setup
    gen = torch.Generator().manual_seed(237)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m))
    g_c = GaugeField(geo, "double").from_complex(u)
    g_g = GaugeField(geo, "double", "cuda").from_complex(u.cuda())
    psi = SpinorField(geo, "double", ls=2).gaussian_(seed=238)
    psi_g = SpinorField(geo, "double", "cuda", ls=2)
    psi_g.from_complex(psi.to_complex().cuda())
    for dagger in (False, True):
        oc = SpinorField(geo, "double", ls=2)
        DiracNdegTwistedMass(g_c, KAPPA, MU, EPS).M(oc, psi, dagger=dagger)
        og = SpinorField(geo, "double", "cuda", ls=2)
        DiracNdegTwistedMass(g_g, KAPPA, MU, EPS).M(og, psi_g, dagger=dagger)
        err = (og.to_complex().cpu() - oc.to_complex()).abs().max().item()
        assert err < 1e-12, (dagger, err)


# ---------------------------------------------------------------------------
# non-degenerate twisted CLOVER doublet
# ---------------------------------------------------------------------------

def _ntc_setup():
    from quda_amd.fields.clover import CloverField
    from quda_amd.ops.reference import clover_matrix
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=641)
    u = g.to_complex()
    kappa, csw = 0.11, 1.3
    A = clover_matrix(u, geo, kappa, csw)
    cl = CloverField(geo, "double").from_matrices(A)
    return geo, g, cl, kappa


def test_ntc_ainv_inverts_a():
    from quda_amd.models import DiracNdegTwistedClover
    geo, g, cl, kappa = _ntc_setup()
    d = DiracNdegTwistedClover(g, cl, kappa, mu=0.3, epsilon=0.15)
    psi = SpinorField(geo, "double", n_parity=1, ls=2).gaussian_(seed=642)
    t = SpinorField(geo, "double", n_parity=1, ls=2)
    b = SpinorField(geo, "double", n_parity=1, ls=2)
    for dag in (False, True):
        for p in (0, 1):
            d._apply_A_p(t, psi, p, dagger=dag)
            d._apply_Ainv_p(b, t, p, dagger=dag)
            err = (b.to_complex() - psi.to_complex()).abs().max().item()
            assert err < 1e-11, (dag, p, err)


def test_ntc_eps_zero_reduces_to_twisted_clover_pair():
    """epsilon=0 decouples the flavors: slice 0 = TwistedClover(+mu),
    slice 1 = TwistedClover(-mu)."""
    from quda_amd.models import DiracNdegTwistedClover, DiracTwistedClover
    geo, g, cl, kappa = _ntc_setup()
    mu = 0.25
    d = DiracNdegTwistedClover(g, cl, kappa, mu=mu, epsilon=0.0)
    psi = SpinorField(geo, "double", n_parity=2, ls=2).gaussian_(seed=643)
    out = SpinorField(geo, "double", n_parity=2, ls=2)
    d.M(out, psi)
    V = geo.volume_cb
    pc = psi.to_complex().reshape(2, 2, V, 4, 3)
    oc = out.to_complex().reshape(2, 2, V, 4, 3)
    for s, sgn in ((0, +1), (1, -1)):
        dtc = DiracTwistedClover(g, cl, kappa, sgn * mu)
        ps = SpinorField(geo, "double", n_parity=2)
        ps.from_complex(pc[:, s].contiguous())
        os_ = SpinorField(geo, "double", n_parity=2)
        dtc.M(os_, ps)
        err = (os_.to_complex() - oc[:, s]).abs().max().item()
        assert err < 1e-11, (s, err)


def test_ntc_adjoint():
    from quda_amd.models import DiracNdegTwistedClover
    geo, g, cl, kappa = _ntc_setup()
    d = DiracNdegTwistedClover(g, cl, kappa, mu=0.3, epsilon=0.15)
    a = SpinorField(geo, "double", n_parity=2, ls=2).gaussian_(seed=644)
    b = SpinorField(geo, "double", n_parity=2, ls=2).gaussian_(seed=645)
    Ma = SpinorField(geo, "double", n_parity=2, ls=2)
    Mdb = SpinorField(geo, "double", n_parity=2, ls=2)
    d.M(Ma, a)
    d.M(Mdb, b, dagger=True)
    lhs = (b.to_complex().conj() * Ma.to_complex()).sum()
    rhs = (Mdb.to_complex().conj() * a.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)


def test_ntc_pc_vs_full_solve():
    from quda_amd.models import (DiracNdegTwistedClover,
                                 DiracNdegTwistedCloverPC)
    from quda_amd.solvers import cgnr_solve
    geo, g, cl, kappa = _ntc_setup()
    full = DiracNdegTwistedClover(g, cl, kappa, mu=0.3, epsilon=0.15)
    pc = DiracNdegTwistedCloverPC(g, cl, kappa, mu=0.3, epsilon=0.15)
    b = SpinorField(geo, "double", n_parity=2, ls=2).gaussian_(seed=646)
    x_full = SpinorField(geo, "double", n_parity=2, ls=2)
    st = cgnr_solve(full, x_full, b, tol=1e-10, maxiter=3000)
    assert st.converged
    # PC: solve on even from b' = Ainv(b_e + kappa D_eo Ainv b_o)
    import math
    from quda_amd.ops import blas
    from quda_amd.ops.dispatch import dslash_wilson_slice, dwf_halo_exchange
    be = SpinorField(geo, "double", n_parity=1, ls=2)
    t = SpinorField(geo, "double", n_parity=1, ls=2)
    pc._apply_Ainv_p(t, b.parity_view(1), 1)
    h = dwf_halo_exchange(t, 1, False)
    u = SpinorField(geo, "double", n_parity=1, ls=2)
    blas.copy(u, b.parity_view(0))
    for s in (0, 1):
        dslash_wilson_slice(u, t, g, 0, s, False, a=kappa, x=u, halo=h)
    pc._apply_Ainv_p(be, u, 0)
    xe = SpinorField(geo, "double", n_parity=1, ls=2)
    st2 = cgnr_solve(pc, xe, be, tol=1e-11, maxiter=3000)
    assert st2.converged
    err = (xe.to_complex() - x_full.to_complex()[0]).abs().max().item()
    assert err < 1e-6, err


@pytest.mark.gpu
def test_ntc_gpu_matches_cpu():
    """Doublet twisted-clover apply on device (sliced clover kernel path)
    matches the CPU oracle composition."""
    from quda_amd.fields.clover import CloverField
    from quda_amd.models import DiracNdegTwistedClover
    from quda_amd.ops.reference import clover_matrix
    geo = LatticeGeometry((4, 4, 4, 4))
    gen = torch.Generator().manual_seed(651)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m))
    kappa, csw = 0.11, 1.3
    A = clover_matrix(u, geo, kappa, csw)
    g_cpu = GaugeField(geo, "double").from_complex(u)
    cl_cpu = CloverField(geo, "double").from_matrices(A)
    d_cpu = DiracNdegTwistedClover(g_cpu, cl_cpu, kappa, 0.3, 0.15)
    g_gpu = GaugeField(geo, "double", "cuda").from_complex(u.cuda())
    cl_gpu = CloverField(geo, "double", "cuda").from_matrices(A.cuda())
    d_gpu = DiracNdegTwistedClover(g_gpu, cl_gpu, kappa, 0.3, 0.15)
    psi = SpinorField(geo, "double", n_parity=2, ls=2).gaussian_(seed=652)
    psi_g = SpinorField(geo, "double", "cuda", n_parity=2, ls=2)
    psi_g.from_complex(psi.to_complex().cuda())
    for dag in (False, True):
        oc = SpinorField(geo, "double", n_parity=2, ls=2)
        og = SpinorField(geo, "double", "cuda", n_parity=2, ls=2)
        d_cpu.M(oc, psi, dagger=dag)
        d_gpu.M(og, psi_g, dagger=dag)
        err = (og.to_complex().cpu() - oc.to_complex()).abs().max().item()
        assert err < 1e-11, (dag, err)
