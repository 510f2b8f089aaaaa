"""Distance-preconditioned Wilson/clover (ref: dslash_wilson.cuh:96-101
t-hop weights w(t+-1)/w(t), spinor_reweight.cuh:29 cosh weight,
arXiv:1006.4028). Our implementation is the exact similarity transform
P^-1 M P; these tests verify it against the reference's per-hop weight
definition built independently from directional hops."""
import math

import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd import api
from quda_amd.api import DslashType, InvertParam, InverterType, SolutionType, GaugeParam
from quda_amd.models.distance import DiracWilsonDistance, DistanceWeight
from quda_amd.models import DiracWilson
from quda_amd.ops import blas
from quda_amd.ops import reference as ref

ALPHA0, T0, KAPPA = 0.35, 2, 0.12


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 4, 4, 8))
    g = GaugeField(geo, "double").random_su3_(seed=611)
    psi = SpinorField(geo, "double").gaussian_(seed=612)
    return geo, g, psi


def hop_weights(nt, alpha0, t0):
    def w(t):
        x = alpha0 * ((t - t0 + nt) % nt - nt // 2)
        return math.cosh(x) if alpha0 > 0 else 1.0 / math.cosh(x)
    return w


def test_matches_per_hop_weight_definition(setup):
    """M_dist psi == psi - kappa*(spatial hops + w-weighted t-hops),
    the reference's in-kernel definition (fwd t-hop * w(t+1)/w(t))."""
    from quda_amd.mg.coarse import _hop_lex
    from quda_amd.ops.reference import _gamma_tensors
    from quda_amd.fields.geometry import checkerboard_join, checkerboard_split
    geo, g, psi = setup
    d = DiracWilsonDistance(g, KAPPA, alpha0=ALPHA0, t0=T0)
    out = SpinorField(geo, "double")
    d.M(out, psi)
    got = checkerboard_join(out.to_complex(), geo)

    u_cb = g.to_complex()
    lo = geo.lex_of_cb
    u_lex = torch.empty((4, geo.volume, 3, 3), dtype=torch.complex128)
    u_lex[:, lo[0]] = u_cb[:, 0]
    u_lex[:, lo[1]] = u_cb[:, 1]
    psi_lex = checkerboard_join(psi.to_complex(), geo)
    P = _gamma_tensors("cpu", torch.complex128)
    w = hop_weights(geo.dims[3], ALPHA0, T0)
    t_of = geo.coords[:, 3].to(torch.int64)
    want = psi_lex.clone()
    for mu in range(4):
        for fwd in (True, False):
            h = _hop_lex(u_lex, geo, psi_lex, mu, fwd, False, P)
            if mu == 3:
                cf = torch.tensor(
                    [w((int(t) + (1 if fwd else -1)) % geo.dims[3])
                     / w(int(t)) for t in t_of], dtype=torch.float64)
                h = h * cf.reshape(-1, 1, 1)
            want -= KAPPA * h
    err = (got - want).abs().max().item()
    assert err < 1e-12, err


def test_invert_distance_full_and_pc(setup):
    """invertQuda with distance params returns the solution of the
    PHYSICAL system M x = b for both MAT and MATPC paths."""
    geo, g, psi = setup
    gp = GaugeParam(X=(4, 4, 4, 8), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.init_quda()
    api.load_gauge_quda(g.to_complex(), gp)
    b = psi.to_complex()
    for sol, inv in ((SolutionType.MAT, InverterType.CGNR),
                     (SolutionType.MATPC, InverterType.CG)):
        p = InvertParam(dslash_type=DslashType.WILSON, kappa=KAPPA,
                        inv_type=inv, solution_type=sol, tol=1e-10,
                        maxiter=600, distance_pc_alpha0=ALPHA0,
                        distance_pc_t0=T0)
        x = api.invert_quda(b, p)
        assert p.true_res < 1e-8, (sol, p.true_res)
        # independent check on the plain operator
        mx = ref.mat_wilson(g.to_complex(), x, geo, KAPPA)
        rel = (mx - b).abs().max().item() / b.abs().max().item()
        assert rel < 1e-7, (sol, rel)


def test_dagger_adjointness(setup):
    geo, g, psi = setup
    d = DiracWilsonDistance(g, KAPPA, alpha0=-0.25, t0=1)
    a = SpinorField(geo, "double").gaussian_(seed=613)
    b = SpinorField(geo, "double").gaussian_(seed=614)
    Ma = SpinorField(geo, "double")
    Mdb = SpinorField(geo, "double")
    d.M(Ma, a)
    d.M(Mdb, b, dagger=True)
    lhs = blas.c_dot(b, Ma)
    rhs = complex(blas.c_dot(Mdb, a))
    assert abs(complex(lhs) - rhs) < 1e-10 * abs(complex(lhs))
