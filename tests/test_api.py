"""quda.h-parity API tests (role of the reference's c_interface_test +
invert_test driver paths through interface_quda.cpp)."""
import math

import pytest
import torch

from quda_amd import api
from quda_amd.api import (DslashType, EigParam, GaugeParam, InvertParam,
                          InverterType, SolutionType)
from quda_amd.fields.gauge import GaugeField
from quda_amd.fields.geometry import LatticeGeometry


@pytest.fixture(scope="module", autouse=True)
def resident():
    geo = LatticeGeometry((4, 4, 4, 4))
    u = GaugeField(geo, "double").random_su3_(seed=131).to_complex()
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.init_quda()
    api.load_gauge_quda(u, gp)
    yield u
    api.end_quda()


def _rand_spinor(shape, seed):
    g = torch.Generator().manual_seed(seed)
    return torch.view_as_complex(torch.randn(*shape, 2, generator=g,
                                             dtype=torch.float64))


def test_invert_wilson_bicgstab(resident):
    p = InvertParam(dslash_type=DslashType.WILSON,
                    inv_type=InverterType.BICGSTAB, kappa=0.12, tol=1e-9,
                    maxiter=500, cuda_prec="double")
    b = _rand_spinor((2, 128, 4, 3), 132)
    x = api.invert_quda(b, p)
    assert p.true_res < 1e-8
    assert p.iter > 0 and p.secs > 0


def test_invert_clover_matpc_cg(resident):
    p = InvertParam(dslash_type=DslashType.CLOVER, clover_csw=1.0,
                    inv_type=InverterType.CG,
                    solution_type=SolutionType.MATPC, kappa=0.12,
                    tol=1e-10, maxiter=500, cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_clover_quda(p)
    b = _rand_spinor((2, 128, 4, 3), 133)
    x = api.invert_quda(b, p)
    assert p.true_res < 1e-7, p.true_res


def test_mat_and_dslash(resident):
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.1)
    psi = _rand_spinor((2, 128, 4, 3), 134)
    out = api.mat_quda(psi, p)
    # M = 1 - kappa D
    from quda_amd.ops import reference as ref
    geo = LatticeGeometry((4, 4, 4, 4))
    expect = ref.mat_wilson(resident, psi, geo, 0.1)
    assert (out - expect).abs().max().item() < 1e-12


def test_multishift(resident):
    p = InvertParam(dslash_type=DslashType.WILSON,
                    solution_type=SolutionType.MATPC, kappa=0.12,
                    tol=1e-10, maxiter=500)
    b = _rand_spinor((1, 128, 4, 3), 135)
    xs = api.invert_multishift_quda(b, p, [0.0, 0.2, 1.0])
    assert len(xs) == 3 and p.iter > 0


def test_eigensolve(resident):
    p = InvertParam(dslash_type=DslashType.WILSON,
                    solution_type=SolutionType.MATPC, kappa=0.1)
    e = EigParam(n_ev=4, n_kr=16, tol=1e-7)
    evals, evecs = api.eigensolve_quda(p, e)
    assert len(evals) == 4 and all(v > 0 for v in evals)


def test_mg_through_api(resident):
    p = InvertParam(dslash_type=DslashType.WILSON,
                    inv_type=InverterType.GCR, kappa=0.145, tol=1e-8,
                    maxiter=300)
    mg = api.new_multigrid_quda(p, block=(2, 2, 2, 2), n_vec=4,
                                null_tol=1e-4, null_maxiter=300)
    p.preconditioner = mg.precond
    b = _rand_spinor((2, 128, 4, 3), 136)
    api.invert_quda(b, p)
    assert p.true_res < 1e-7


def test_observables_and_smear(resident):
    obs = api.gauge_observables_quda()
    assert -0.5 < obs["plaquette"][0] < 1  # random field: ~0
    p0 = obs["plaquette"][0]
    api.perform_gauge_smear_quda("stout", 2, 0.1)
    assert api.plaq_quda()[0] > p0


def test_twisted_and_staggered_paths(resident):
    p = InvertParam(dslash_type=DslashType.TWISTED_MASS, kappa=0.12,
                    mu=0.05, inv_type=InverterType.CGNR, tol=1e-9,
                    maxiter=800)
    b = _rand_spinor((2, 128, 4, 3), 137)
    api.invert_quda(b, p)
    assert p.true_res < 1e-8
    ps = InvertParam(dslash_type=DslashType.STAGGERED, mass=0.1,
                     solution_type=SolutionType.MATPC,
                     inv_type=InverterType.CG, tol=1e-10, maxiter=800)
    bs = _rand_spinor((2, 128, 3), 138)
    api.invert_quda(bs, ps)
    assert ps.true_res < 1e-8


def test_hmc_entry_points(resident):
    import torch
    F = api.compute_gauge_force_quda(5.5)
    assert F.shape == (4, 2, 128, 3, 3)
    P = api.gauss_mom_quda(seed=301)
    k0 = api.mom_action_quda(P)
    assert k0 > 0
    p_before = api.plaq_quda()[0]
    api.update_gauge_field_quda(P, 0.01)
    assert abs(api.plaq_quda()[0] - p_before) > 1e-8  # moved
    fat, lng = api.compute_ks_link_quda()
    assert fat.shape == lng.shape == (4, 2, 128, 3, 3)


def test_gauge_fixing_entry(resident):
    from quda_amd.gauge.fix import gauge_fix_quality
    # reload a fresh field (earlier tests mutate the resident gauge)
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    _, th0 = gauge_fix_quality(api._R.u_complex, api._R.geo, 4)
    api.compute_gauge_fixing_ovr_quda("landau", max_iter=300, tol=1e-6)
    _, th = gauge_fix_quality(api._R.u_complex, api._R.geo, 4)
    # hot random fields converge slowly; deep convergence is covered by
    # tests/test_gauge_fix.py on a smooth field
    assert th < 1e-4 and th < th0 * 1e-2


def test_matdagmat_and_save_gauge(resident):
    # earlier tests smear the resident field in place: reload the original
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12)
    b = _rand_spinor((2, 128, 4, 3), 501)
    mm = api.mat_dag_mat_quda(b, p)
    m1 = api.mat_quda(b, p)
    # compare against M^dag(M b) via dagger entry
    from quda_amd.api import _make_dirac, _wrap
    d = _make_dirac(p)
    t = _wrap(m1, p, 2)
    o = t.clone_empty()
    d.M(o, t, dagger=True)
    assert (mm - o.to_complex()).abs().max().item() < 1e-12
    u = api.save_gauge_quda()
    assert (u - resident).abs().max().item() == 0.0


def test_multi_src_api(resident):
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12, tol=1e-9,
                    maxiter=300)
    bs = [_rand_spinor((2, 128, 4, 3), 510 + j) for j in range(3)]
    xs = api.invert_multi_src_quda(bs, p)
    for b, x in zip(bs, xs):
        x1 = api.invert_quda(b, p)
        assert (x - x1).abs().max().item() < 1e-8


def test_blas_gemm_contract_api(resident):
    a = torch.randn(4, 8, 8, dtype=torch.complex128)
    b = torch.randn(4, 8, 8, dtype=torch.complex128)
    c = torch.randn(4, 8, 8, dtype=torch.complex128)
    r = api.blas_gemm_quda(a, b, alpha=2.0, beta=0.5, c=c)
    assert (r - (2.0 * a @ b + 0.5 * c)).abs().max().item() < 1e-12
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12)
    x = _rand_spinor((2, 128, 4, 3), 520)
    y = _rand_spinor((2, 128, 4, 3), 521)
    open_c = api.contract_quda(x, y, p, mode="open")
    assert open_c.shape[-2:] == (4, 4)
    ft = api.contract_ft_quda(x, y, p, [(0, 0, 0), (1, 0, 0)])
    assert ft.shape == (2, 4, 16)


def test_deflation_and_chrono_api(resident):
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.10, tol=1e-8,
                    maxiter=400)
    e = EigParam(n_ev=4, n_kr=16, tol=1e-6)
    defl = api.new_deflation_quda(p, e)
    assert len(defl.evals) == 4
    ch = api.chrono_forecaster(0)
    assert api.chrono_forecaster(0) is ch
    api.flush_chrono_quda(0)
    assert api.chrono_forecaster(0) is not ch
    api.flush_chrono_quda()


def test_fermion_smear_api(resident):
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12)
    src = torch.zeros((2, 128, 4, 3), dtype=torch.complex128)
    src[0, 0, 0, 0] = 1.0
    sm = api.perform_fermion_smear_quda(src, p, n_steps=3, width=0.5)
    assert (sm.abs() > 1e-12).sum() > (src.abs() > 1e-12).sum()
    n0 = src.abs().square().sum()


def test_clover_force_api(resident):
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.1)
    phi = _rand_spinor((2, 128, 4, 3), 530)
    F = api.compute_clover_force_quda(0.1, 1.2, phi)
    assert F.shape == (4, 2, 128, 3, 3)
    # traceless-antihermitian output
    ta = F + F.conj().transpose(-1, -2)
    assert ta.abs().max().item() < 1e-10
    tr = torch.diagonal(F, dim1=-2, dim2=-1).sum(-1)
    assert tr.abs().max().item() < 1e-10


def test_update_multigrid_api(resident):
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12, tol=1e-8,
                    maxiter=300, inv_type=InverterType.GCR)
    mg = api.new_multigrid_quda(p, block=(2, 2, 2, 2), n_vec=4)
    api.update_multigrid_quda(mg, p)
    p2 = InvertParam(**{**p.__dict__, "preconditioner": mg.precond})
    b = _rand_spinor((2, 128, 4, 3), 540)
    x = api.invert_quda(b, p2)
    assert p2.true_res < 1e-7


def test_verbosity_api(capsys):
    api.set_verbosity_quda(1)
    api.log_quda(1, "visible")
    api.log_quda(2, "hidden")
    out = capsys.readouterr().out
    assert "visible" in out and "hidden" not in out


def test_gmresdr_and_eigcg_through_api(resident):
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    b = _rand_spinor((2, 128, 4, 3), 551)
    for inv in (InverterType.GMRESDR, InverterType.EIGCG):
        p = InvertParam(dslash_type=DslashType.WILSON, inv_type=inv,
                        kappa=0.12, tol=1e-9, maxiter=800)
        x = api.invert_quda(b, p)
        assert p.true_res < 1e-7, (inv, p.true_res)


def test_deflated_invert_through_api(resident):
    """newDeflationQuda + deflated invertQuda (deflated_invert_test
    role): the deflated solve converges in no more iterations and to the
    same solution."""
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.124, tol=1e-9,
                    maxiter=600)
    e = EigParam(n_ev=6, n_kr=24, tol=1e-7)
    defl = api.new_deflation_quda(p, e)
    b = _rand_spinor((2, 128, 4, 3), 560)
    x0 = api.invert_quda(b, p)
    it0 = p.iter
    p2 = InvertParam(**{**p.__dict__, "deflation": defl})
    x1 = api.invert_quda(b, p2)
    assert p2.iter <= it0, (p2.iter, it0)
    assert (x1 - x0).abs().max().item() < 1e-6


def test_all_action_types_through_api(resident):
    """Every DslashType constructs and solves through invertQuda (the
    invert_test dslash-type axis)."""
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    pcl = InvertParam(dslash_type=DslashType.CLOVER, kappa=0.11,
                      clover_csw=1.2)
    api.load_clover_quda(pcl)
    api.compute_ks_link_quda()
    cases = [
        (DslashType.DOMAIN_WALL_4D, dict(mass=0.04, Ls=4), (4, 3), 4),
        (DslashType.ZMOBIUS,
         dict(mass=0.04, Ls=4,
              b5_z=[1.5 + 0.1j, 1.4 - 0.05j, 1.6 + 0.02j, 1.5 - 0.08j],
              c5_z=[0.5 + 0.1j, 0.4 - 0.05j, 0.6 + 0.02j, 0.5 - 0.08j]),
         (4, 3), 4),
        (DslashType.MOBIUS_EOFA,
         dict(Ls=4, mq1=0.04, eofa_shift=-0.2), (4, 3), 4),
        (DslashType.NDEG_TWISTED_MASS,
         dict(kappa=0.11, mu=0.2, epsilon=0.1), (4, 3), 2),
        (DslashType.NDEG_TWISTED_CLOVER,
         dict(kappa=0.11, mu=0.2, epsilon=0.1), (4, 3), 2),
        (DslashType.ASQTAD, dict(mass=0.1), (3,), 1),
    ]
    g = torch.Generator().manual_seed(570)
    for t, kw, site, ls in cases:
        p = InvertParam(dslash_type=t, inv_type=InverterType.CGNR,
                        tol=1e-8, maxiter=3000, **kw)
        b = torch.view_as_complex(torch.randn(
            (2, 128 * ls, *site, 2), generator=g, dtype=torch.float64))
        x = api.invert_quda(b, p)
        assert p.true_res < 1e-6, (t, p.true_res)


def test_covdev_and_gauge_save_api(resident, tmp_path):
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12)
    src = _rand_spinor((2, 128, 4, 3), 580)
    fwd = api.covdev_quda(src, p, mu=2, forward=True)
    bwd = api.covdev_quda(fwd, p, mu=2, forward=False)
    # U^d(x-mu) [U(x-mu) psi(x)] = psi  => bwd(fwd(psi)) = psi
    assert (bwd - src).abs().max().item() < 1e-12
    path = str(tmp_path / "resident.pt")
    api.save_gauge_quda(path)
    from quda_amd.utils.io import load_gauge
    u2, geo2, _ = load_gauge(path)
    assert (u2 - resident).abs().max().item() == 0.0


def test_antiperiodic_t_boundary(resident):
    """t_boundary="anti": the loaded operator equals the periodic one on
    a T-link-negated field, and free-field pion correlators genuinely
    differ between the two boundary conditions."""
    geo = LatticeGeometry((4, 4, 4, 4))
    u = resident.clone()
    gp_a = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                      cuda_prec_sloppy="double", t_boundary="anti")
    api.load_gauge_quda(u, gp_a)
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12)
    b = _rand_spinor((2, 128, 4, 3), 590)
    out_a = api.mat_quda(b, p)
    # manual phase application + periodic load
    u2 = u.clone()
    from quda_amd.fields.geometry import LatticeGeometry as LG
    g2 = LG((4, 4, 4, 4))
    for par in (0, 1):
        idx = g2.face_index_cb(par, 3, 3)
        u2[3, par, idx] = -u2[3, par, idx]
    # manually-negated boundary links are not SU(3): recon must be off
    gp_p = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                      cuda_prec_sloppy="double", reconstruct_sloppy="none")
    api.load_gauge_quda(u2, gp_p)
    out_p = api.mat_quda(b, p)
    assert (out_a - out_p).abs().max().item() < 1e-13
    # and the boundary matters: differs from the purely periodic op
    api.load_gauge_quda(u, gp_p)
    out_per = api.mat_quda(b, p)
    assert (out_a - out_per).abs().max().item() > 1e-3


def test_anisotropy_folds_into_spatial_links(resident):
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double", anisotropy=2.5)
    api.load_gauge_quda(resident, gp)
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12)
    b = _rand_spinor((2, 128, 4, 3), 595)
    out_a = api.mat_quda(b, p)
    u2 = resident.clone()
    u2[0:3] = u2[0:3] / 2.5
    gp_p = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                      cuda_prec_sloppy="double",
                      reconstruct_sloppy="none")  # scaled links: no recon
    api.load_gauge_quda(u2, gp_p)
    out_m = api.mat_quda(b, p)
    assert (out_a - out_m).abs().max().item() < 1e-13


def test_staggered_phase_applied_interop(resident):
    """MILC-convention phase-folded links load to the same staggered
    operator as bare links."""
    from quda_amd.ops.reference import staggered_phases
    geo = LatticeGeometry((4, 4, 4, 4))
    u = resident.clone()
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(u, gp)
    p = InvertParam(dslash_type=DslashType.STAGGERED, mass=0.1)
    g = torch.Generator().manual_seed(598)
    b = torch.view_as_complex(torch.randn((2, 128, 3, 2), generator=g,
                                          dtype=torch.float64))
    out_bare = api.mat_quda(b, p)
    u_ph = u.clone()
    for par in (0, 1):
        ph = staggered_phases(geo, par)
        for mu in range(4):
            u_ph[mu, par] = u_ph[mu, par] * ph[:, mu].to(u.dtype).reshape(
                -1, 1, 1)
    gp2 = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                     cuda_prec_sloppy="double",
                     staggered_phase_applied=True)
    api.load_gauge_quda(u_ph, gp2)
    out_ph = api.mat_quda(b, p)
    assert (out_bare - out_ph).abs().max().item() < 1e-13


def test_chrono_guess_accelerates_sequence(resident):
    """use_resident_chrono role: successive related solves reuse the
    chrono basis and converge in fewer iterations."""
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    api.flush_chrono_quda()
    g = torch.Generator().manual_seed(599)
    b0 = torch.view_as_complex(torch.randn((2, 128, 4, 3, 2), generator=g,
                                           dtype=torch.float64))
    db = torch.view_as_complex(torch.randn((2, 128, 4, 3, 2), generator=g,
                                           dtype=torch.float64))
    iters_plain, iters_chrono = [], []
    for i in range(3):
        b = b0 + 0.01 * i * db  # slowly-varying source sequence
        p1 = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12,
                         tol=1e-9, maxiter=500)
        api.invert_quda(b, p1)
        iters_plain.append(p1.iter)
        p2 = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12,
                         tol=1e-9, maxiter=500, chrono_index=1)
        api.invert_quda(b, p2)
        iters_chrono.append(p2.iter)
    api.flush_chrono_quda()
    assert iters_chrono[-1] < iters_plain[-1], (iters_chrono, iters_plain)



def test_gauge_path_force_matches_plaquette(resident):
    """computeGaugeForceQuda with explicit plaquette paths reproduces
    the analytic plaquette force."""
    from quda_amd.gauge import gauge_force
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    paths, coeffs = [], []
    for mu in range(4):
        for nu in range(mu + 1, 4):
            paths.append((mu + 1, nu + 1, -(mu + 1), -(nu + 1)))
            coeffs.append(1.0)
    beta = 5.5
    F = api.compute_gauge_path_force_quda(paths, coeffs, beta)
    F_ref = gauge_force(resident, api._R.geo, beta)
    assert (F - F_ref).abs().max().item() < 1e-12
    # resident momentum round trip
    api.mom_resident_quda(F)
    assert api.mom_resident_quda() is F


def test_mixed_precision_and_dwf_matpc_through_api(resident):
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="single")
    api.load_gauge_quda(resident, gp)
    # mixed-precision CG (reliable updates) through the API
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12, tol=1e-9,
                    maxiter=600, cuda_prec="double",
                    cuda_prec_sloppy="single")
    b = _rand_spinor((2, 128, 4, 3), 601)
    api.invert_quda(b, p)
    assert p.true_res < 1e-8, p.true_res
    # Mobius MATPC solve + reconstruct through the API
    p2 = InvertParam(dslash_type=DslashType.MOBIUS, mass=0.04, Ls=4,
                     solution_type=SolutionType.MATPC,
                     inv_type=InverterType.CGNR, tol=1e-9, maxiter=2000,
                     cuda_prec="double", cuda_prec_sloppy="double")
    b5 = _rand_spinor((2, 128 * 4, 4, 3), 602)
    api.invert_quda(b5, p2)
    assert p2.true_res < 1e-7, p2.true_res


def test_init_guess_and_flow_measure(resident):
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12,
                    inv_type=InverterType.CGNR, tol=1e-9, maxiter=500)
    b = _rand_spinor((2, 128, 4, 3), 611)
    x1 = api.invert_quda(b, p)
    it_cold = p.iter
    x2 = api.invert_quda(b, p, x0=x1)  # warm start at the solution
    assert p.iter < max(it_cold // 2, 2), (p.iter, it_cold)
    hist = api.perform_gauge_smear_quda("wilson_flow", 3, 0.02,
                                        measure=True)
    assert len(hist) == 3 and hist[0][0] < hist[-1][0]


def test_hasenbusch_and_laplace_types(resident):
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.load_gauge_quda(resident, gp)
    pcl = InvertParam(dslash_type=DslashType.CLOVER, kappa=0.11,
                      clover_csw=1.2)
    api.load_clover_quda(pcl)
    p = InvertParam(dslash_type=DslashType.CLOVER_HASENBUSCH_TWIST,
                    kappa=0.11, mu=0.2, inv_type=InverterType.CGNR,
                    tol=1e-8, maxiter=800)
    b = _rand_spinor((2, 128, 4, 3), 621)
    api.invert_quda(b, p)
    assert p.true_res < 1e-6, p.true_res
    p2 = InvertParam(dslash_type=DslashType.LAPLACE, mass=0.5,
                     inv_type=InverterType.CG, tol=1e-9, maxiter=400)
    x = api.invert_quda(b, p2)
    assert p2.true_res < 1e-7, p2.true_res


def test_anti_boundary_forces_reconstruct_off(resident):
    """anti-periodic t-boundary negates U_t on the last
    timeslice; recon-12 cannot represent negated SU(3) links, so the load
    must force reconstruction off for both residencies, and the sloppy
    operator must agree with the precise one."""
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double", t_boundary="anti")
    assert gp.reconstruct_sloppy == "twelve"  # the default, pre-load
    api.load_gauge_quda(resident.clone(), gp)
    assert gp.reconstruct_sloppy == "none"
    # sloppy links bit-match the precise ones (same precision here)
    d = (api._R.gauge.to_complex() - api._R.gauge_sloppy.to_complex())
    assert d.abs().max().item() < 1e-14
