"""Domain-wall / Moebius tests (analogue of the reference's DWF coverage
in dslash_ctest + invert_test for dslash_type domain_wall/mobius)."""
import math

import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.models.dwf import (DiracDomainWall, DiracDomainWallPC,
                                 DiracMobius, DiracMobiusPC)
from quda_amd.ops import blas
from quda_amd.ops import reference as ref
from quda_amd.ops.dispatch import dwf5_op
from quda_amd.solvers import cgnr_solve, cg_solve

MF, M5, LS = 0.04, 1.8, 6


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=101)
    return geo, g


def spin5(geo, seed, n_parity=2):
    return SpinorField(geo, "double", n_parity=n_parity, ls=LS).gaussian_(seed=seed)


def test_m5inv_oracle_inverts(setup):
    geo, g = setup
    psi = spin5(geo, 102, n_parity=1).to_complex()[0]
    for dagger in (False, True):
        y = ref.m5inv(psi, LS, 1.3, -0.4, MF, dagger)
        back = ref.dslash5(y, LS, 1.3, -0.4, MF, dagger)
        assert (back - psi).abs().max().item() < 1e-10


def test_dslash5_dagger_adjoint(setup):
    geo, g = setup
    a = spin5(geo, 103, n_parity=1).to_complex()[0]
    b = spin5(geo, 104, n_parity=1).to_complex()[0]
    Da = ref.dslash5(a, LS, 0.7, -0.3, MF, dagger=False)
    Ddag_b = ref.dslash5(b, LS, 0.7, -0.3, MF, dagger=True)
    lhs = (b.conj() * Da).sum()
    rhs = (Ddag_b.conj() * a).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)


def test_mobius_M_dagger_adjoint(setup):
    geo, g = setup
    d = DiracMobius(g, MF, M5, LS)
    a = spin5(geo, 105)
    b = spin5(geo, 106)
    Ma = SpinorField(geo, "double", ls=LS)
    Mdb = SpinorField(geo, "double", ls=LS)
    d.M(Ma, a)
    d.M(Mdb, b, dagger=True)
    lhs = (b.to_complex().conj() * Ma.to_complex()).sum()
    rhs = (Mdb.to_complex().conj() * a.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)


def test_dwf_pc_vs_full_solve(setup):
    geo, g = setup
    full = DiracDomainWall(g, MF, M5, LS)
    pc = DiracDomainWallPC(g, MF, M5, LS)
    b = spin5(geo, 107)
    x_full = SpinorField(geo, "double", ls=LS)
    st = cgnr_solve(full, x_full, b, tol=1e-10, maxiter=3000)
    assert st.converged
    be = pc.prepare(b)
    xe = SpinorField(geo, "double", n_parity=1, ls=LS)
    st2 = cgnr_solve(pc, xe, be, tol=1e-11, maxiter=3000)
    assert st2.converged
    x_rec = SpinorField(geo, "double", ls=LS)
    pc.reconstruct(x_rec, xe, b)
    err = (x_rec.to_complex() - x_full.to_complex()).abs().max().item()
    assert err < 1e-6, err


def test_mobius_pc_vs_full_solve(setup):
    geo, g = setup
    full = DiracMobius(g, MF, M5, LS)
    pc = DiracMobiusPC(g, MF, M5, LS)
    b = spin5(geo, 108)
    x_full = SpinorField(geo, "double", ls=LS)
    st = cgnr_solve(full, x_full, b, tol=1e-10, maxiter=4000)
    assert st.converged
    be = pc.prepare(b)
    xe = SpinorField(geo, "double", n_parity=1, ls=LS)
    st2 = cgnr_solve(pc, xe, be, tol=1e-11, maxiter=4000)
    assert st2.converged
    x_rec = SpinorField(geo, "double", ls=LS)
    pc.reconstruct(x_rec, xe, b)
    err = (x_rec.to_complex() - x_full.to_complex()).abs().max().item()
    assert err < 1e-6, err


# ---------------------------------------------------------------------------
# GPU numerics
# ---------------------------------------------------------------------------

@pytest.mark.gpu
@pytest.mark.parametrize("prec", ["double", "single"])
@pytest.mark.parametrize("kind,dagger", [(0, False), (0, True), (1, False),
                                         (1, True)])
def test_dwf5_gpu_vs_oracle(setup, prec, kind, dagger):
    geo, _ = setup
    inp = SpinorField(geo, prec, "cuda", n_parity=1, ls=LS).gaussian_(seed=111)
    out = SpinorField(geo, prec, "cuda", n_parity=1, ls=LS)
    dwf5_op(out, inp, 1.9, -0.55, MF, kind=kind, dagger=dagger)
    psi = inp.to_complex()[0]
    if kind == 0:
        expect = ref.dslash5(psi, LS, 1.9, -0.55, MF, dagger)
    else:
        expect = ref.m5inv(psi, LS, 1.9, -0.55, MF, dagger)
    err = (out.to_complex()[0] - expect).abs().max().item()
    tol = {"double": 1e-11, "single": 1e-4}[prec]
    assert err < tol, f"{prec} kind={kind} dag={dagger}: {err}"


@pytest.mark.gpu
def test_mobius_M_gpu_vs_cpu(setup):
    geo, _ = setup
    gen = torch.Generator().manual_seed(112)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m))
    g_cpu = GaugeField(geo, "double").from_complex(u)
    g_gpu = GaugeField(geo, "double", "cuda").from_complex(u.cuda())
    psi = SpinorField(geo, "double", ls=LS).gaussian_(seed=113)
    psi_g = SpinorField(geo, "double", "cuda", ls=LS)
    psi_g.from_complex(psi.to_complex().cuda())
    for dagger in (False, True):
        out_c = SpinorField(geo, "double", ls=LS)
        DiracMobius(g_cpu, MF, M5, LS).M(out_c, psi, dagger=dagger)
        out_g = SpinorField(geo, "double", "cuda", ls=LS)
        DiracMobius(g_gpu, MF, M5, LS).M(out_g, psi_g, dagger=dagger)
        err = (out_g.to_complex().cpu() - out_c.to_complex()).abs().max().item()
        assert err < 1e-11, f"dag={dagger}: {err}"


@pytest.mark.gpu
def test_mobius_pc_cg_gpu(setup):
    geo, _ = setup
    geo = LatticeGeometry((8, 8, 8, 8))
    gen = torch.Generator().manual_seed(114)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m)).cuda()
    g = GaugeField(geo, "double", "cuda").from_complex(u)
    pc = DiracMobiusPC(g, MF, M5, 8)
    b = SpinorField(geo, "double", "cuda", n_parity=1, ls=8).gaussian_(seed=115)
    x = SpinorField(geo, "double", "cuda", n_parity=1, ls=8)
    st = cgnr_solve(pc, x, b, tol=1e-8, maxiter=2000)
    assert st.converged


# ---------------------------------------------------------------------------
# 5-d halo exchange
# ---------------------------------------------------------------------------

def test_mobius_self_wraparound_cpu(setup):
    from quda_amd.parallel import comms
    geo, g = setup
    d = DiracMobius(g, MF, M5, LS)
    psi = spin5(geo, 181)
    out_ref = SpinorField(geo, "double", ls=LS)
    d.M(out_ref, psi)
    try:
        comms.set_forced_partition(0b1010)
        g2 = GaugeField(geo, "double").from_complex(g.to_complex())
        d2 = DiracMobius(g2, MF, M5, LS)
        out = SpinorField(geo, "double", ls=LS)
        d2.M(out, psi)
    finally:
        comms.set_forced_partition(0)
    err = (out.to_complex() - out_ref.to_complex()).abs().max().item()
    assert err < 1e-12, err


@pytest.mark.gpu
def test_mobius_self_wraparound_gpu(setup):
    from quda_amd.parallel import comms
    geo, _ = setup
    gen = torch.Generator().manual_seed(182)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m)).cuda()
    g = GaugeField(geo, "double", "cuda").from_complex(u)
    d = DiracMobius(g, MF, M5, LS)
    psi = SpinorField(geo, "double", "cuda", ls=LS).gaussian_(seed=183)
    out_ref = SpinorField(geo, "double", "cuda", ls=LS)
    d.M(out_ref, psi)
    for dagger in (False, True):
        d.M(out_ref, psi, dagger=dagger)
        try:
            comms.set_forced_partition(0b1111)
            g2 = GaugeField(geo, "double", "cuda").from_complex(u)
            d2 = DiracMobius(g2, MF, M5, LS)
            out = SpinorField(geo, "double", "cuda", ls=LS)
            d2.M(out, psi, dagger=dagger)
        finally:
            comms.set_forced_partition(0)
        err = (out.to_complex() - out_ref.to_complex()).abs().max().item()
        assert err < 1e-12, (dagger, err)


# ---------------------------------------------------------------------------
# zMobius (complex per-slice b5/c5)
# ---------------------------------------------------------------------------

def _zcoefs():
    b5 = [1.5 + 0.1j, 1.4 - 0.05j, 1.6 + 0.02j, 1.5 - 0.08j, 1.45 + 0.06j,
          1.55 - 0.03j]
    c5 = [b - 1.0 for b in b5]  # standard zMobius pairing c5 = b5 - 1
    return b5[:LS], c5[:LS]


def test_zm5inv_oracle_inverts(setup):
    geo, _ = setup
    b5, c5 = _zcoefs()
    d4 = 4.0 - M5
    diag = [1.0 + b * d4 for b in b5]
    hop = [c * d4 - 1.0 for c in c5]
    psi = spin5(geo, 301, n_parity=1).to_complex()[0]
    for dag in (False, True):
        y = ref.zm5inv(psi, LS, diag, hop, MF, dag)
        back = ref.zdslash5(y, LS, diag, hop, MF, dag)
        assert (back - psi).abs().max().item() < 1e-11


def test_ztables_match_dense(setup):
    """The host-assembled sequence tables (what the GPU kernel consumes)
    must reproduce the dense oracle when replayed in plain python."""
    import numpy as np
    from quda_amd.ops.dispatch import _ztables
    b5, c5 = _zcoefs()
    d4 = 4.0 - M5
    diag = [1.0 + b * d4 for b in b5]
    hop = [c * d4 - 1.0 for c in c5]
    rng = np.random.default_rng(5)
    r = rng.normal(size=LS) + 1j * rng.normal(size=LS)
    for dag in (False, True):
        t = _ztables(LS, diag, hop, MF, dag)
        for blk in ("u", "l"):
            ordl = t["ord_" + blk]
            di = t["di" + blk]
            e = t["e" + blk]
            cw = complex(t["cw" + blk + "_re"], t["cw" + blk + "_im"])
            di = [complex(di[2 * i], di[2 * i + 1]) for i in range(LS)]
            e = [complex(e[2 * i], e[2 * i + 1]) for i in range(LS)]
            # replay: forward substitution + Sherman-Morrison
            y = np.zeros(LS, dtype=complex)
            z = np.zeros(LS, dtype=complex)
            for i in range(LS):
                s = ordl[i]
                y[s] = di[i] * (r[s] - (e[i] * y[ordl[i - 1]] if i else 0))
                z[s] = di[i] * (cw if i == 0 else 0) - (di[i] * e[i] * z[ordl[i - 1]] if i else 0)
            last = ordl[LS - 1]
            y = y - z * y[last] / (1 + z[last])
            # dense truth
            A = ref._zm5_matrix(LS, diag, hop, MF, blk == "u", dag)
            yd = np.linalg.solve(A, r)
            assert np.abs(y - yd).max() < 1e-12, (blk, dag)


def test_zmobius_dagger_adjoint(setup):
    geo, g = setup
    b5, c5 = _zcoefs()
    from quda_amd.models import DiracZMobius
    op = DiracZMobius(g, MF, M5, LS, b5, c5)
    psi = spin5(geo, 302)
    chi = spin5(geo, 303)
    Mp = spin5(geo, 0)
    Mdc = spin5(geo, 0)
    op.M(Mp, psi)
    op.M(Mdc, chi, dagger=True)
    lhs = blas.c_dot(chi, Mp)
    rhs = blas.c_dot(Mdc, psi)
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)


def test_zmobius_reduces_to_mobius(setup):
    """Real constant b5/c5 must reproduce the scalar Moebius operator."""
    geo, g = setup
    op_r = DiracMobius(g, MF, M5, LS, b5=1.5, c5=0.5)
    from quda_amd.models import DiracZMobius
    op_z = DiracZMobius(g, MF, M5, LS, [1.5] * LS, [0.5] * LS)
    psi = spin5(geo, 304)
    a = spin5(geo, 0)
    b = spin5(geo, 0)
    for dag in (False, True):
        op_r.M(a, psi, dagger=dag)
        op_z.M(b, psi, dagger=dag)
        assert (a.to_complex() - b.to_complex()).abs().max().item() < 1e-11


def test_zmobius_pc_vs_full_solve(setup):
    geo, g = setup
    b5, c5 = _zcoefs()
    from quda_amd.models import DiracZMobius, DiracZMobiusPC
    full = DiracZMobius(g, MF, M5, LS, b5, c5)
    pc = DiracZMobiusPC(g, MF, M5, LS, b5, c5)
    b = spin5(geo, 305)
    x_full = spin5(geo, 0)
    st = cgnr_solve(full, x_full, b, tol=1e-10, maxiter=4000)
    assert st.converged
    be = pc.prepare(b)
    xe = SpinorField(geo, "double", n_parity=1, ls=LS)
    from quda_amd.solvers import cgnr_solve as _c
    st2 = _c(pc, xe, be, tol=1e-11, maxiter=4000)
    assert st2.converged
    x_rec = spin5(geo, 0)
    pc.reconstruct(x_rec, xe, b)
    err = (x_rec.to_complex() - x_full.to_complex()).abs().max().item()
    assert err < 1e-6, err


@pytest.mark.gpu
@pytest.mark.parametrize("prec", ["double", "single"])
@pytest.mark.parametrize("kind", [0, 1])
@pytest.mark.parametrize("dagger", [False, True])
def test_zdwf5_gpu_vs_oracle(setup, prec, kind, dagger):
    from quda_amd.ops.dispatch import zdwf5_op
    geo, _ = setup
    b5, c5 = _zcoefs()
    d4 = 4.0 - M5
    diag = [1.0 + b * d4 for b in b5]
    hop = [c * d4 - 1.0 for c in c5]
    inp = SpinorField(geo, prec, "cuda", n_parity=1, ls=LS).gaussian_(seed=306)
    out = SpinorField(geo, prec, "cuda", n_parity=1, ls=LS)
    zdwf5_op(out, inp, diag, hop, MF, kind, dagger=dagger)
    psi = inp.to_complex()[0]
    if kind == 0:
        expect = ref.zdslash5(psi, LS, diag, hop, MF, dagger)
    else:
        expect = ref.zm5inv(psi, LS, diag, hop, MF, dagger)
    got = out.to_complex()[0]
    tol = 1e-11 if prec == "double" else 1e-4
    assert (got - expect).abs().max().item() < tol


@pytest.mark.gpu
def test_zmobius_pc_cg_gpu(setup):
    geo, _ = setup
    b5, c5 = _zcoefs()
    gen = torch.Generator().manual_seed(307)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m)).cuda()
    g = GaugeField(geo, "double", "cuda").from_complex(u)
    from quda_amd.models import DiracZMobiusPC
    pc = DiracZMobiusPC(g, MF, M5, LS, b5, c5)
    b = SpinorField(geo, "double", "cuda", n_parity=1, ls=LS).gaussian_(seed=308)
    x = SpinorField(geo, "double", "cuda", n_parity=1, ls=LS)
    st = cgnr_solve(pc, x, b, tol=1e-10, maxiter=4000)
    assert st.converged
    # true residual of M x = b
    r = SpinorField(geo, "double", "cuda", n_parity=1, ls=LS)
    pc.M(r, x)
    import math as _m
    tr = _m.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b))
    assert tr < 1e-8, tr


# ---------------------------------------------------------------------------
# MADWF (Ls -> Ls' accelerated solver)
# ---------------------------------------------------------------------------

def test_madwf_transfer_adjoint(setup):
    from quda_amd.solvers.madwf import TransferLs
    geo, _ = setup
    T = TransferLs(LS, LS // 2)
    # randomize so the adjoint check is non-trivial
    gen = torch.Generator().manual_seed(320)
    T.Tp = torch.randn((LS // 2, LS, 2), generator=gen,
                       dtype=torch.float64)
    T.Tp = torch.view_as_complex(T.Tp)
    T.Tm = torch.view_as_complex(
        torch.randn((LS // 2, LS, 2), generator=gen, dtype=torch.float64))
    big = spin5(geo, 321, n_parity=1)
    small = SpinorField(geo, "double", n_parity=1, ls=LS // 2).gaussian_(seed=322)
    tb = SpinorField(geo, "double", n_parity=1, ls=LS // 2)
    T.apply(tb, big)
    td = SpinorField(geo, "double", n_parity=1, ls=LS)
    T.apply_dag(td, small)
    lhs = (small.to_complex().conj() * tb.to_complex()).sum()
    rhs = (td.to_complex().conj() * big.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-12 * abs(lhs)


def test_madwf_identity_transfer_exact(setup):
    """Ls' = Ls with the identity transfer makes the preconditioner an
    exact solve: outer GCR must converge in one iteration."""
    from quda_amd.models import DiracMobiusPC
    from quda_amd.solvers.madwf import TransferLs, madwf_solve
    geo, g = setup
    big = DiracMobiusPC(g, MF, M5, LS, b5=1.5, c5=0.5)
    b = spin5(geo, 323, n_parity=1)
    x = SpinorField(geo, "double", n_parity=1, ls=LS)
    st = madwf_solve(big, big, x, b, T=TransferLs(LS, LS), tol=1e-8,
                     inner_tol=1e-10, inner_maxiter=500)
    assert st.converged and st.iters <= 2, st.iters
    r = SpinorField(geo, "double", n_parity=1, ls=LS)
    big.M(r, x)
    import math as _m
    assert _m.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b)) < 1e-7


def test_madwf_truncated_solver(setup):
    """Ls'=Ls/2 truncation: preconditioned solve converges to the same
    solution, in no more outer iterations than plain GCR."""
    from quda_amd.models import DiracMobiusPC
    from quda_amd.solvers import gcr_solve
    from quda_amd.solvers.madwf import madwf_solve
    geo, g = setup
    big = DiracMobiusPC(g, MF, M5, LS, b5=1.5, c5=0.5)
    small = DiracMobiusPC(g, MF, M5, LS // 2, b5=1.5, c5=0.5)
    b = spin5(geo, 324, n_parity=1)
    x0 = SpinorField(geo, "double", n_parity=1, ls=LS)
    st0 = gcr_solve(big, x0, b, tol=1e-8, maxiter=300)
    x1 = SpinorField(geo, "double", n_parity=1, ls=LS)
    st1 = madwf_solve(big, small, x1, b, tol=1e-8, inner_tol=1e-4)
    assert st0.converged and st1.converged
    assert st1.iters <= st0.iters + 1, (st1.iters, st0.iters)
    err = (x1.to_complex() - x0.to_complex()).abs().max().item()
    assert err < 1e-6, err


def test_madwf_training_reduces_chi2(setup):
    from quda_amd.models import DiracMobiusPC
    from quda_amd.solvers.madwf import TransferLs, train_transfer
    geo, g = setup
    big = DiracMobiusPC(g, MF, M5, LS, b5=1.5, c5=0.5)
    small = DiracMobiusPC(g, MF, M5, LS // 2, b5=1.5, c5=0.5)
    T = TransferLs(LS, LS // 2)
    hist = train_transfer(big, small, T, n_samples=2, iters=10, lr=0.05,
                          inner_tol=1e-6)
    assert hist[-1] < hist[0], hist


# ---------------------------------------------------------------------------
# EOFA (rank-1 chiral extension of M5)
# ---------------------------------------------------------------------------

EOFA_KW = dict(mq1=MF, eofa_pm=1, eofa_shift=-0.21)


def _eofa_vecs(op):
    return dict(alpha=op.alpha, beta=op.beta, mf=op.mf, sh=op.eofa_shift,
                pm=op.eofa_pm, u=op.eofa_u, w=op.eofa_u)


@pytest.mark.parametrize("pm", [1, -1])
@pytest.mark.parametrize("dagger", [False, True])
def test_m5inv_eofa_oracle_inverts(setup, pm, dagger):
    geo, g = setup
    from quda_amd.models import DiracMobiusEofa
    op = DiracMobiusEofa(g, M5, LS, mq1=MF, eofa_pm=pm, eofa_shift=-0.21)
    kw = _eofa_vecs(op)
    psi = spin5(geo, 331, n_parity=1).to_complex()[0]
    y = ref.m5inv_eofa(psi, LS, kw["alpha"], kw["beta"], kw["mf"], kw["sh"],
                       kw["pm"], kw["u"], kw["w"], dagger)
    back = ref.m5_eofa(y, LS, kw["alpha"], kw["beta"], kw["mf"], kw["sh"],
                       kw["pm"], kw["u"], kw["w"], dagger)
    assert (back - psi).abs().max().item() < 1e-11


@pytest.mark.parametrize("dagger", [False, True])
def test_eofa_dispatch_ainv_inverts_a(setup, dagger):
    """apply_Ainv(apply_A(psi)) == psi through the dispatch layer (the
    GPU variant of this exercises the host Sherman-Morrison setup)."""
    geo, g = setup
    from quda_amd.models import DiracMobiusEofa
    op = DiracMobiusEofa(g, M5, LS, **EOFA_KW)
    psi = spin5(geo, 332, n_parity=1)
    t = SpinorField(geo, "double", n_parity=1, ls=LS)
    b = SpinorField(geo, "double", n_parity=1, ls=LS)
    op.apply_A(t, psi, dagger=dagger)
    op.apply_Ainv(b, t, dagger=dagger)
    assert (b.to_complex() - psi.to_complex()).abs().max().item() < 1e-11


def test_eofa_M_dagger_adjoint(setup):
    geo, g = setup
    from quda_amd.models import DiracMobiusEofa
    op = DiracMobiusEofa(g, M5, LS, **EOFA_KW)
    a = spin5(geo, 333)
    b = spin5(geo, 334)
    Ma = SpinorField(geo, "double", ls=LS)
    Mdb = SpinorField(geo, "double", ls=LS)
    op.M(Ma, a)
    op.M(Mdb, b, dagger=True)
    lhs = (b.to_complex().conj() * Ma.to_complex()).sum()
    rhs = (Mdb.to_complex().conj() * a.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)


def test_eofa_shift_zero_reduces_to_mobius(setup):
    geo, g = setup
    from quda_amd.models import DiracMobiusEofa
    op0 = DiracMobius(g, MF, M5, LS)
    ope = DiracMobiusEofa(g, M5, LS, mq1=MF, eofa_shift=0.0)
    psi = spin5(geo, 335)
    a = spin5(geo, 0)
    b = spin5(geo, 0)
    op0.M(a, psi)
    ope.M(b, psi)
    assert (a.to_complex() - b.to_complex()).abs().max().item() < 1e-12


def test_eofa_pc_solve(setup):
    geo, g = setup
    from quda_amd.models import DiracMobiusEofaPC
    pc = DiracMobiusEofaPC(g, M5, LS, **EOFA_KW)
    b = spin5(geo, 336, n_parity=1)
    x = SpinorField(geo, "double", n_parity=1, ls=LS)
    st = cgnr_solve(pc, x, b, tol=1e-10, maxiter=4000)
    assert st.converged
    r = SpinorField(geo, "double", n_parity=1, ls=LS)
    pc.M(r, x)
    tr = math.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b))
    assert tr < 1e-8, tr


@pytest.mark.gpu
@pytest.mark.parametrize("prec", ["double", "single"])
@pytest.mark.parametrize("kind", [0, 1])
@pytest.mark.parametrize("dagger", [False, True])
@pytest.mark.parametrize("pm", [1, -1])
def test_eofa5_gpu_vs_oracle(setup, prec, kind, dagger, pm):
    from quda_amd.ops.dispatch import eofa5_op
    geo, g = setup
    from quda_amd.models import DiracMobiusEofa
    op = DiracMobiusEofa(g, M5, LS, mq1=MF, eofa_pm=pm, eofa_shift=-0.21)
    kw = _eofa_vecs(op)
    inp = SpinorField(geo, prec, "cuda", n_parity=1, ls=LS).gaussian_(seed=337)
    out = SpinorField(geo, prec, "cuda", n_parity=1, ls=LS)
    eofa5_op(out, inp, kw["alpha"], kw["beta"], kw["mf"], kind, kw["sh"],
             kw["pm"], kw["u"], kw["w"], dagger=dagger)
    psi = inp.to_complex()[0]
    fn = ref.m5_eofa if kind == 0 else ref.m5inv_eofa
    expect = fn(psi, LS, kw["alpha"], kw["beta"], kw["mf"], kw["sh"],
                kw["pm"], kw["u"], kw["w"], dagger)
    tol = 1e-11 if prec == "double" else 1e-4
    assert (out.to_complex()[0] - expect).abs().max().item() < tol


def test_madwf_with_zmobius_inner(setup):
    """The production MADWF configuration: a cheap COMPLEX-coefficient
    (zMobius) inner operator accelerating a Moebius outer solve."""
    from quda_amd.models import DiracMobiusPC, DiracZMobiusPC
    from quda_amd.solvers import gcr_solve
    from quda_amd.solvers.madwf import madwf_solve
    geo, g = setup
    big = DiracMobiusPC(g, MF, M5, LS, b5=1.5, c5=0.5)
    b5z, c5z = _zcoefs()
    small = DiracZMobiusPC(g, MF, M5, LS // 2, b5z[:LS // 2], c5z[:LS // 2])
    b = spin5(geo, 341, n_parity=1)
    x0 = SpinorField(geo, "double", n_parity=1, ls=LS)
    st0 = gcr_solve(big, x0, b, tol=1e-8, maxiter=300)
    x1 = SpinorField(geo, "double", n_parity=1, ls=LS)
    st1 = madwf_solve(big, small, x1, b, tol=1e-8, inner_tol=1e-4)
    assert st0.converged and st1.converged
    err = (x1.to_complex() - x0.to_complex()).abs().max().item()
    assert err < 1e-6, err
