"""Improved-staggered (asqtad/HISQ-style) tests: link fattening
properties + fat/long dslash consistency + PC solves
(analogue of the reference's hisq_stencil_ctest + llfat coverage)."""
import math

import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.gauge.hisq import (KSLinkCoeffs, asqtad_coefficients,
                                 fat7_coefficients, fat_links, naik_links,
                                 unitarize_links)
from quda_amd.models import DiracImprovedStaggered, DiracImprovedStaggeredPC
from quda_amd.ops import blas
from quda_amd.ops import reference as ref
from quda_amd.ops.dispatch import dslash_staggered
from quda_amd.solvers import cg_solve, multishift_cg_solve

MASS = 0.08


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 6, 4, 8))
    g = GaugeField(geo, "double").random_su3_(seed=141)
    u = g.to_complex()
    fat = fat_links(u, geo, asqtad_coefficients())
    lng = naik_links(u, geo)
    return geo, u, fat, lng


def test_fat7_unit_gauge_closure():
    """Fat7 weights sum to 1: unit gauge gives unit fat links."""
    geo = LatticeGeometry((4, 4, 4, 4))
    u = GaugeField(geo, "double").unit_().to_complex()
    f = fat_links(u, geo, fat7_coefficients())
    eye = torch.eye(3, dtype=torch.complex128)
    assert (f - eye).abs().max().item() < 1e-12


def test_fat_links_gauge_covariance(setup):
    """F_mu transforms like a link: F'(x) = g(x) F(x) g(x+mu)^d."""
    geo, u, fat, lng = setup
    from quda_amd.fields.gauge import project_su3
    from quda_amd.gauge.ops import _from_lex, _to_lex
    gen = torch.Generator().manual_seed(142)
    gt = project_su3(torch.view_as_complex(
        torch.randn((geo.volume, 3, 3, 2), generator=gen, dtype=torch.float64)))
    U = _to_lex(u, geo)
    U2 = torch.empty_like(U)
    for mu in range(4):
        idx = geo.neighbor_lex(mu, +1)
        U2[mu] = gt @ U[mu] @ gt[idx].conj().mT
    u2 = _from_lex(U2, geo)
    fat2 = fat_links(u2, geo, asqtad_coefficients())
    F = _to_lex(fat, geo)
    F2 = _to_lex(fat2, geo)
    for mu in range(4):
        idx = geo.neighbor_lex(mu, +1)
        expect = gt @ F[mu] @ gt[idx].conj().mT
        assert (F2[mu] - expect).abs().max().item() < 1e-10


def test_naik_links_3hop(setup):
    geo, u, fat, lng = setup
    from quda_amd.gauge.ops import _to_lex, _shift
    U = _to_lex(u, geo)
    N = _to_lex(lng, geo)
    mu = 2
    expect = U[mu] @ _shift(U[mu], geo, mu, 1) @ _shift(_shift(U[mu], geo, mu, 1), geo, mu, 1)
    assert (N[mu] - expect).abs().max().item() < 1e-12


def test_unitarize(setup):
    geo, u, fat, lng = setup
    w = unitarize_links(fat)
    eye = torch.eye(3, dtype=torch.complex128)
    err = (w @ w.conj().mT - eye).abs().max().item()
    assert err < 1e-10


def test_improved_antihermiticity(setup):
    geo, u, fat, lng = setup
    gf = GaugeField(geo, "double").from_complex(fat)
    gl = GaugeField(geo, "double", shift=3).from_complex(lng)
    d = DiracImprovedStaggered(gf, gl, 0.0)  # pure D (mass 0)
    a = SpinorField(geo, "double", nspin=1).gaussian_(seed=143)
    b = SpinorField(geo, "double", nspin=1).gaussian_(seed=144)
    Da = SpinorField(geo, "double", nspin=1)
    Db = SpinorField(geo, "double", nspin=1)
    d.M(Da, a)
    d.M(Db, b)
    lhs = (b.to_complex().conj() * Da.to_complex()).sum()
    rhs = -(Db.to_complex().conj() * a.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-10 * max(abs(lhs), 1.0)


def test_improved_dispatch_vs_oracle(setup):
    geo, u, fat, lng = setup
    gf = GaugeField(geo, "double").from_complex(fat)
    gl = GaugeField(geo, "double", shift=3).from_complex(lng)
    full = SpinorField(geo, "double", nspin=1).gaussian_(seed=145)
    out = SpinorField(geo, "double", n_parity=1, nspin=1)
    dslash_staggered(out, full.parity_view(1), gf, 0, long_gauge=gl)
    fc = full.to_complex()
    expect = (ref.dslash_staggered_parity(fat, fc[1], geo, 0)
              + ref.dslash_staggered_naik_parity(lng, fc[1], geo, 0))
    assert (out.to_complex()[0] - expect).abs().max().item() < 1e-11


def test_improved_pc_multishift(setup):
    """The BASELINE config-3 workload shape: HISQ-style multi-shift CG."""
    geo, u, fat, lng = setup
    gf = GaugeField(geo, "double").from_complex(fat)
    gl = GaugeField(geo, "double", shift=3).from_complex(lng)
    pc = DiracImprovedStaggeredPC(gf, gl, MASS)
    b = SpinorField(geo, "double", n_parity=1, nspin=1).gaussian_(seed=146)
    shifts = [0.0, 0.05, 0.5]
    xs = [SpinorField(geo, "double", n_parity=1, nspin=1) for _ in shifts]
    st = multishift_cg_solve(pc, xs, b, shifts, tol=1e-10, maxiter=2000)
    assert st.converged
    for i, s in enumerate(shifts):
        r = SpinorField(geo, "double", n_parity=1, nspin=1)
        t = SpinorField(geo, "double", n_parity=1, nspin=1)
        pc.MdagM(r, xs[i], t)
        blas.axpy(s, xs[i], r)
        tr = math.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b))
        assert tr < 1e-7, f"shift {s}: {tr}"


def test_improved_pc_vs_full(setup):
    geo, u, fat, lng = setup
    gf = GaugeField(geo, "double").from_complex(fat)
    gl = GaugeField(geo, "double", shift=3).from_complex(lng)
    full = DiracImprovedStaggered(gf, gl, MASS)
    pc = DiracImprovedStaggeredPC(gf, gl, MASS)
    b = SpinorField(geo, "double", nspin=1).gaussian_(seed=147)
    from quda_amd.solvers import cgnr_solve
    x_full = SpinorField(geo, "double", nspin=1)
    st = cgnr_solve(full, x_full, b, tol=1e-11, maxiter=3000)
    assert st.converged
    be = pc.prepare(b)
    xe = SpinorField(geo, "double", n_parity=1, nspin=1)
    st2 = cg_solve(pc, xe, be, tol=1e-12, maxiter=3000)
    assert st2.converged
    x_rec = SpinorField(geo, "double", nspin=1)
    pc.reconstruct(x_rec, xe, b)
    err = (x_rec.to_complex() - x_full.to_complex()).abs().max().item()
    assert err < 1e-6, err


@pytest.mark.gpu
def test_improved_gpu_vs_oracle(setup):
    geo, u, fat, lng = setup
    fat_g, lng_g = fat.cuda(), lng.cuda()
    gf = GaugeField(geo, "double", "cuda").from_complex(fat_g)
    gl = GaugeField(geo, "double", "cuda", shift=3).from_complex(lng_g)
    full = SpinorField(geo, "double", "cuda", nspin=1).gaussian_(seed=148)
    out = SpinorField(geo, "double", "cuda", n_parity=1, nspin=1)
    dslash_staggered(out, full.parity_view(1), gf, 0, a=2 * MASS, b=1.0,
                     x=full.parity_view(0), long_gauge=gl)
    fc = full.to_complex()
    expect = (2 * MASS * fc[0]
              + ref.dslash_staggered_parity(fat_g, fc[1], geo, 0)
              + ref.dslash_staggered_naik_parity(lng_g, fc[1], geo, 0))
    err = (out.to_complex()[0] - expect).abs().max().item()
    assert err < 1e-11, err


# ---------------------------------------------------------------------------
# multi-rank Naik (nFace=3 halos)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("mask", [0b1000, 0b1010])
def test_improved_halo_self_wraparound_cpu(setup, mask):
    """Forced self-partition: depth-3 ghost path must reproduce the local
    periodic wrap (exercises the shift-3 gauge boundary exchange and the
    Naik oracle ghost indexing)."""
    from quda_amd.parallel import comms
    geo, u, fat, lng = setup
    gf0 = GaugeField(geo, "double").from_complex(fat)
    gl0 = GaugeField(geo, "double", shift=3).from_complex(lng)
    src = SpinorField(geo, "double", n_parity=1, nspin=1).gaussian_(seed=401)
    out_ref = SpinorField(geo, "double", n_parity=1, nspin=1)
    dslash_staggered(out_ref, src, gf0, 0, long_gauge=gl0)
    try:
        comms.set_forced_partition(mask)
        gf = GaugeField(geo, "double").from_complex(fat)
        gl = GaugeField(geo, "double", shift=3).from_complex(lng)
        out = SpinorField(geo, "double", n_parity=1, nspin=1)
        dslash_staggered(out, src, gf, 0, long_gauge=gl)
    finally:
        comms.set_forced_partition(0)
    err = (out.to_complex() - out_ref.to_complex()).abs().max().item()
    assert err < 1e-12, err


@pytest.mark.gpu
@pytest.mark.parametrize("mask", [0b1000, 0b1111])
def test_improved_halo_self_wraparound_gpu(setup, mask):
    """Same self-partition check on the HIP fused kernel with depth-3
    ghost buffers."""
    from quda_amd.parallel import comms
    geo, u, fat, lng = setup
    fat_d, lng_d = fat.cuda(), lng.cuda()
    gf0 = GaugeField(geo, "double", "cuda").from_complex(fat_d)
    gl0 = GaugeField(geo, "double", "cuda", shift=3).from_complex(lng_d)
    src = SpinorField(geo, "double", "cuda", n_parity=1,
                      nspin=1).gaussian_(seed=402)
    out_ref = SpinorField(geo, "double", "cuda", n_parity=1, nspin=1)
    dslash_staggered(out_ref, src, gf0, 0, long_gauge=gl0)
    try:
        comms.set_forced_partition(mask)
        gf = GaugeField(geo, "double", "cuda").from_complex(fat_d)
        gl = GaugeField(geo, "double", "cuda", shift=3).from_complex(lng_d)
        out = SpinorField(geo, "double", "cuda", n_parity=1, nspin=1)
        dslash_staggered(out, src, gf, 0, long_gauge=gl)
    finally:
        comms.set_forced_partition(0)
    err = (out.to_complex() - out_ref.to_complex()).abs().max().item()
    assert err < 1e-12, err


def _worker_improved(rank, world, init_file):
    import torch.distributed as dist
    from quda_amd.parallel import comms
    from quda_amd.fields.geometry import checkerboard_split
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        grid = (1, 1, 1, world)
        comms.init_comms(grid=grid)
        GD = (4, 4, 4, 8)
        gg = LatticeGeometry(GD)
        gen = torch.Generator().manual_seed(403)
        from quda_amd.fields.gauge import project_su3
        m = torch.randn((4, gg.volume, 3, 3, 2), generator=gen,
                        dtype=torch.float64)
        fat_lex = project_su3(torch.view_as_complex(m))
        m2 = torch.randn((4, gg.volume, 3, 3, 2), generator=gen,
                         dtype=torch.float64)
        lng_lex = project_su3(torch.view_as_complex(m2))
        s = torch.randn((gg.volume, 3, 2), generator=gen, dtype=torch.float64)
        src_lex = torch.view_as_complex(s)

        ldims = tuple(GD[i] // grid[i] for i in range(4))
        lg = LatticeGeometry(ldims)
        coords = comms.grid_coords()
        off = torch.tensor([coords[i] * ldims[i] for i in range(4)])
        c = lg.coords.to(torch.int64) + off
        X, Y, Z, _ = GD
        glex = ((c[:, 3] * Z + c[:, 2]) * Y + c[:, 1]) * X + c[:, 0]

        def loc(lex_field):
            return lex_field[glex]

        fat_loc = checkerboard_split(loc(fat_lex.movedim(0, 1)), lg)
        fat_loc = fat_loc.permute(2, 0, 1, 3, 4).contiguous()
        lng_loc = checkerboard_split(loc(lng_lex.movedim(0, 1)), lg)
        lng_loc = lng_loc.permute(2, 0, 1, 3, 4).contiguous()
        src_loc = checkerboard_split(loc(src_lex), lg)

        gf = GaugeField(lg, "double").from_complex(fat_loc)
        gl = GaugeField(lg, "double", shift=3).from_complex(lng_loc)
        src = SpinorField(lg, "double", nspin=1)
        src.from_complex(src_loc)
        out = SpinorField(lg, "double", n_parity=1, nspin=1)
        dslash_staggered(out, src.parity_view(1), gf, 0, long_gauge=gl)

        # global truth (single lattice, no comms)
        fat_g = checkerboard_split(fat_lex.movedim(0, 1), gg).permute(
            2, 0, 1, 3, 4).contiguous()
        lng_g = checkerboard_split(lng_lex.movedim(0, 1), gg).permute(
            2, 0, 1, 3, 4).contiguous()
        src_g = checkerboard_split(src_lex, gg)
        truth = (ref.dslash_staggered_parity(fat_g, src_g[1], gg, 0)
                 + ref.dslash_staggered_naik_parity(lng_g, src_g[1], gg, 0))
        truth_lex = torch.zeros((gg.volume, 3), dtype=torch.complex128)
        truth_lex[gg.lex_of_cb[0]] = truth
        truth_loc = checkerboard_split(loc(truth_lex), lg)[0]
        err = (out.to_complex()[0] - truth_loc).abs().max().item()
        assert err < 1e-12, f"rank{rank} improved dslash err={err}"
    finally:
        dist.destroy_process_group()


def test_improved_multiproc_gloo(tmp_path):
    import torch.multiprocessing as mp
    init_file = str(tmp_path / "init_imp")
    mp.spawn(_worker_improved, args=(2, init_file), nprocs=2, join=True)


def _worker_improved_2d(rank, world, init_file):
    import torch.distributed as dist
    from quda_amd.parallel import comms
    from quda_amd.fields.geometry import checkerboard_split
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        grid = (1, 1, 2, 2)
        comms.init_comms(grid=grid)
        GD = (4, 4, 8, 8)
        gg = LatticeGeometry(GD)
        gen = torch.Generator().manual_seed(411)
        from quda_amd.fields.gauge import project_su3
        fat_lex = project_su3(torch.view_as_complex(torch.randn(
            (4, gg.volume, 3, 3, 2), generator=gen, dtype=torch.float64)))
        lng_lex = project_su3(torch.view_as_complex(torch.randn(
            (4, gg.volume, 3, 3, 2), generator=gen, dtype=torch.float64)))
        src_lex = torch.view_as_complex(torch.randn(
            (gg.volume, 3, 2), generator=gen, dtype=torch.float64))
        ldims = tuple(GD[i] // grid[i] for i in range(4))
        lg = LatticeGeometry(ldims)
        coords = comms.grid_coords()
        off = torch.tensor([coords[i] * ldims[i] for i in range(4)])
        c = lg.coords.to(torch.int64) + off
        X, Y, Z, _ = GD
        glex = ((c[:, 3] * Z + c[:, 2]) * Y + c[:, 1]) * X + c[:, 0]
        fat_loc = checkerboard_split(fat_lex.movedim(0, 1)[glex], lg
                                     ).permute(2, 0, 1, 3, 4).contiguous()
        lng_loc = checkerboard_split(lng_lex.movedim(0, 1)[glex], lg
                                     ).permute(2, 0, 1, 3, 4).contiguous()
        gf = GaugeField(lg, "double").from_complex(fat_loc)
        gl = GaugeField(lg, "double", shift=3).from_complex(lng_loc)
        src = SpinorField(lg, "double", nspin=1)
        src.from_complex(checkerboard_split(src_lex[glex], lg))
        out = SpinorField(lg, "double", n_parity=1, nspin=1)
        dslash_staggered(out, src.parity_view(1), gf, 0, long_gauge=gl)
        fat_g = checkerboard_split(fat_lex.movedim(0, 1), gg).permute(
            2, 0, 1, 3, 4).contiguous()
        lng_g = checkerboard_split(lng_lex.movedim(0, 1), gg).permute(
            2, 0, 1, 3, 4).contiguous()
        src_g = checkerboard_split(src_lex, gg)
        truth = (ref.dslash_staggered_parity(fat_g, src_g[1], gg, 0)
                 + ref.dslash_staggered_naik_parity(lng_g, src_g[1], gg, 0))
        truth_lex = torch.zeros((gg.volume, 3), dtype=torch.complex128)
        truth_lex[gg.lex_of_cb[0]] = truth
        want = checkerboard_split(truth_lex[glex], lg)[0]
        err = (out.to_complex()[0] - want).abs().max().item()
        assert err < 1e-12, f"rank{rank} err={err}"
    finally:
        dist.destroy_process_group()


def test_improved_multiproc_2d_grid_gloo(tmp_path):
    """4 ranks on a (1,1,2,2) grid: depth-3 Naik halos across TWO
    partitioned dims simultaneously (corner-adjacent exchange paths)."""
    import torch.multiprocessing as mp
    init_file = str(tmp_path / "init_imp2d")
    mp.spawn(_worker_improved_2d, args=(4, init_file), nprocs=4, join=True)
