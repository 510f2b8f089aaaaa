"""Multigrid tests (analogue of the reference's MG verify() checks +
multigrid_benchmark_test convergence behaviour)."""
import math

import numpy as np
import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.mg import MG, MGParam, Transfer, build_coarse_op
from quda_amd.mg.coarse import coarse_bicgstab
from quda_amd.models import DiracClover, DiracWilson
from quda_amd.ops import blas
from quda_amd.ops import reference as ref
from quda_amd.solvers import gcr_solve

KAPPA = 0.145  # near-critical for a random 4^4 field: hard system


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=121)
    d = DiracWilson(g, KAPPA)
    mg = MG(d, MGParam(block=(2, 2, 2, 2), n_vec=4, nu_post=4,
                       coarse_tol=5e-2, null_tol=1e-4, null_maxiter=300))
    return geo, g, d, mg


def test_transfer_identities(setup):
    geo, g, d, mg = setup
    out = mg.verify()
    assert out["RP_identity"] < 1e-12, out
    assert out["galerkin"] < 1e-10, out


def test_null_vectors_preserved(setup):
    """(1 - P R) v ~ 0 for the (block-orthonormalized) null vectors."""
    t = mgt = setup[3].transfer
    # check on the stored V itself: prolong(restrict(V_k)) == V_k
    for k in range(t.nvec):
        fine = setup[2].new_spinor(n_parity=2)
        # rebuild fine field from V column k
        Na, B = t.n_agg, t.block_vol
        col = t.V[:, :, :, :, k]
        from quda_amd.fields.geometry import checkerboard_split
        V = t.geo.volume
        lex = torch.empty((V, 4, 3), dtype=col.dtype)
        lex[t.sites_by_agg.reshape(-1)] = col.reshape(V, 4, 3)
        fine.from_complex(checkerboard_split(lex, t.geo))
        c = t.restrict(fine)
        back = setup[2].new_spinor(n_parity=2)
        t.prolong(c, back)
        err = (back.to_complex() - fine.to_complex()).abs().max().item()
        assert err < 1e-10, (k, err)


def test_coarse_dagger_adjoint(setup):
    co = setup[3].coarse
    gen = torch.Generator().manual_seed(2)
    a = torch.view_as_complex(torch.randn((co.Na, co.Nc, 2), generator=gen,
                                          dtype=torch.float64))
    b = torch.view_as_complex(torch.randn((co.Na, co.Nc, 2), generator=gen,
                                          dtype=torch.float64))
    lhs = (b.conj() * co.apply(a)).sum()
    rhs = (co.apply(b, dagger=True).conj() * a).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)


def test_coarse_bicgstab_solves(setup):
    co = setup[3].coarse
    gen = torch.Generator().manual_seed(3)
    b = torch.view_as_complex(torch.randn((co.Na, co.Nc, 2), generator=gen,
                                          dtype=torch.float64))
    x = coarse_bicgstab(co, b, tol=1e-10, maxiter=2000)
    r = b - co.apply(x)
    assert (r.conj() * r).sum().real.item() < 1e-16 * (b.conj() * b).sum().real.item() * 1e4


def test_mg_preconditioner_beats_plain_gcr(setup):
    geo, g, d, mg = setup
    b = SpinorField(geo, "double").gaussian_(seed=122)
    x0 = SpinorField(geo, "double")
    st_plain = gcr_solve(d, x0, b, tol=1e-8, maxiter=400, nkrylov=16)
    x1 = SpinorField(geo, "double")
    st_mg = gcr_solve(d, x1, b, tol=1e-8, maxiter=400, nkrylov=16,
                      precond=mg.precond)
    assert st_mg.converged
    assert st_mg.iters < st_plain.iters / 2, (st_mg.iters, st_plain.iters)
    # true residual
    r = SpinorField(geo, "double")
    d.M(r, x1)
    tr = math.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b))
    assert tr < 1e-7


def test_mg_clover(setup):
    """MG on the clover operator (clover enters the coarse X)."""
    geo, g, _, _ = setup
    u = g.to_complex()
    from quda_amd.fields.clover import CloverField
    A = ref.clover_matrix(u, geo, KAPPA, 1.0)
    cl = CloverField(geo, "double").from_matrices(A)
    d = DiracClover(g, cl, KAPPA)
    mg = MG(d, MGParam(block=(2, 2, 2, 2), n_vec=4, nu_post=4,
                       null_tol=1e-4, null_maxiter=300))
    chk = mg.verify()
    assert chk["galerkin"] < 1e-10, chk
    b = SpinorField(geo, "double").gaussian_(seed=123)
    x = SpinorField(geo, "double")
    st = gcr_solve(d, x, b, tol=1e-8, maxiter=200, nkrylov=16,
                   precond=mg.precond)
    assert st.converged


@pytest.mark.gpu
def test_mg_on_gpu():
    """Whole MG stack on device: HIP dslash for null vectors/smoothing,
    rocBLAS batched GEMM for transfer/coarse apply."""
    geo = LatticeGeometry((8, 8, 8, 8))
    gen = torch.Generator().manual_seed(221)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64)
    u = project_su3(torch.view_as_complex(m)).cuda()
    g = GaugeField(geo, "double", "cuda").from_complex(u)
    d = DiracWilson(g, 0.14)
    mg = MG(d, MGParam(block=(2, 2, 2, 2), n_vec=4, nu_post=4,
                       null_tol=1e-4, null_maxiter=200))
    chk = mg.verify()
    assert chk["galerkin"] < 1e-9, chk
    b = SpinorField(geo, "double", "cuda").gaussian_(seed=222)
    x0 = SpinorField(geo, "double", "cuda")
    st_plain = gcr_solve(d, x0, b, tol=1e-8, maxiter=300, nkrylov=16)
    x1 = SpinorField(geo, "double", "cuda")
    st_mg = gcr_solve(d, x1, b, tol=1e-8, maxiter=300, nkrylov=16,
                      precond=mg.precond)
    assert st_mg.converged
    assert st_mg.iters < st_plain.iters


def test_three_level_galerkin_and_convergence():
    """Level-2 Galerkin exactness (R2 A1 P2 == A2) + 3-level MG-GCR
    converges on an 8^4 system."""
    from quda_amd.mg.coarse_level import (CoarseMG, CoarseTransfer,
                                          build_coarse2_op,
                                          generate_coarse_null_vectors)
    geo = LatticeGeometry((8, 8, 8, 8))
    g = GaugeField(geo, "double").random_su3_(seed=271)
    d = DiracWilson(g, 0.14)
    mg = MG(d, MGParam(block=(2, 2, 2, 2), n_vec=4, nu_post=4, levels=3,
                       block2=(2, 2, 2, 2), n_vec2=4,
                       null_tol=1e-4, null_maxiter=200))
    co, cmg = mg.coarse, mg.coarse_mg
    # Galerkin exactness at level 2
    gen = torch.Generator().manual_seed(272)
    c = torch.view_as_complex(torch.randn((co.Na, co.Nc, 2), generator=gen,
                                          dtype=torch.float64))
    # A2 c2 must equal R2 A1 P2 c2
    c2 = cmg.t2.restrict(c)
    lhs = cmg.t2.restrict(co.apply(cmg.t2.prolong(c2)))
    rhs = cmg.co2.apply(c2)
    assert (lhs - rhs).abs().max().item() < 1e-10
    # R2 P2 = 1
    back = cmg.t2.restrict(cmg.t2.prolong(c2))
    assert (back - c2).abs().max().item() < 1e-12
    # convergence of the full 3-level preconditioner
    b = SpinorField(geo, "double").gaussian_(seed=273)
    x = SpinorField(geo, "double")
    st = gcr_solve(d, x, b, tol=1e-8, maxiter=300, nkrylov=16,
                   precond=mg.precond)
    assert st.converged
    x0 = SpinorField(geo, "double")
    st_plain = gcr_solve(d, x0, b, tol=1e-8, maxiter=300, nkrylov=16)
    assert st.iters < st_plain.iters


# ---------------------------------------------------------------------------
# staggered transfer + Galerkin coarse machinery (KD-basis cycle = round 2)
# ---------------------------------------------------------------------------

def test_staggered_transfer_verify():
    """Parity-blocked staggered transfer: R P = identity on the coarse
    space and the dense Galerkin coarse op is consistent (A_c c = R A P c)
    — the MG::verify() checks of the reference (multigrid.cpp:762)."""
    from quda_amd.models import DiracStaggered
    from quda_amd.mg.staggered import StaggeredMG
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=606)
    d = DiracStaggered(g, 0.05)
    mg = StaggeredMG(d, geo, block=(2, 2, 2, 2), n_vec=6)
    v = mg.verify()
    assert v["RP_identity"] < 1e-12, v
    assert v["galerkin"] < 1e-12, v


def test_staggered_coarse_op_epsilon_hermiticity():
    """The parity blocking preserves epsilon-hermiticity on the coarse
    level: E A_c E = A_c^dag with E = diag(+1 even-block, -1 odd-block)."""
    import numpy as np
    from quda_amd.models import DiracStaggered
    from quda_amd.mg.staggered import StaggeredMG
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=607)
    d = DiracStaggered(g, 0.05)
    mg = StaggeredMG(d, geo, block=(2, 2, 2, 2), n_vec=4)
    A = mg.coarse.A.numpy()
    na, nv = mg.transfer.n_agg, mg.transfer.nvec
    E = np.zeros(na * 2 * nv)
    E.reshape(na, 2, nv)[:, 0, :] = 1.0
    E.reshape(na, 2, nv)[:, 1, :] = -1.0
    lhs = (E[:, None] * A) * E[None, :]
    err = np.abs(lhs - A.conj().T).max()
    assert err < 1e-10, err


def test_staggered_null_vectors_are_near_null():
    from quda_amd.models import DiracStaggered
    from quda_amd.mg.staggered import generate_stag_null_vectors
    from quda_amd.ops import blas
    import math
    geo = LatticeGeometry((4, 4, 4, 8))
    g = GaugeField(geo, "double").random_su3_(seed=608)
    d = DiracStaggered(g, 0.01)
    vs = generate_stag_null_vectors(d, 3, geo)
    for v in vs:
        w = SpinorField(geo, "double", nspin=1)
        d.M(w, v)
        assert math.sqrt(blas.norm2(w)) < 0.1  # ||v|| = 1


def test_mg_survives_gauge_evolution():
    """updateMultigridQuda role across real HMC evolution: evolve the
    resident gauge a few MD steps, refresh the hierarchy, and the
    MG-preconditioned solve still converges on the NEW links
    (multigrid_evolve_test analogue)."""
    from quda_amd import api
    from quda_amd.api import (DslashType, GaugeParam, InvertParam,
                              InverterType)
    from quda_amd.fields.gauge import GaugeField
    from quda_amd.fields.geometry import LatticeGeometry
    import torch
    geo = LatticeGeometry((4, 4, 4, 4))
    u = GaugeField(geo, "double").random_su3_(seed=695).to_complex()
    gp = GaugeParam(X=(4, 4, 4, 4), device="cpu", cuda_prec="double",
                    cuda_prec_sloppy="double")
    api.init_quda()
    api.load_gauge_quda(u, gp)
    p = InvertParam(dslash_type=DslashType.WILSON, kappa=0.12,
                    inv_type=InverterType.GCR, tol=1e-8, maxiter=400)
    mg = api.new_multigrid_quda(p, block=(2, 2, 2, 2), n_vec=4)
    # evolve: a few leapfrog steps on the resident field
    P = api.gauss_mom_quda(seed=696)
    for _ in range(3):
        F = api.compute_gauge_force_quda(5.5)
        P = P + 0.05 * F
        api.update_gauge_field_quda(P, 0.05)
    api.update_multigrid_quda(mg, p)
    p2 = InvertParam(**{**p.__dict__, "preconditioner": mg.precond})
    g = torch.Generator().manual_seed(697)
    b = torch.view_as_complex(torch.randn((2, 128, 4, 3, 2), generator=g,
                                          dtype=torch.float64))
    api.invert_quda(b, p2)
    assert p2.true_res < 1e-7, p2.true_res
    api.end_quda()


# ---------------------------------------------------------------------------
# Multi-rank multigrid: the distributed coarse op, V-cycle
# and 3-level recursion must match the single-rank build of the same
# GLOBAL system (ref: coarse ghost machinery, lib/dslash_coarse.hpp:30).
# ---------------------------------------------------------------------------

def _mg_worker(rank, world, grid, init_file):
    import torch.distributed as dist
    from quda_amd.fields.geometry import (checkerboard_join,
                                          checkerboard_split)
    from quda_amd.mg.coarse_level import CoarseMG
    from quda_amd.parallel import comms
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        comms.init_comms(grid=grid)
        GD = (4, 4, 4, 8)
        gg = LatticeGeometry(GD)
        gen = torch.Generator().manual_seed(411)
        m = torch.randn((4, 2, gg.volume_cb, 3, 3, 2), generator=gen,
                        dtype=torch.float64)
        from quda_amd.fields.gauge import project_su3
        u_g = project_su3(torch.view_as_complex(m))
        kappa = 0.14

        ldims = tuple(GD[i] // grid[i] for i in range(4))
        lg = LatticeGeometry(
            ldims, parity_offset=comms.parity_offset_of_rank(ldims))
        coords = comms.grid_coords()
        off = torch.tensor([coords[i] * ldims[i] for i in range(4)])

        def local_lex(lex_field):
            c = lg.coords.to(torch.int64) + off
            X, Y, Z, _ = GD
            glex = ((c[:, 3] * Z + c[:, 2]) * Y + c[:, 1]) * X + c[:, 0]
            return lex_field[glex]

        # per-direction lex join of the global links, sliced to this rank
        u_loc = torch.empty((4, 2, lg.volume_cb, 3, 3),
                            dtype=torch.complex128)
        for mu in range(4):
            lexd = checkerboard_join(u_g[mu], gg)  # [V,3,3]
            loc = local_lex(lexd)
            u_loc[mu] = checkerboard_split(loc, lg)

        g_loc = GaugeField(lg, "double").from_complex(u_loc)
        d_loc = DiracWilson(g_loc, kappa)

        # single-rank global truth (each rank computes it identically)
        with comms.solo_mode():
            g_glob = GaugeField(gg, "double").from_complex(u_g)
            d_glob = DiracWilson(g_glob, kappa)
            from quda_amd.mg.transfer import generate_null_vectors
            vecs_g = generate_null_vectors(d_glob, 3, tol=1e-3,
                                           maxiter=150, seed=500)
            mg_g = MG(d_glob, MGParam(block=(2, 2, 2, 2), n_vec=3,
                                      nu_post=2, coarse_tol=1e-2,
                                      coarse_maxiter=100),
                      vectors=vecs_g)

        # distributed build from the SLICED global null vectors
        vecs_l = []
        for v in vecs_g:
            lexv = checkerboard_join(v.to_complex(), gg)
            vecs_l.append(SpinorField(lg, "double").from_complex(
                checkerboard_split(local_lex(lexv), lg)))
        mg_l = MG(d_loc, MGParam(block=(2, 2, 2, 2), n_vec=3, nu_post=2,
                                 coarse_tol=1e-2, coarse_maxiter=100),
                  vectors=vecs_l)

        # (a) coarse tensors: local X/Y equal the global slice
        cd_g = mg_g.transfer.coarse_dims
        cd_l = mg_l.transfer.coarse_dims
        cidx = torch.arange(cd_l[0] * cd_l[1] * cd_l[2] * cd_l[3])
        cc = torch.stack([cidx % cd_l[0],
                          (cidx // cd_l[0]) % cd_l[1],
                          (cidx // (cd_l[0] * cd_l[1])) % cd_l[2],
                          cidx // (cd_l[0] * cd_l[1] * cd_l[2])], dim=1)
        coff = torch.tensor([coords[i] * cd_l[i] for i in range(4)])
        gc = cc + coff
        c_glex = (((gc[:, 3] * cd_g[2] + gc[:, 2]) * cd_g[1] + gc[:, 1])
                  * cd_g[0] + gc[:, 0])
        errX = (mg_l.coarse.X - mg_g.coarse.X[c_glex]).abs().max().item()
        assert errX < 1e-11, f"rank{rank} X err {errX}"
        for dd in range(8):
            errY = (mg_l.coarse.Y[dd]
                    - mg_g.coarse.Y[dd][c_glex]).abs().max().item()
            assert errY < 1e-11, f"rank{rank} Y[{dd}] err {errY}"

        # (b) coarse apply + dagger match the global apply's local slice
        genc = torch.Generator().manual_seed(97)
        cg_vec = torch.view_as_complex(
            torch.randn((mg_g.coarse.Na, mg_g.coarse.Nc, 2), generator=genc,
                        dtype=torch.float64))
        cl_vec = cg_vec[c_glex].clone()
        for dag in (False, True):
            with comms.solo_mode():
                want = mg_g.coarse.apply(cg_vec, dagger=dag)
            got = mg_l.coarse.apply(cl_vec, dagger=dag)
            err = (got - want[c_glex]).abs().max().item()
            assert err < 1e-11, f"rank{rank} apply dag={dag} err {err}"

        # (c) full V-cycle matches the single-rank V-cycle
        genr = torch.Generator().manual_seed(98)
        r_lex = torch.view_as_complex(
            torch.randn((gg.volume, 4, 3, 2), generator=genr,
                        dtype=torch.float64))
        r_g = SpinorField(gg, "double").from_complex(
            checkerboard_split(r_lex, gg))
        r_l = SpinorField(lg, "double").from_complex(
            checkerboard_split(local_lex(r_lex), lg))
        z_g = SpinorField(gg, "double")
        z_l = SpinorField(lg, "double")
        with comms.solo_mode():
            mg_g.precond(z_g, r_g)
        mg_l.precond(z_l, r_l)
        zg_lex = checkerboard_join(z_g.to_complex(), gg)
        err = (z_l.to_complex()
               - checkerboard_split(local_lex(zg_lex), lg)).abs().max().item()
        assert err < 1e-8, f"rank{rank} V-cycle err {err}"

        # (d) 3-level: level-2 Galerkin tensors match the global build
        from quda_amd.mg.coarse_level import (CoarseTransfer,
                                              build_coarse2_op,
                                              generate_coarse_null_vectors)
        with comms.solo_mode():
            v2_g = generate_coarse_null_vectors(mg_g.coarse, 2, tol=1e-2,
                                                maxiter=60, seed=700)
            cmg_g = CoarseMG(mg_g.coarse, block2=(2, 2, 2, 2), n_vec2=2,
                             vectors=v2_g)
        v2_l = [v[c_glex].clone() for v in v2_g]
        cmg_l = CoarseMG(mg_l.coarse, block2=(2, 2, 2, 2), n_vec2=2,
                         vectors=v2_l)
        cd2_g, cd2_l = cmg_g.t2.cd2, cmg_l.t2.cd2
        c2idx = torch.arange(cd2_l[0] * cd2_l[1] * cd2_l[2] * cd2_l[3])
        c2c = torch.stack([c2idx % cd2_l[0],
                           (c2idx // cd2_l[0]) % cd2_l[1],
                           (c2idx // (cd2_l[0] * cd2_l[1])) % cd2_l[2],
                           c2idx // (cd2_l[0] * cd2_l[1] * cd2_l[2])], dim=1)
        c2off = torch.tensor([coords[i] * cd2_l[i] for i in range(4)])
        g2 = c2c + c2off
        c2_glex = (((g2[:, 3] * cd2_g[2] + g2[:, 2]) * cd2_g[1] + g2[:, 1])
                   * cd2_g[0] + g2[:, 0])
        errX2 = (cmg_l.co2.X - cmg_g.co2.X[c2_glex]).abs().max().item()
        assert errX2 < 1e-10, f"rank{rank} X2 err {errX2}"
        for dd in range(8):
            errY2 = (cmg_l.co2.Y[dd]
                     - cmg_g.co2.Y[dd][c2_glex]).abs().max().item()
            assert errY2 < 1e-10, f"rank{rank} Y2[{dd}] err {errY2}"
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("grid,world", [((1, 1, 1, 2), 2)])
def test_multirank_mg_matches_global(grid, world):
    import os
    import tempfile

    import torch.multiprocessing as mp
    with tempfile.NamedTemporaryFile(delete=False) as f:
        init_file = f.name
    os.unlink(init_file)
    mp.spawn(_mg_worker, args=(world, grid, init_file), nprocs=world,
             join=True)


@pytest.mark.gpu
@pytest.mark.parametrize("nr", [1, 8, 16])
def test_coarse_mfma_vs_torch_gpu(nr):
    """k_coarse_dslash_mfma (f32 matrix cores) vs the torch einsum path,
    including the self-wraparound ghost path (forced partition)."""
    from quda_amd.mg.coarse import CoarseOp
    from quda_amd.parallel import comms
    gen = torch.Generator().manual_seed(314)
    cd = (4, 4, 4, 4)
    Na, Nc = 256, 16
    X = torch.view_as_complex(torch.randn((Na, Nc, Nc, 2), generator=gen,
                                          dtype=torch.float64)).cuda()
    Y = [torch.view_as_complex(torch.randn((Na, Nc, Nc, 2), generator=gen,
                                           dtype=torch.float64)).cuda()
         for _ in range(8)]
    C = torch.view_as_complex(torch.randn((Na, Nc, nr, 2), generator=gen,
                                          dtype=torch.float64)).cuda()
    for mask in (0, 0b1010):
        if mask:
            comms.set_forced_partition(mask)
        try:
            co = CoarseOp(X, Y, cd, mask=mask)
            co.use_hip = False
            want = co.apply_block(C)
            co2 = CoarseOp(X, Y, cd, mask=mask)
            got = co2.apply_block(C)
            rel = ((got - want).abs().max()
                   / want.abs().max()).item()
            assert rel < 5e-6, (nr, mask, rel)
            # single-RHS entry point too
            got1 = co2.apply(C[:, :, 0])
            rel1 = ((got1 - want[:, :, 0]).abs().max()
                    / want.abs().max()).item()
            assert rel1 < 5e-6, (nr, mask, rel1)
        finally:
            comms.set_forced_partition(0)


def test_staggered_kd_free_spectrum_circle():
    """ROUND2_PLAN §3 derivation check: on the free field the
    Kahler-Dirac-preconditioned staggered operator X^-1 M has its whole
    spectrum on the Wilson-like circle |lambda - 1| = sqrt(1 - (2m)^2/..)
    ~ 1, tangent to zero from the right — the property that makes
    staggered MG coarsening contract (ref dirac_staggered_kd.cpp role)."""
    import numpy as np
    from quda_amd.models import DiracStaggered
    from quda_amd.models.staggered_kd import KDBlockInverse
    geo = LatticeGeometry((4, 4, 2, 2))
    u = torch.zeros((4, 2, geo.volume_cb, 3, 3), dtype=torch.complex128)
    for c in range(3):
        u[..., c, c] = 1.0
    m = 0.1
    d = DiracStaggered(GaugeField(geo, "double").from_complex(u), m)
    kd = KDBlockInverse(u, geo, m)

    def apply_op(f):
        s = SpinorField(geo, "double", n_parity=2, nspin=1)
        s.from_complex(f)
        w = SpinorField(geo, "double", n_parity=2, nspin=1)
        d.M(w, s)
        t = SpinorField(geo, "double", n_parity=2, nspin=1)
        kd.apply(t, w)
        return t.to_complex()

    n = geo.volume * 3
    cols = []
    E = torch.zeros((2, geo.volume_cb, 3), dtype=torch.complex128)
    for p in range(2):
        for i in range(geo.volume_cb):
            for c in range(3):
                E.zero_()
                E[p, i, c] = 1.0
                cols.append(apply_op(E).reshape(-1).numpy())
    A = np.array(cols).T
    ev = np.linalg.eigvals(A)
    assert ev.real.min() > 0, ev.real.min()           # right half plane
    r = np.abs(ev - 1.0)
    assert r.max() - r.min() < 1e-6, (r.min(), r.max())  # one circle
    assert abs(r.max() - np.sqrt(1 - ev.real.min() * (2 - ev.real.min()))) \
        < 1.0  # radius ~ <=1


def test_staggered_kd_mg_contracts():
    """VERDICT r1 #10 'Done' criterion: measured iteration reduction on
    the staggered operator — the V-cycle over the KD-preconditioned op
    (spectrum on the Wilson circle) contracts where coarsening plain
    staggered could not (round-1 finding, mg/staggered.py)."""
    from quda_amd.models.staggered_kd import DiracStaggeredKD
    from quda_amd.mg.staggered import StaggeredMG
    geo = LatticeGeometry((4, 4, 4, 8))
    g = GaugeField(geo, "double").random_su3_(seed=17)
    dkd = DiracStaggeredKD(g, 0.3)
    b = SpinorField(geo, "double", n_parity=2, nspin=1).gaussian_(seed=5)
    bp = dkd.prepare(b)
    x0 = SpinorField(geo, "double", n_parity=2, nspin=1)
    st_p = gcr_solve(dkd, x0, bp, tol=1e-8, maxiter=300, nkrylov=16)
    mg = StaggeredMG(dkd, geo, block=(2, 2, 2, 4), n_vec=8, n_smooth=4)
    x1 = SpinorField(geo, "double", n_parity=2, nspin=1)
    st_m = gcr_solve(dkd, x1, bp, tol=1e-8, maxiter=300, nkrylov=16,
                     precond=mg.precond)
    assert st_p.converged and st_m.converged
    assert st_m.iters * 2 < st_p.iters, (st_m.iters, st_p.iters)
    # solution solves the PLAIN staggered system
    from quda_amd.models import DiracStaggered
    dpl = DiracStaggered(g, 0.3)
    r = SpinorField(geo, "double", n_parity=2, nspin=1)
    dpl.M(r, x1)
    tr = math.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b))
    assert tr < 1e-6, tr
