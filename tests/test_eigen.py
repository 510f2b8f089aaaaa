"""Eigensolver tests: TRLM/IRAM vs dense numpy eigensolve of the same
operator on a tiny lattice (the reference validates eigensolve_test
against residual norms; we additionally have exact dense spectra)."""
import math

import numpy as np
import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.models import DiracWilson, DiracWilsonPC
from quda_amd.ops import blas
from quda_amd.solvers import cg_solve
from quda_amd.solvers.eigen import (ChebyshevOp, Deflation, EigResult,
                                    iram_solve, trlm_solve)

KAPPA = 0.1


@pytest.fixture(scope="module")
def small():
    geo = LatticeGeometry((4, 4, 2, 2))
    g = GaugeField(geo, "double").random_su3_(seed=91)
    d = DiracWilsonPC(g, KAPPA)
    return geo, g, d


def dense_matrix(d, geo, n_parity=1, nspin=4, op="MdagM"):
    """Build the dense operator matrix column by column."""
    dim = geo.volume_cb * n_parity * nspin * 3
    A = np.zeros((dim, dim), dtype=complex)
    x = SpinorField(geo, "double", n_parity=n_parity, nspin=nspin)
    y = SpinorField(geo, "double", n_parity=n_parity, nspin=nspin)
    t = SpinorField(geo, "double", n_parity=n_parity, nspin=nspin)
    for j in range(dim):
        c = torch.zeros(dim, dtype=torch.complex128)
        c[j] = 1.0
        x.from_complex(c.reshape(n_parity, geo.volume_cb, *x.site_shape))
        if op == "MdagM":
            d.MdagM(y, x, t)
        else:
            d.M(y, x)
        A[:, j] = y.to_complex().reshape(-1).numpy()
    return A


def test_trlm_vs_dense(small):
    geo, g, d = small
    A = dense_matrix(d, geo)
    w_exact = np.sort(np.linalg.eigvalsh(A))
    x0 = SpinorField(geo, "double", n_parity=1)
    res = trlm_solve(d, n_ev=6, n_kr=24, x0=x0, tol=1e-8, max_restarts=200)
    assert res.converged
    got = np.sort(res.evals)
    assert np.allclose(got, w_exact[:6], rtol=1e-6), (got, w_exact[:6])
    # residuals small, vectors orthonormal
    for i, v in enumerate(res.evecs):
        assert res.residuals[i] < 1e-5
        for j in range(i):
            assert abs(blas.c_dot(res.evecs[j], v)) < 1e-6


def test_trlm_largest(small):
    geo, g, d = small
    A = dense_matrix(d, geo)
    w_exact = np.sort(np.linalg.eigvalsh(A))
    x0 = SpinorField(geo, "double", n_parity=1)
    res = trlm_solve(d, n_ev=4, n_kr=16, x0=x0, tol=1e-8, which="largest",
                     max_restarts=200)
    got = np.sort(res.evals)
    assert np.allclose(got, w_exact[-4:], rtol=1e-6)


def test_trlm_chebyshev_accelerated(small):
    geo, g, d = small
    A = dense_matrix(d, geo)
    w_exact = np.sort(np.linalg.eigvalsh(A))
    lam_max = w_exact[-1] * 1.05
    x0 = SpinorField(geo, "double", n_parity=1)
    poly = ChebyshevOp(d, a_min=w_exact[7] * 1.1, a_max=lam_max, degree=8)
    res = trlm_solve(d, n_ev=4, n_kr=16, x0=x0, tol=1e-8, poly=poly,
                     which="largest", max_restarts=100)
    # p(A) ordering: smallest A-eigenvalues are amplified -> 'largest' of
    # p(A) are the smallest of A; evals recomputed as Rayleigh quotients
    got = np.sort(res.evals)
    assert np.allclose(got, w_exact[:4], rtol=1e-5), (got, w_exact[:4])


def test_iram_vs_dense_extremal(small):
    """Largest-|.| modes of the non-hermitian Wilson M (well separated);
    eigenvalues must match the dense spectrum and residuals be small."""
    geo, g, _ = small
    d = DiracWilson(g, KAPPA)
    A = dense_matrix(d, geo, n_parity=2, op="M")
    w = np.linalg.eigvals(A)
    w_big = w[np.argsort(-np.abs(w))][:4]
    x0 = SpinorField(geo, "double", n_parity=2)
    res = iram_solve(d, n_ev=4, n_kr=20, x0=x0, tol=1e-8, max_restarts=300,
                     which="largest_abs")
    assert res.converged
    key = lambda z: (round(-abs(z), 7), round(z.imag, 7))
    got = sorted(res.evals, key=key)
    exact = sorted(w_big, key=key)
    for a, b in zip(got, exact):
        assert abs(a - b) < 1e-6 * max(1.0, abs(b)), (got, exact)
    assert max(res.residuals) < 1e-6


def test_iram_smallest_residuals(small):
    """Interior (smallest-|.|) search on a clustered spectrum: converged
    Ritz pairs must at least be true approximate eigenpairs (small
    residual ||M v - lambda v||)."""
    geo, g, _ = small
    d = DiracWilson(g, KAPPA)
    x0 = SpinorField(geo, "double", n_parity=2)
    res = iram_solve(d, n_ev=4, n_kr=24, x0=x0, tol=1e-6, max_restarts=200)
    assert max(res.residuals) < 1e-4


def test_deflated_cg_fewer_iters(small):
    geo, g, _ = small
    d = DiracWilsonPC(g, 0.135)  # near-critical: ill-conditioned MdagM
    b = SpinorField(geo, "double", n_parity=1).gaussian_(seed=92)
    x_plain = SpinorField(geo, "double", n_parity=1)
    st0 = cg_solve(d, x_plain, b, tol=1e-10, maxiter=1000)
    x0 = SpinorField(geo, "double", n_parity=1)
    res = trlm_solve(d, n_ev=8, n_kr=28, x0=x0, tol=1e-8, max_restarts=200)
    defl = Deflation(res.evals, res.evecs)
    x_defl = SpinorField(geo, "double", n_parity=1)
    defl.guess(x_defl, b)
    # the deflated guess removes the low-mode content of the residual:
    # r0 = b - A x0 must be orthogonal to the deflated eigenvectors
    r0 = SpinorField(geo, "double", n_parity=1)
    t = SpinorField(geo, "double", n_parity=1)
    d.MdagM(r0, x_defl, t)
    blas.xmy_norm2(b, r0)
    for lam, v in zip(res.evals, res.evecs):
        assert abs(blas.c_dot(v, r0)) < 1e-6 * math.sqrt(blas.norm2(b))
    st1 = cg_solve(d, x_defl, b, tol=1e-10, maxiter=1000)
    assert st1.converged
    assert st1.iters <= st0.iters


def test_block_trlm_matches_trlm():
    """Block TRLM finds the same lowest spectrum as scalar TRLM."""
    from quda_amd.models import DiracWilson
    from quda_amd.solvers.eigen import block_trlm_solve, trlm_solve
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=71)
    d = DiracWilson(g, 0.12)
    r1 = trlm_solve(d, 6, 24, SpinorField(geo, "double"), tol=1e-8)
    r2 = block_trlm_solve(d, 6, 32, SpinorField(geo, "double"),
                          block_size=4, tol=1e-8, max_restarts=200)
    assert r1.converged and r2.converged
    for a, b in zip(r1.evals, r2.evals):
        assert abs(a - b) < 1e-7 * max(abs(a), 1e-10), (a, b)
    assert all(r < 1e-6 for r in r2.residuals), r2.residuals


def test_trlm_on_g5m_indefinite():
    """TRLM on the hermitian INDEFINITE g5 M: both spectral edges are
    reachable and eigenpairs satisfy the residual check (the role of
    QUDA's gamma5-hermitian eigensolves)."""
    from quda_amd.models import DiracG5M, DiracWilson
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=681)
    h = DiracG5M(DiracWilson(g, 0.12))

    class _Sq:
        """TRLM consumes MdagM; feed it (g5 M)^2 and recover +-lambda."""
        def MdagM(self, out, inp, tmp):
            return h.MdagM(out, inp, tmp)

    from quda_amd.solvers.eigen import trlm_solve
    r = trlm_solve(_Sq(), 4, 24, SpinorField(geo, "double"), tol=1e-8)
    assert r.converged
    # eigenvalues of (g5M)^2 are lambda^2 > 0; verify the smallest
    # eigenvector is a +-lambda eigenvector of g5M itself
    lam2, v = r.evals[0], r.evecs[0]
    Hv = SpinorField(geo, "double")
    h.M(Hv, v)
    lam = (v.to_complex().conj() * Hv.to_complex()).sum().real.item()
    assert abs(lam * lam - lam2) < 1e-5 * lam2
    t = SpinorField(geo, "double")
    from quda_amd.ops import blas
    blas.copy(t, Hv)
    blas.axpy(-lam, v, t)
    import math
    assert math.sqrt(blas.norm2(t)) < 1e-3


def test_iram_on_kd_operator_vs_dense():
    """Krylov-Schur Arnoldi on the origin-wrapping nonsymmetric KD
    operator: largest-|lambda| eigenvalues match the dense spectrum."""
    import numpy as np
    from quda_amd.models import DiracStaggeredKD
    from quda_amd.solvers.eigen import iram_solve
    geo = LatticeGeometry((2, 2, 2, 4))
    g = GaugeField(geo, "double").random_su3_(seed=685)
    kd = DiracStaggeredKD(g, 0.05)
    n = 2 * geo.volume_cb * 3
    A = np.zeros((n, n), dtype=complex)
    e = SpinorField(geo, "double", nspin=1)
    o = SpinorField(geo, "double", nspin=1)
    import torch
    for j in range(n):
        c = torch.zeros((2, geo.volume_cb, 3), dtype=torch.complex128)
        c.view(-1)[j] = 1.0
        e.from_complex(c)
        kd.M(o, e)
        A[:, j] = o.to_complex().reshape(-1).numpy()
    w = np.linalg.eigvals(A)
    w_big = sorted(np.abs(w))[-4:]
    x0 = SpinorField(geo, "double", nspin=1)
    r = iram_solve(kd, 4, 20, x0, tol=1e-8, which="largest_abs")
    got = sorted(abs(complex(v)) for v in r.evals)
    for a, b in zip(got, w_big):
        assert abs(a - b) < 1e-6 * b, (got, w_big)
