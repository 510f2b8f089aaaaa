"""Laplace/covdev/Wuppertal tests (ref: laplace + covariant_derivative
kernels, LapH workflows)."""
import numpy as np
import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.models.laplace import (GaugeLaplace, covdev_apply,
                                     laplace_apply, wuppertal_smear)
from quda_amd.ops import blas
from quda_amd.solvers import cg_solve
from quda_amd.solvers.eigen import trlm_solve


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=241)
    return geo, g


def test_laplace_hermitian_nsd(setup):
    geo, g = setup
    a = SpinorField(geo, "double").gaussian_(seed=242)
    b = SpinorField(geo, "double").gaussian_(seed=243)
    La = SpinorField(geo, "double")
    Lb = SpinorField(geo, "double")
    laplace_apply(g, a, La)
    laplace_apply(g, b, Lb)
    lhs = (b.to_complex().conj() * La.to_complex()).sum()
    rhs = (Lb.to_complex().conj() * a.to_complex()).sum()
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)
    # negative semi-definite
    quad = (a.to_complex().conj() * La.to_complex()).sum().real
    assert quad < 1e-10


def test_covdev_unitary(setup):
    geo, g = setup
    a = SpinorField(geo, "double").gaussian_(seed=244)
    f = covdev_apply(g, a, 2, forward=True)
    assert abs(blas.norm2(f) - blas.norm2(a)) < 1e-8
    # backward of forward = identity
    back = covdev_apply(g, f, 2, forward=False)
    assert (back.to_complex() - a.to_complex()).abs().max().item() < 1e-12


def test_laplace_cg_solve(setup):
    geo, g = setup
    op = GaugeLaplace(g, m2=0.5)
    b = SpinorField(geo, "double").gaussian_(seed=245)
    x = SpinorField(geo, "double")
    st = cg_solve(op, x, b, tol=1e-10, maxiter=500)
    assert st.converged


def test_laplace_eigensolve(setup):
    """Lowest Laplace eigenmodes (the LapH basis construction)."""
    geo, g = setup
    op = GaugeLaplace(g, m2=1.0)
    x0 = SpinorField(geo, "double", n_parity=2, nspin=1)

    class Wrap:
        def MdagM(self, out, inp, tmp):
            return laplace_apply(g, inp, out, a=-1.0, b=1.0)

    res = trlm_solve(Wrap(), n_ev=4, n_kr=20, x0=x0, tol=1e-7,
                     max_restarts=200)
    assert res.converged
    assert all(v > 0 for v in res.evals)


def test_wuppertal_preserves_norm_roughly_and_smooths(setup):
    geo, g = setup
    a = SpinorField(geo, "double")
    # point source
    c = torch.zeros((2, geo.volume_cb, 4, 3), dtype=torch.complex128)
    c[0, 0, 0, 0] = 1.0
    a.from_complex(c)
    sm = wuppertal_smear(g, a, alpha=3.0, n_steps=5)
    c2 = sm.to_complex()
    # support spread beyond the source site
    assert (c2.abs() > 1e-8).sum().item() > 100
    # norm decreases (averaging) but stays finite
    n = blas.norm2(sm)
    assert 0 < n <= blas.norm2(a) + 1e-12
