"""BLAS layer identity coverage on CPU for every precision and ncomp
(role of the reference's blas_test; the GPU kernel equivalence is in
test_gpu_kernels)."""
import math

import pytest
import torch

from quda_amd import LatticeGeometry, SpinorField
from quda_amd.ops import blas


@pytest.fixture(params=["double", "single"])
def prec(request):
    return request.param


@pytest.fixture(params=[4, 1])
def nspin(request):
    return request.param


def mk(prec, nspin, seed):
    geo = LatticeGeometry((4, 4, 4, 4))
    return SpinorField(geo, prec, nspin=nspin).gaussian_(seed=seed)


def test_axpy_family(prec, nspin):
    x = mk(prec, nspin, 701)
    y = mk(prec, nspin, 702)
    y0 = y.to_complex().clone()
    blas.axpy(0.7, x, y)
    want = y0 + 0.7 * x.to_complex()
    tol = 1e-12 if prec == "double" else 1e-5
    assert (y.to_complex() - want).abs().max().item() < tol
    blas.xpay(x, -0.3, y)
    want = x.to_complex() - 0.3 * want
    assert (y.to_complex() - want).abs().max().item() < 3 * tol
    blas.axpby(0.2, x, 1.1, y)
    want = 0.2 * x.to_complex() + 1.1 * want
    assert (y.to_complex() - want).abs().max().item() < 6 * tol


def test_complex_family(prec, nspin):
    x = mk(prec, nspin, 703)
    y = mk(prec, nspin, 704)
    a = 0.3 - 0.8j
    b = -0.1 + 0.4j
    y0 = y.to_complex().clone()
    blas.caxpy(a, x, y)
    want = y0 + a * x.to_complex()
    tol = 1e-12 if prec == "double" else 1e-5
    assert (y.to_complex() - want).abs().max().item() < tol
    blas.caxpby(a, x, b, y)
    want = a * x.to_complex() + b * want
    assert (y.to_complex() - want).abs().max().item() < 3 * tol


def test_reductions(prec, nspin):
    x = mk(prec, nspin, 705)
    y = mk(prec, nspin, 706)
    xc, yc = x.to_complex(), y.to_complex()
    rtol = 1e-12 if prec == "double" else 1e-5
    assert abs(blas.norm2(x) - xc.abs().square().sum().item()) \
        < rtol * xc.abs().square().sum().item()
    cd = blas.c_dot(x, y)
    want = (xc.conj() * yc).sum()
    assert abs(cd - complex(want)) < rtol * abs(want) * 100
    rd = blas.re_dot(x, y)
    assert abs(rd - want.real.item()) < rtol * abs(want) * 100
    n2 = blas.xmy_norm2(x, y)
    assert abs(n2 - (xc - yc).abs().square().sum().item()) \
        < rtol * n2 * 100


def test_fused_axpy_norm2(prec, nspin):
    x = mk(prec, nspin, 707)
    y = mk(prec, nspin, 708)
    want_y = y.to_complex() - 0.4 * x.to_complex()
    n2 = blas.axpy_norm2(-0.4, x, y)
    tol = 1e-10 if prec == "double" else 1e-3
    assert abs(n2 - want_y.abs().square().sum().item()) \
        < tol * max(n2, 1.0)
    assert (y.to_complex() - want_y).abs().max().item() < tol


def test_deterministic_mode_consistency():
    from quda_amd.ops.blas import set_deterministic
    x = mk("double", 4, 709)
    y1 = mk("double", 4, 710)
    n_plain = blas.norm2(x)
    try:
        set_deterministic(True)
        n_det1 = blas.norm2(x)
        n_det2 = blas.norm2(x)
    finally:
        set_deterministic(False)
    assert n_det1 == n_det2
    assert abs(n_det1 - n_plain) < 1e-9 * n_plain
