"""Oracle-op self-consistency tests (the analogue of the reference's
host_reference sanity: free-field dispersion, gamma5-hermiticity, clover
hermiticity, plaquette on known configurations)."""
import math

import numpy as np
import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.fields import gamma
from quda_amd.ops import reference as ref


def _random_setup(geo, seed=7):
    u = GaugeField(geo, "double").random_su3_(seed=seed).to_complex()
    psi = SpinorField(geo, "double").gaussian_(seed=seed + 1).to_complex()
    return u, psi


def _inner(a, b):
    return (a.conj() * b).sum()


def test_free_field_plane_wave(small_geo):
    """Unit gauge: D on a plane wave equals sum_mu (cos p - i sin p gamma_mu)."""
    geo = small_geo
    u = GaugeField(geo, "double").unit_().to_complex()
    X = geo.dims
    p = [2 * math.pi * k / X[i] for i, k in enumerate((1, 0, 2, 1))]
    coords = geo.coords.to(torch.float64)
    phase = torch.exp(1j * (coords * torch.tensor(p, dtype=torch.float64)).sum(-1))  # [V]
    spinor = torch.randn(4, 3, dtype=torch.complex128)
    lo = geo.lex_of_cb
    psi = torch.empty((2, geo.volume_cb, 4, 3), dtype=torch.complex128)
    psi[0] = phase[lo[0]].reshape(-1, 1, 1) * spinor
    psi[1] = phase[lo[1]].reshape(-1, 1, 1) * spinor
    out = ref.dslash_wilson_full(u, psi, geo)
    # expected spin matrix: sum_mu (cos p_mu - i sin p_mu gamma_mu)
    M = sum(math.cos(p[mu]) * np.eye(4) - 1j * math.sin(p[mu]) * gamma.GAMMA[mu]
            for mu in range(4))
    expect_sp = torch.tensor(M, dtype=torch.complex128) @ spinor.reshape(4, 3)
    expect = torch.empty_like(psi)
    expect[0] = phase[lo[0]].reshape(-1, 1, 1) * expect_sp
    expect[1] = phase[lo[1]].reshape(-1, 1, 1) * expect_sp
    assert (out - expect).abs().max().item() < 1e-12


def test_gamma5_hermiticity(rect_geo):
    """gamma5 D gamma5 = D^dag  <=>  <chi, D psi> = <gamma5 D gamma5 chi, psi>*"""
    geo = rect_geo
    u, psi = _random_setup(geo)
    chi = SpinorField(geo, "double").gaussian_(seed=42).to_complex()
    Dpsi = ref.dslash_wilson_full(u, psi, geo)
    Ddag_chi = ref.dslash_wilson_full(u, chi, geo, dagger=True)
    lhs = _inner(chi, Dpsi)
    rhs = _inner(Ddag_chi, psi)
    assert abs(lhs - rhs) < 1e-10 * abs(lhs)
    # and dagger == g5 D g5
    g5Dg5 = ref.apply_gamma5(ref.dslash_wilson_full(u, ref.apply_gamma5(chi), geo))
    assert (g5Dg5 - Ddag_chi).abs().max().item() < 1e-12


def test_dslash_linearity_and_locality(small_geo):
    geo = small_geo
    u, psi = _random_setup(geo)
    out2 = ref.dslash_wilson_full(u, 2.5 * psi, geo)
    out1 = ref.dslash_wilson_full(u, psi, geo)
    assert (out2 - 2.5 * out1).abs().max().item() < 1e-12


def test_plaquette_unit_and_random(small_geo):
    geo = small_geo
    u = GaugeField(geo, "double").unit_().to_complex()
    tot, s, t = ref.plaquette(u, geo)
    assert abs(tot - 1.0) < 1e-14
    u = GaugeField(geo, "double").random_su3_(seed=9).to_complex()
    tot, s, t = ref.plaquette(u, geo)
    assert -1.0 <= tot <= 1.0
    assert abs(tot) < 0.5  # random links decorrelate the trace


def test_clover_hermitian_and_chiral_block(small_geo):
    geo = small_geo
    u = GaugeField(geo, "double").random_su3_(seed=11).to_complex()
    A = ref.clover_matrix(u, geo, kappa=0.1, csw=1.2)
    assert (A - A.conj().mT).abs().max().item() < 1e-12
    # chirality block-diagonal in DeGrand-Rossi: <upper|A|lower> = 0
    Ablk = A.reshape(2, -1, 2, 6, 2, 6)
    assert Ablk[:, :, 0, :, 1, :].abs().max().item() < 1e-13
    assert Ablk[:, :, 1, :, 0, :].abs().max().item() < 1e-13


def test_clover_unit_gauge_is_identity(small_geo):
    geo = small_geo
    u = GaugeField(geo, "double").unit_().to_complex()
    A = ref.clover_matrix(u, geo, kappa=0.1, csw=1.0)
    eye = torch.eye(12, dtype=torch.complex128)
    assert (A - eye).abs().max().item() < 1e-13


@pytest.mark.parametrize("dims,kappa,seed", [
    ((2, 4, 6, 8), 0.09, 11), ((6, 6, 2, 4), 0.21, 12),
    ((4, 2, 8, 2), 0.13, 13)])
def test_dslash_adjoint_identity_shape_sweep(dims, kappa, seed):
    """Property sweep: gamma5-hermiticity <chi, D psi> = <g5 D g5 chi,
    psi>* holds on arbitrary even anisotropic shapes (layout corner
    cases)."""
    import torch
    from quda_amd import GaugeField, LatticeGeometry, SpinorField
    from quda_amd.ops import reference as ref
    geo = LatticeGeometry(dims)
    u = GaugeField(geo, "double").random_su3_(seed=seed).to_complex()
    gen = torch.Generator().manual_seed(seed + 1)
    psi = torch.view_as_complex(torch.randn(
        (geo.volume_cb, 4, 3, 2), generator=gen, dtype=torch.float64))
    chi = torch.view_as_complex(torch.randn(
        (geo.volume_cb, 4, 3, 2), generator=gen, dtype=torch.float64))
    Dp = ref.dslash_wilson_parity(u, psi, geo, 0, False)
    Ddc = ref.dslash_wilson_parity(u, chi, geo, 1, True)
    lhs = (chi.conj() * Dp).sum()
    rhs = (Ddc.conj() * psi).sum()
    assert abs(lhs - rhs) < 1e-10 * max(abs(lhs), 1e-30)
