"""Gauge fixing tests (role of the reference's gauge_alg_test fixing
coverage)."""
import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry
from quda_amd.gauge import plaquette
from quda_amd.gauge.fix import gauge_fix_ovr, gauge_fix_quality


@pytest.fixture(scope="module")
def setup():
    geo = LatticeGeometry((4, 4, 4, 4))
    from quda_amd.fields.gauge import project_su3
    gen = torch.Generator().manual_seed(171)
    eye = torch.eye(3, dtype=torch.complex128)
    m = eye + 0.4 * torch.view_as_complex(
        torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float64))
    return geo, project_su3(m)


@pytest.mark.parametrize("gauge,dirs", [("landau", 4), ("coulomb", 3)])
def test_gauge_fixing_converges(setup, gauge, dirs):
    geo, u = setup
    f0, th0 = gauge_fix_quality(u, geo, dirs)
    uf = gauge_fix_ovr(u, geo, gauge=gauge, max_iter=400, tol=1e-9)
    f1, th1 = gauge_fix_quality(uf, geo, dirs)
    assert th1 < 1e-8, (th0, th1)
    assert f1 > f0
    # gauge-invariant observable unchanged
    p0, _, _ = plaquette(u, geo)
    p1, _, _ = plaquette(uf, geo)
    assert abs(p0 - p1) < 1e-10
    # links still SU(3)
    det = torch.linalg.det(uf.reshape(-1, 3, 3))
    assert (det - 1).abs().max().item() < 1e-8


def test_gauge_fix_fft(setup):
    from quda_amd.gauge.fix import gauge_fix_fft
    geo, u = setup
    f0, th0 = gauge_fix_quality(u, geo, 4)
    uf = gauge_fix_fft(u, geo, max_iter=600, tol=1e-9, alpha=0.08)
    f1, th1 = gauge_fix_quality(uf, geo, 4)
    assert th1 < 1e-8, (th0, th1)
    assert f1 > f0
    p0, _, _ = plaquette(u, geo)
    p1, _, _ = plaquette(uf, geo)
    assert abs(p0 - p1) < 1e-9
    det = torch.linalg.det(uf.reshape(-1, 3, 3))
    assert (det - 1).abs().max().item() < 1e-7
