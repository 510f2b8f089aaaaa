"""Field layer tests: gamma algebra, layouts, geometry indexing."""
import numpy as np
import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.fields import gamma


def test_gamma_algebra():
    assert gamma.check_algebra()


def test_gamma5_product():
    g5 = gamma.GAMMA[0] @ gamma.GAMMA[1] @ gamma.GAMMA[2] @ gamma.GAMMA[3]
    assert np.allclose(g5, gamma.GAMMA5)


def test_geometry_parity_counts(rect_geo):
    geo = rect_geo
    assert geo.parity.to(torch.int64).sum().item() == geo.volume_cb
    # cb <-> lex roundtrip
    lo = geo.lex_of_cb
    assert torch.unique(lo.flatten()).numel() == geo.volume


def test_neighbor_tables_inverse(rect_geo):
    geo = rect_geo
    for mu in range(4):
        fwd = geo.neighbor_lex(mu, +1)
        bwd = geo.neighbor_lex(mu, -1)
        idx = torch.arange(geo.volume)
        assert torch.equal(bwd[fwd], idx)
        assert torch.equal(fwd[bwd], idx)


def test_neighbor_cb_parity_flip(small_geo):
    geo = small_geo
    for p in (0, 1):
        for mu in range(4):
            nbr = geo.neighbor_cb(p, mu, +1)
            # displaced site must live on the opposite parity
            nbr_lex = geo.lex_of_cb[1 - p][nbr]
            expect = geo.neighbor_lex(mu, +1)[geo.lex_of_cb[p]]
            assert torch.equal(nbr_lex, expect)


@pytest.mark.parametrize("prec", ["double", "single", "half"])
def test_spinor_roundtrip(small_geo, prec):
    s = SpinorField(small_geo, "double").gaussian_(seed=3)
    c = s.to_complex()
    s2 = SpinorField(small_geo, prec).from_complex(c)
    back = s2.to_complex()
    tol = {"double": 1e-14, "single": 1e-6, "half": 1e-3}[prec]
    assert (back - c).abs().max().item() <= tol * c.abs().max().item()


@pytest.mark.parametrize("recon", ["none", "twelve"])
@pytest.mark.parametrize("prec", ["double", "single"])
def test_gauge_roundtrip(small_geo, prec, recon):
    g = GaugeField(small_geo, "double").random_su3_(seed=4)
    u = g.to_complex()
    g2 = GaugeField(small_geo, prec, reconstruct=recon).from_complex(u)
    back = g2.to_complex()
    tol = {"double": 1e-12, "single": 1e-5}[prec]
    assert (back - u).abs().max().item() <= tol


def test_gauge_su3(small_geo):
    g = GaugeField(small_geo, "double").random_su3_(seed=5)
    u = g.to_complex()
    eye = torch.eye(3, dtype=torch.complex128)
    assert (u @ u.conj().mT - eye).abs().max().item() < 1e-12
    assert (torch.linalg.det(u) - 1).abs().max().item() < 1e-12


def test_parity_view_shares_storage(small_geo):
    s = SpinorField(small_geo, "double").gaussian_(seed=6)
    e = s.parity_view(0)
    e.data.mul_(2.0)
    assert torch.allclose(s.data[0], e.data[0])


def test_interop_roundtrips(small_geo):
    import torch
    from quda_amd import GaugeField, SpinorField
    from quda_amd.fields import interop
    geo = small_geo
    u = GaugeField(geo, "double").random_su3_(seed=261).to_complex()
    s = SpinorField(geo, "double").gaussian_(seed=262).to_complex()
    for to, frm in ((interop.gauge_to_qdp, interop.gauge_from_qdp),
                    (interop.gauge_to_milc, interop.gauge_from_milc),
                    (interop.gauge_to_cps, interop.gauge_from_cps)):
        back = frm(to(u, geo), geo)
        assert (back - u).abs().max().item() == 0
    for to, frm in ((interop.spinor_to_qdp, interop.spinor_from_qdp),
                    (interop.spinor_to_milc, interop.spinor_from_milc),
                    (interop.spinor_to_cps, interop.spinor_from_cps)):
        back = frm(to(s, geo), geo)
        assert (back - s).abs().max().item() == 0
    # MILC gauge layout structure: site-major with direction index
    m = interop.gauge_to_milc(u, geo)
    assert m.shape == (geo.volume, 4, 3, 3)
    assert (m[0, 2] - u[2, 0, 0]).abs().max().item() == 0


def test_recon12_rejects_nonunitary_links(small_geo):
    """Fat links loaded into a recon-12 field would be silently
    corrupted; the load must refuse."""
    import pytest
    import torch
    from quda_amd.fields.gauge import GaugeField
    geo = small_geo
    gen = torch.Generator().manual_seed(17)
    u = torch.view_as_complex(torch.randn(
        (4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
        dtype=torch.float64))  # NOT unitary
    g = GaugeField(geo, "double", reconstruct="twelve")
    with pytest.raises(ValueError):
        g.from_complex(u)


def test_recon8_codec_roundtrip_and_field():
    """recon-8 (arXiv:0911.3191): pack/unpack round-trips SU(3) exactly
    at fp64, and a recon-8 GaugeField reproduces its links."""
    import torch
    from quda_amd.fields.gauge import (GaugeField, pack_recon8,
                                       unpack_recon8)
    from quda_amd.fields.geometry import LatticeGeometry
    geo = LatticeGeometry((4, 4, 4, 4))
    g = GaugeField(geo, "double").random_su3_(seed=771)
    u = g.to_complex()
    p = pack_recon8(u)
    assert (unpack_recon8(p) - u).abs().max().item() < 1e-13
    g8 = GaugeField(geo, "double", reconstruct="eight").from_complex(u)
    assert (g8.to_complex() - u).abs().max().item() < 1e-13
    # CPU oracle dslash through a recon-8 field
    from quda_amd import SpinorField
    from quda_amd.ops import reference as ref
    from quda_amd.ops.dispatch import dslash_wilson
    psi = SpinorField(geo, "double").gaussian_(seed=772)
    out = SpinorField(geo, "double", n_parity=1)
    dslash_wilson(out, psi.parity_view(1), g8, 0)
    want = ref.dslash_wilson_parity(u, psi.to_complex()[1], geo, 0)
    assert (out.to_complex()[0] - want).abs().max().item() < 1e-11
