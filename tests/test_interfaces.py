"""MILC-convention interface shim tests (role of the reference's
milc_interface.cpp consumers)."""
import math

import pytest
import torch

from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.fields.interop import gauge_to_milc, spinor_from_milc
from quda_amd.interfaces import (qudaInvert, qudaLoadGauge,
                                 qudaMultishiftInvert, qudaPlaquette)
from quda_amd.interfaces.milc import qudaLoadKSLink
from quda_amd.ops import reference as ref


@pytest.fixture(scope="module")
def milc_setup():
    geo = LatticeGeometry((4, 4, 4, 8))
    u = GaugeField(geo, "double").random_su3_(seed=171).to_complex()
    um = gauge_to_milc(u, geo)
    qudaLoadGauge((4, 4, 4, 8), um, precision="double", device="cpu")
    return geo, u


def test_milc_plaquette_and_kslink(milc_setup):
    from quda_amd.gauge import plaquette
    geo, u = milc_setup
    tot, sp, tm = qudaPlaquette()
    tot2, _, _ = plaquette(u, geo)
    assert abs(tot - tot2) < 1e-12
    fat, lng = qudaLoadKSLink()
    assert fat.shape == (4, 2, geo.volume_cb, 3, 3)
    assert lng.shape == (4, 2, geo.volume_cb, 3, 3)


def test_milc_invert(milc_setup):
    geo, u = milc_setup
    mass = 0.1
    g = torch.Generator().manual_seed(172)
    bm = torch.view_as_complex(
        torch.randn((2 * geo.volume_cb, 3, 2), generator=g,
                    dtype=torch.float64))
    xm = qudaInvert(mass, bm, tol=1e-10, maxiter=2000)
    # residual in engine order: (2m + D) x = b
    x = spinor_from_milc(xm, geo)
    b = spinor_from_milc(bm, geo)
    r = ref.mat_staggered(u, x, geo, mass)
    err = (r - b).abs().max().item()
    assert err < 1e-6, err


def test_milc_multishift(milc_setup):
    geo, u = milc_setup
    mass = 0.1
    g = torch.Generator().manual_seed(173)
    be = torch.view_as_complex(
        torch.randn((geo.volume_cb, 3, 2), generator=g,
                    dtype=torch.float64))
    shifts = [0.0, 0.1]
    xs = qudaMultishiftInvert(shifts, mass, be, tol=1e-10)
    from quda_amd.models import DiracStaggeredPC
    gf = GaugeField(geo, "double").from_complex(u)
    pc = DiracStaggeredPC(gf, mass)
    from quda_amd.ops import blas
    for s, xm in zip(shifts, xs):
        xf = SpinorField(geo, "double", n_parity=1, nspin=1)
        xf.from_complex(xm if xm.dim() == 3 else xm.unsqueeze(0))
        r = SpinorField(geo, "double", n_parity=1, nspin=1)
        t = SpinorField(geo, "double", n_parity=1, nspin=1)
        pc.MdagM(r, xf, t)
        blas.axpy(s, xf, r)
        bf = SpinorField(geo, "double", n_parity=1, nspin=1)
        bf.from_complex(be.unsqueeze(0))
        tr = math.sqrt(blas.xmy_norm2(bf, r) / blas.norm2(bf))
        assert tr < 1e-7, (s, tr)


def test_milc_extended_surface():
    """The round-2 MILC entries (lifecycle, dslash, clover solves,
    multi-src, rephase, unitarized links, observables, momenta) run
    end-to-end on a small lattice."""
    import torch
    from quda_amd import api
    from quda_amd.interfaces import milc
    dims = (4, 4, 4, 4)
    milc.qudaInit()
    milc.qudaSetLayout(dims)
    geo_v = 4 ** 4
    gen = torch.Generator().manual_seed(909)
    from quda_amd.fields.gauge import GaugeField, project_su3
    from quda_amd.fields.geometry import LatticeGeometry
    from quda_amd.fields.interop import gauge_to_milc
    geo = LatticeGeometry(dims)
    u = GaugeField(geo, "double").random_su3_(seed=909).to_complex()
    milc.qudaLoadGauge(dims, gauge_to_milc(u, geo))
    # dslash + staggered solve
    src = torch.view_as_complex(torch.randn(geo_v, 3, 2, generator=gen,
                                            dtype=torch.float64))
    out = milc.qudaDslash(src, parity=0)
    assert out.shape == src.shape and out.abs().max() > 0
    x = milc.qudaEigCGInvert(0.3, src, tol=1e-7, maxiter=500)
    assert x.abs().max() > 0
    xs = milc.qudaInvertMsrc(0.3, [src, src], tol=1e-7, maxiter=500)
    assert len(xs) == 2
    # clover solve through the MILC entry
    wsrc = torch.view_as_complex(torch.randn(geo_v, 4, 3, 2, generator=gen,
                                             dtype=torch.float64))
    xc = milc.qudaCloverInvert(0.12, 1.0, wsrc, tol=1e-8, maxiter=500)
    assert xc.shape == wsrc.shape
    # links/observables/momenta
    w = milc.qudaLoadUnitarizedLink()
    assert w.shape == u.shape
    pl = milc.qudaPolyakovLoop()
    tr = milc.qudaGaugeLoopTrace([[1, 2, -1, -2]])  # (0,1) plaquette
    mom = torch.zeros_like(u)
    milc.qudaMomLoad(mom)
    assert milc.qudaMomSave() is mom
    milc.qudaRephase(True)
    milc.qudaFreeCloverField()
    milc.qudaFreeGaugeField()
    milc.qudaFinalize()


def test_milc_round2_surface():
    """Round-2 additions: memory helpers, gauge-field handles, phased
    wrappers, shift/spin-taste/two-link smear, gauge fixing, DD invert,
    MG create, force-chain pieces (oprod, clover derivative/trace)."""
    import torch
    from quda_amd import api
    from quda_amd.interfaces import milc
    from quda_amd.fields.gauge import GaugeField
    from quda_amd.fields.geometry import LatticeGeometry
    from quda_amd.fields.interop import gauge_to_milc
    dims = (4, 4, 4, 4)
    milc.qudaInit()
    geo = LatticeGeometry(dims)
    V = geo.volume
    gen = torch.Generator().manual_seed(321)
    u = GaugeField(geo, "double").random_su3_(seed=321).to_complex()
    milc.qudaLoadGaugeField(dims, gauge_to_milc(u, geo))

    # memory + comm helpers
    buf = milc.qudaAllocatePinned(256)
    assert buf.numel() == 256
    milc.qudaFreePinned(buf)
    milc.qudaFreeManaged(milc.qudaAllocateManaged(64))
    milc.qudaSetMPICommHandle(0)

    # handles
    h = milc.qudaCreateGaugeField(dims)
    um = milc.qudaSaveGaugeField(h)
    assert torch.allclose(um[0, 0], torch.eye(3, dtype=um.dtype))
    milc.qudaDestroyGaugeField(h)
    he = milc.qudaCreateExtendedGaugeField(gauge_to_milc(u, geo), dims, 2)
    milc.qudaResidentExtendedGaugeField(he)

    # phased observables: plaquette is a closed loop -> phases cancel in
    # pairs only for 2x identical eta factors; compare against explicit
    # double-rephase instead of the unphased value
    p_phased = milc.qudaPlaquettePhased()
    assert all(abs(x) <= 1.0 + 1e-12 for x in p_phased)
    meas = milc.qudaGaugeMeasurementsPhased()
    assert set(meas) == {"plaquette", "polyakov_loop", "qcharge"}
    # after every phased entry the resident field must be back unphased
    p0 = milc.qudaPlaquette()
    tot2, _, _ = __import__("quda_amd.gauge", fromlist=["plaquette"]
                            ).plaquette(api._R.u_complex, geo)
    assert abs(p0[0] - tot2) < 1e-12

    # covariant shift: free-field shift of a constant vector is itself
    milc.qudaLoadGaugeField(dims, gauge_to_milc(
        torch.eye(3, dtype=torch.complex128).expand(4, 2, geo.volume_cb,
                                                    3, 3).contiguous(), geo))
    cvec = torch.ones((V, 3), dtype=torch.complex128)
    sh = milc.qudaShift(cvec, 0, True)
    assert (sh - cvec).abs().max() < 1e-14
    milc.qudaLoadGaugeField(dims, gauge_to_milc(u, geo))

    # spin-taste + two-link smear
    src = torch.view_as_complex(torch.randn(V, 3, 2, generator=gen,
                                            dtype=torch.float64))
    st = milc.qudaSpinTaste(src, "g5-g5")
    assert st.abs().max() > 0 and st.shape == src.shape
    sm = milc.qudaTwoLinkGaussianSmear(src, width=1.0, n_steps=2)
    assert sm.shape == src.shape
    milc.qudaFreeTwoLink()

    # gauge fixing refreshes the resident field and improves quality
    from quda_amd.gauge.fix import gauge_fix_quality
    q0 = gauge_fix_quality(api._R.u_complex, geo)
    milc.qudaGaugeFixingOVR(4, max_iter=12, tol=1e-30)
    q1 = gauge_fix_quality(api._R.u_complex, geo)
    assert q1[0] > q0[0]
    milc.qudaLoadGaugeField(dims, gauge_to_milc(u, geo))

    # DD invert solves (2m + D) x = b
    from quda_amd.ops import reference as refops
    from quda_amd.fields.interop import spinor_from_milc
    xm = milc.qudaDDInvert(0.3, src, tol=1e-8, maxiter=300)
    x = spinor_from_milc(xm, geo)
    b = spinor_from_milc(src, geo)
    r = refops.mat_staggered(u, x, geo, 0.3)
    assert (r - b).abs().max() < 1e-6

    # MG create + destroy
    mg = milc.qudaMultigridCreate(0.12, block=(2, 2, 2, 2), n_vec=2)
    milc.qudaMultigridDestroy(mg)

    # oprod: free-field check O_mu = psi(x+mu) psi(x)^dag
    o = milc.qudaComputeOprod([1.0], [src])
    from quda_amd.parallel.halo import shift_lex
    psi = src.reshape(V, 3)
    want = torch.einsum(
        "xa,xb->xab",
        shift_lex(psi.unsqueeze(-1), geo, 1, +1).squeeze(-1), psi.conj())
    assert (o[1] - want).abs().max() < 1e-14

    # clover derivative returns a traceless antihermitian field
    F = milc.qudaCloverDerivative(torch.randn(2, geo.volume_cb, 3, 3,
                                              dtype=torch.complex128,
                                              generator=gen), 0, 1)
    assert (F + F.conj().mT).abs().max() < 1e-10
    assert F.diagonal(dim1=-2, dim2=-1).sum(-1).abs().max() < 1e-10

    # clover sigma-trace: antihermitian color matrices per mu<nu pair
    tr = milc.qudaCloverTrace(0.12, 1.0)
    assert len(tr) == 6
    t01 = tr[(0, 1)]
    assert t01.shape == (2, geo.volume_cb, 3, 3)

    milc.qudaFreeGaugeField()
    milc.qudaFinalize()
