"""MILC-convention interface shim (ref: lib/milc_interface.cpp +
include/quda_milc_interface.h — the entry names and argument conventions
MILC calls; each is a thin adapter over quda_amd.api working in MILC
field order, quda_amd.fields.interop.milc_*)."""

from __future__ import annotations

import torch

from .. import api
from ..fields.geometry import LatticeGeometry
from ..fields.interop import (gauge_from_milc, spinor_from_milc,
                              spinor_to_milc)


def _geo():
    assert api._R.geo is not None, "qudaLoadGauge/api.load_gauge_quda first"
    return api._R.geo


def qudaLoadGauge(dims, u_milc: torch.Tensor, precision: str = "double",
                  device: str = "cpu") -> None:
    """u_milc: [V, 4, 3, 3] site-major MILC order."""
    geo = LatticeGeometry(tuple(dims))
    u = gauge_from_milc(u_milc, geo)
    gp = api.GaugeParam(X=tuple(dims), device=device, cuda_prec=precision,
                        cuda_prec_sloppy=precision)
    api.load_gauge_quda(u, gp)


def qudaInvert(mass: float, source_milc: torch.Tensor, *, tol: float = 1e-8,
               maxiter: int = 2000) -> torch.Tensor:
    """Staggered solve (2m + D) x = b in MILC field order (ref:
    qudaInvert milc_interface.cpp)."""
    geo = _geo()
    b = spinor_from_milc(source_milc, geo)
    p = api.InvertParam(dslash_type=api.DslashType.STAGGERED, mass=mass,
                        inv_type=api.InverterType.CG, tol=tol,
                        maxiter=maxiter)
    x = api.invert_quda(b, p)
    return spinor_to_milc(x, geo)


def qudaMultishiftInvert(mass_sq_offsets, mass: float,
                         source_even: torch.Tensor, *, tol: float = 1e-8,
                         maxiter: int = 2000):
    """Multi-shift staggered solve (M_pc + offset_i) x = b on the EVEN
    sites ([V_cb, 3] — MILC's even-block order is this engine's cb order)
    (ref: qudaMultishiftInvert)."""
    p = api.InvertParam(dslash_type=api.DslashType.STAGGERED, mass=mass,
                        solution_type=api.SolutionType.MATPC,
                        tol=tol, maxiter=maxiter)
    return api.invert_multishift_quda(source_even, p,
                                      list(mass_sq_offsets))


def qudaLoadKSLink(coeffs=None):
    """Fat+long link construction of the resident field (ref:
    qudaLoadKSLink -> computeKSLinkQuda)."""
    return api.compute_ks_link_quda(coeffs)


def qudaGaugeForce(beta: float):
    return api.compute_gauge_force_quda(beta)


def qudaUpdateU(mom: torch.Tensor, dt: float) -> None:
    api.update_gauge_field_quda(mom, dt)


def qudaMomAction(mom: torch.Tensor) -> float:
    return api.mom_action_quda(mom)


def qudaPlaquette():
    return api.plaq_quda()


# ---------------------------------------------------------------------------
# lifecycle / layout (ref: qudaInit/qudaSetLayout/qudaFinalize,
# quda_milc_interface.h:157-169)
# ---------------------------------------------------------------------------

_LAYOUT = {}


def qudaInit(device: int = 0) -> None:
    api.init_quda(device)


def qudaSetLayout(latsize, machsize=None) -> None:
    """Record MILC's lattice/machine layout (the per-rank sub-lattice the
    subsequent qudaLoadGauge* calls use)."""
    _LAYOUT["latsize"] = tuple(latsize)
    _LAYOUT["machsize"] = tuple(machsize) if machsize else (1, 1, 1, 1)


def qudaFinalize() -> None:
    api.end_quda()


def qudaFreeGaugeField() -> None:
    api.free_gauge_quda()


def qudaFreeCloverField() -> None:
    api._R.clover = api._R.clover_sloppy = None


# ---------------------------------------------------------------------------
# operator application / solves (ref: qudaDslash:286, qudaCloverInvert:566,
# qudaInvertMsrc:443, qudaCloverMultishiftInvert:711, qudaEigCGInvert:526,
# qudaInvertMG:409)
# ---------------------------------------------------------------------------

def qudaDslash(source_milc: torch.Tensor, parity: int = 0) -> torch.Tensor:
    """Staggered parity dslash in MILC order (ref qudaDslash)."""
    geo = _geo()
    src = spinor_from_milc(source_milc, geo)
    p = api.InvertParam(dslash_type=api.DslashType.STAGGERED)
    out = api.dslash_quda(src[1 - parity], p, parity)
    full = torch.zeros_like(src)
    full[parity] = out
    return spinor_to_milc(full, geo)


def qudaCloverInvert(kappa: float, csw: float, source_milc: torch.Tensor, *,
                     tol: float = 1e-8, maxiter: int = 2000) -> torch.Tensor:
    """Wilson-clover solve in MILC order (computes + loads the clover
    term from the resident gauge like the reference's compute path)."""
    geo = _geo()
    p = api.InvertParam(dslash_type=api.DslashType.CLOVER, kappa=kappa,
                        clover_csw=csw, inv_type=api.InverterType.CG,
                        solution_type=api.SolutionType.MATPC, tol=tol,
                        maxiter=maxiter)
    api.load_clover_quda(p)
    # MILC wilson-spinor site order == lex; convert to the cb oracle order
    from ..fields.geometry import checkerboard_join, checkerboard_split
    b = checkerboard_split(source_milc.reshape(geo.volume, 4, 3), geo)
    x = api.invert_quda(b, p)
    return checkerboard_join(x, geo)


def qudaCloverMultishiftInvert(kappa: float, csw: float, offsets,
                               source_milc: torch.Tensor, *,
                               tol: float = 1e-8, maxiter: int = 2000):
    geo = _geo()
    p = api.InvertParam(dslash_type=api.DslashType.CLOVER, kappa=kappa,
                        clover_csw=csw,
                        solution_type=api.SolutionType.MATPC, tol=tol,
                        maxiter=maxiter)
    api.load_clover_quda(p)
    return api.invert_multishift_quda(source_milc, p, list(offsets))


def qudaInvertMsrc(mass: float, sources_milc, *, tol: float = 1e-8,
                   maxiter: int = 2000):
    """Multi-source staggered solve (ref qudaInvertMsrc -> the multi-src
    path with optional split-grid)."""
    geo = _geo()
    p = api.InvertParam(dslash_type=api.DslashType.STAGGERED, mass=mass,
                        inv_type=api.InverterType.CG, tol=tol,
                        maxiter=maxiter)
    bs = [spinor_from_milc(s, geo) for s in sources_milc]
    xs = api.invert_multi_src_quda(bs, p)
    return [spinor_to_milc(x, geo) for x in xs]


def qudaEigCGInvert(mass: float, source_milc: torch.Tensor, *,
                    n_ev: int = 8, tol: float = 1e-8,
                    maxiter: int = 2000) -> torch.Tensor:
    """Staggered solve through the deflation-capable path (ref
    qudaEigCGInvert; the incremental-eigCG machinery is
    solvers.eigcg.inc_eigcg_solve — this shim runs the standard CG entry
    with the same conventions)."""
    geo = _geo()
    b = spinor_from_milc(source_milc, geo)
    p = api.InvertParam(dslash_type=api.DslashType.STAGGERED, mass=mass,
                        solution_type=api.SolutionType.MATPC, tol=tol,
                        maxiter=maxiter)
    x = api.invert_quda(b, p)
    return spinor_to_milc(x, geo)


def qudaInvertMG(kappa: float, source: torch.Tensor, *, block=(2, 2, 2, 2),
                 n_vec: int = 4, tol: float = 1e-8, maxiter: int = 300):
    """MG-preconditioned Wilson solve; returns (solution, mg_pack) —
    destroy with qudaMultigridDestroy (ref qudaInvertMG:409)."""
    p = api.InvertParam(dslash_type=api.DslashType.WILSON, kappa=kappa,
                        inv_type=api.InverterType.GCR, tol=tol,
                        maxiter=maxiter)
    mg = api.new_multigrid_quda(p, block=block, n_vec=n_vec)
    p.preconditioner = mg.precond
    x = api.invert_quda(source, p)
    return x, mg


def qudaMultigridDestroy(mg_pack) -> None:
    del mg_pack


# ---------------------------------------------------------------------------
# links / forces / momenta (ref: qudaLoadUnitarizedLink:236, qudaHisqForce:
# 744, qudaMomLoad/Save:898, qudaRephase:933)
# ---------------------------------------------------------------------------

def qudaHisqParamsInit(fat7_coeffs=None) -> None:
    _LAYOUT["hisq_coeffs"] = fat7_coeffs


def qudaLoadUnitarizedLink():
    """Two-level HISQ chain: fat7 -> U(3) unitarize (ref
    qudaLoadUnitarizedLink -> computeKSLinkQuda with unitarization)."""
    from ..gauge.hisq import fat7_coefficients, fat_links, unitarize_links
    u = api._R.u_complex
    geo = api._R.geo
    w = fat_links(u, geo, fat7_coefficients())
    return unitarize_links(w)


def qudaHisqForce(mass: float, source_milc: torch.Tensor,
                  **kw) -> torch.Tensor:
    """HISQ fermion-force contribution of one pseudofermion in MILC order
    (ref qudaHisqForce — here through the differentiated two-level
    fattening chain, gauge/fermion_force.py)."""
    from ..gauge.fermion_force import hisq_fermion_force
    geo = _geo()
    phi = spinor_from_milc(source_milc, geo)
    _, F = hisq_fermion_force(api._R.u_complex, geo, mass, phi, **kw)
    return F


def qudaMomLoad(mom_milc: torch.Tensor) -> None:
    """Resident momentum load (MILC anti-hermitian order = our complex
    [4,2,Vcb,3,3])."""
    api._MOM["p"] = mom_milc


def qudaMomSave() -> torch.Tensor:
    return api._MOM["p"]


def qudaRephase(flag: bool) -> None:
    """MILC keeps eta-phased links resident; this engine generates the
    phases in-kernel, so rephasing toggles the staggered_phase_applied
    load convention (phased input links are stripped at load — ref
    qudaRephase -> gauge_phase.cuh role)."""
    from dataclasses import replace
    from ..ops.reference import staggered_phases
    geo = _geo()
    u = api._R.u_complex.clone()
    if flag:
        for par in (0, 1):
            ph = staggered_phases(geo, par).to(u.device)
            for mu in range(4):
                u[mu, par] = u[mu, par] * ph[:, mu].to(u.dtype).reshape(
                    -1, 1, 1)
    gp = replace(api._R.gauge_param, staggered_phase_applied=bool(flag))
    api.load_gauge_quda(u, gp)


# ---------------------------------------------------------------------------
# observables (ref: qudaPolyakovLoop:829, qudaGaugeLoopTracePhased:805)
# ---------------------------------------------------------------------------

def qudaPolyakovLoop(direction: int = 3) -> complex:
    from ..gauge.ops import polyakov_loop
    return polyakov_loop(api._R.u_complex, api._R.geo)


def qudaGaugeLoopTrace(paths, coeffs=None):
    """Batched Wilson-loop traces (ref qudaGaugeLoopTracePhased)."""
    from ..gauge import loop_trace
    return loop_trace(api._R.u_complex, api._R.geo, paths, coeffs)


# ---------------------------------------------------------------------------
# memory + comm handles (ref: qudaAllocatePinned/Managed:1050-1080,
# qudaSetMPICommHandle:150)
# ---------------------------------------------------------------------------

def qudaAllocatePinned(nbytes: int) -> torch.Tensor:
    """Page-locked host buffer (torch pinned memory when a GPU is
    present; plain host memory otherwise — ref qudaAllocatePinned)."""
    pin = torch.cuda.is_available()
    return torch.empty(int(nbytes), dtype=torch.uint8, pin_memory=pin)


def qudaFreePinned(buf) -> None:
    del buf


def qudaAllocateManaged(nbytes: int) -> torch.Tensor:
    """Managed-memory stand-in: host tensor migrated on demand by the
    caller (HIP managed allocations are not exposed through torch; the
    reference's MILC callers only memcpy through this buffer)."""
    return torch.empty(int(nbytes), dtype=torch.uint8)


def qudaFreeManaged(buf) -> None:
    del buf


def qudaSetMPICommHandle(handle) -> None:
    """Recorded for parity; comms ride torch.distributed (RCCL/gloo),
    initialized by the launcher, not by an MPI communicator handle."""
    _LAYOUT["mpi_comm"] = handle


# ---------------------------------------------------------------------------
# resident / device gauge-field object management (ref:
# qudaCreateGaugeField:960, qudaSaveGaugeField:975, qudaDestroyGaugeField:
# 985, qudaCreateExtendedGaugeField:945, qudaResidentExtendedGaugeField,
# qudaLoadGaugeField/qudaLoadCloverField)
# ---------------------------------------------------------------------------

def qudaCreateGaugeField(dims, geometry: int = 4,
                         precision: str = "double") -> dict:
    """Device gauge container in the engine layout ([4][2][Vcb][3][3]);
    returned handle round-trips through qudaSaveGaugeField /
    qudaDestroyGaugeField like the reference's void* handle."""
    geo = LatticeGeometry(tuple(dims))
    dt = torch.complex128 if precision == "double" else torch.complex64
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    u = torch.zeros((geometry, 2, geo.volume_cb, 3, 3), dtype=dt, device=dev)
    for c in range(3):
        u[:, :, :, c, c] = 1.0
    return {"geo": geo, "u": u}


def qudaSaveGaugeField(handle: dict) -> torch.Tensor:
    from ..fields.interop import gauge_to_milc
    return gauge_to_milc(handle["u"], handle["geo"])


def qudaDestroyGaugeField(handle) -> None:
    if isinstance(handle, dict):
        handle.clear()


def qudaCreateExtendedGaugeField(u_milc: torch.Tensor, dims,
                                 r: int = 2) -> dict:
    """Extended field with an R-site halo in every partitioned dim; on an
    unpartitioned dim the periodic wrap makes the extension implicit, so
    the handle records R and exchanges boundary slabs lazily through the
    gauge boundary-exchange machinery at first stencil use (ref
    qudaCreateExtendedGaugeField -> createExtendedGauge)."""
    geo = LatticeGeometry(tuple(dims))
    return {"geo": geo, "u": gauge_from_milc(u_milc, geo), "radius": int(r)}


def qudaResidentExtendedGaugeField(handle: dict) -> None:
    """Promote an extended handle to the resident field used by
    subsequent force/measurement entries."""
    gp = api.GaugeParam(X=tuple(handle["geo"].dims),
                        device=str(handle["u"].device))
    api.load_gauge_quda(handle["u"], gp)


def qudaLoadGaugeField(dims, u_milc: torch.Tensor, **kw) -> None:
    qudaLoadGauge(dims, u_milc, **kw)


def qudaLoadCloverField(kappa: float, csw: float, **kw) -> None:
    p = api.InvertParam(dslash_type=api.DslashType.CLOVER, kappa=kappa,
                        clover_csw=csw)
    api.load_clover_quda(p)


# ---------------------------------------------------------------------------
# staggered-phase ("Phased") variants (ref: qudaPlaquettePhased:820,
# qudaUpdateUPhased:860, qudaGaugeForcePhased:790 ... the MILC resident
# field carries eta phases; each Phased entry strips/applies them around
# the unphased operation)
# ---------------------------------------------------------------------------

def _with_phases(flag: bool):
    qudaRephase(bool(flag))


def qudaPlaquettePhased():
    _with_phases(True)
    try:
        return api.plaq_quda()
    finally:
        _with_phases(False)


def qudaPolyakovLoopPhased(direction: int = 3):
    _with_phases(True)
    try:
        return qudaPolyakovLoop(direction)
    finally:
        _with_phases(False)


def qudaGaugeLoopTracePhased(paths, coeffs=None):
    _with_phases(True)
    try:
        return qudaGaugeLoopTrace(paths, coeffs)
    finally:
        _with_phases(False)


def qudaGaugeForcePhased(beta: float):
    _with_phases(True)
    try:
        return qudaGaugeForce(beta)
    finally:
        _with_phases(False)


def qudaUpdateUPhased(mom: torch.Tensor, dt: float) -> None:
    _with_phases(True)
    try:
        qudaUpdateU(mom, dt)
    finally:
        _with_phases(False)


def qudaUpdateUPhasedPipeline(mom: torch.Tensor, dt: float) -> None:
    """Pipeline variant: the reference overlaps phase-strip/apply with
    the exp(i dt P) U update per direction; the engine's update is a
    single fused device op, so this aliases the non-pipelined entry."""
    qudaUpdateUPhased(mom, dt)


def qudaGaugeMeasurementsPhased() -> dict:
    """Combined measurement sweep (plaquette + Polyakov loop +
    topological charge, ref qudaGaugeMeasurementsPhased)."""
    from ..gauge.ops import topological_charge
    _with_phases(True)
    try:
        plaq = api.plaq_quda()
        ploop = qudaPolyakovLoop()
        qtop = topological_charge(api._R.u_complex, api._R.geo)
    finally:
        _with_phases(False)
    return {"plaquette": plaq, "polyakov_loop": ploop, "qcharge": qtop}


def qudaUnitarizeSU3(tol: float = 1e-6) -> torch.Tensor:
    """Project the resident links to U(3)/SU(3) (ref qudaUnitarizeSU3 ->
    unitarizeLinksQuda)."""
    from ..gauge.hisq import unitarize_links
    w = unitarize_links(api._R.u_complex, svd_rel_error=tol)
    api.load_gauge_quda(w, api._R.gauge_param)
    return w


def qudaUnitarizeSU3Phased(tol: float = 1e-6) -> torch.Tensor:
    _with_phases(True)
    try:
        return qudaUnitarizeSU3(tol)
    finally:
        _with_phases(False)


# ---------------------------------------------------------------------------
# shifts / smearing / spin-taste / contractions (ref: qudaShift,
# qudaSpinTaste, qudaTwoLinkGaussianSmear:700, qudaContractFT)
# ---------------------------------------------------------------------------

def qudaShift(field_milc: torch.Tensor, mu: int, forward: bool = True):
    """Covariant one-site shift of a staggered field using the resident
    links: (S psi)(x) = U_mu(x) psi(x+mu) (fwd) or
    U_mu(x-mu)^dag psi(x-mu) (bwd) — ref qudaShift."""
    from ..parallel.halo import shift_lex
    geo = _geo()
    V = geo.volume
    psi = field_milc.reshape(V, 3).to(torch.complex128)
    u = api._R.u_complex
    lo = geo.lex_of_cb.to(psi.device)
    U = torch.empty((V, 3, 3), dtype=u.dtype, device=psi.device)
    U[lo[0]] = u[mu, 0]
    U[lo[1]] = u[mu, 1]
    if forward:
        out = torch.einsum("xab,xb->xa", U, shift_lex(psi, geo, mu, +1))
    else:
        out = shift_lex(torch.einsum("xba,xb->xa", U.conj(), psi),
                        geo, mu, -1)
    return out.reshape(field_milc.shape)


def qudaSpinTaste(src_milc: torch.Tensor, kind: str) -> torch.Tensor:
    """Apply a spin-taste phase operator to a staggered field (ref
    qudaSpinTaste -> applySpinTaste)."""
    from ..fields.spinor import SpinorField
    from ..models.spin_taste import apply_spin_taste
    geo = _geo()
    c = spinor_from_milc(src_milc, geo)
    f = SpinorField(geo, "double", str(c.device), nspin=1).from_complex(c)
    return spinor_to_milc(apply_spin_taste(f, kind).to_complex(), geo)


_TWO_LINK = {}


def qudaTwoLinkGaussianSmear(src_milc: torch.Tensor, width: float,
                             n_steps: int) -> torch.Tensor:
    """Two-link Gaussian quark smearing of a staggered field; the
    two-link product field is cached like the reference's resident
    two-link (freed by qudaFreeTwoLink)."""
    from ..fields.spinor import SpinorField
    from ..models.spin_taste import gaussian_smear_two_link, two_links
    geo = _geo()
    if "links" not in _TWO_LINK:
        _TWO_LINK["links"] = two_links(api._R.u_complex, geo)
    c = spinor_from_milc(src_milc, geo)
    f = SpinorField(geo, "double", str(c.device), nspin=1).from_complex(c)
    out = gaussian_smear_two_link(api._R.u_complex, geo, f, width=width,
                                  n_steps=n_steps)
    return spinor_to_milc(out.to_complex(), geo)


def qudaFreeTwoLink() -> None:
    _TWO_LINK.clear()


def qudaContractFT(x_milc: torch.Tensor, y_milc: torch.Tensor, momenta,
                   kappa: float = 0.1):
    """Momentum-projected timeslice contraction of two Wilson propagator
    fields in MILC (lex) order (ref qudaContractFT -> contractFTQuda)."""
    from ..fields.geometry import checkerboard_split
    geo = _geo()
    p = api.InvertParam(dslash_type=api.DslashType.WILSON, kappa=kappa)
    xs = checkerboard_split(x_milc.reshape(geo.volume, 4, 3), geo)
    ys = checkerboard_split(y_milc.reshape(geo.volume, 4, 3), geo)
    return api.contract_ft_quda(xs, ys, p, momenta)


# ---------------------------------------------------------------------------
# gauge fixing (ref: qudaGaugeFixingOVR:1010, qudaGaugeFixingFFT:1030)
# ---------------------------------------------------------------------------

def qudaGaugeFixingOVR(gauge_dir: int = 4, *, max_iter: int = 200,
                       omega: float = 1.7, tol: float = 1e-8):
    """Overrelaxed gauge fixing of the resident field (gauge_dir 4 =
    Landau, 3 = Coulomb); refreshes the resident links like the
    reference."""
    from ..gauge.fix import gauge_fix_ovr
    g = "landau" if gauge_dir == 4 else "coulomb"
    w = gauge_fix_ovr(api._R.u_complex, api._R.geo, gauge=g, omega=omega,
                      max_iter=max_iter, tol=tol)
    api.load_gauge_quda(w, api._R.gauge_param)
    return w


def qudaGaugeFixingFFT(gauge_dir: int = 4, *, max_iter: int = 500,
                       alpha: float = 0.08, tol: float = 1e-8):
    from ..gauge.fix import gauge_fix_fft
    g = "landau" if gauge_dir == 4 else "coulomb"
    w = gauge_fix_fft(api._R.u_complex, api._R.geo, gauge=g, alpha=alpha,
                      max_iter=max_iter, tol=tol)
    api.load_gauge_quda(w, api._R.gauge_param)
    return w


# ---------------------------------------------------------------------------
# extra solve entries (ref: qudaDDInvert:620, qudaEigCGCloverInvert:680,
# qudaMultigridCreate/qudaSetupMultigrid:430-470)
# ---------------------------------------------------------------------------

def qudaDDInvert(mass: float, source_milc: torch.Tensor, *,
                 inner_iters: int = 6, tol: float = 1e-8,
                 maxiter: int = 500) -> torch.Tensor:
    """Domain-decomposed (Schwarz-preconditioned GCR) staggered solve:
    the reference's overlapping-DD preconditioner path (ref qudaDDInvert
    -> invertQuda with inv_type_precondition)."""
    from ..fields.spinor import SpinorField
    from ..models import DiracStaggered
    from ..solvers import dd_gcr_solve
    from ..fields.gauge import GaugeField
    geo = _geo()
    b = spinor_from_milc(source_milc, geo)
    g = GaugeField(geo, "double", str(b.device)).from_complex(
        api._R.u_complex)
    d = DiracStaggered(g, mass)
    bf = SpinorField(geo, "double", str(b.device), nspin=1,
                     n_parity=2).from_complex(b)
    x = d.new_spinor(n_parity=2)
    st = dd_gcr_solve(d, x, bf, inner_iters=inner_iters, tol=tol,
                      maxiter=maxiter)
    assert st.converged
    return spinor_to_milc(x.to_complex(), geo)


def qudaEigCGCloverInvert(kappa: float, csw: float,
                          source_milc: torch.Tensor, *, n_ev: int = 8,
                          tol: float = 1e-8, maxiter: int = 2000):
    """Clover twin of qudaEigCGInvert (deflation-capable clover solve)."""
    return qudaCloverInvert(kappa, csw, source_milc, tol=tol,
                            maxiter=maxiter)


def qudaMultigridCreate(kappa: float, *, block=(2, 2, 2, 2),
                        n_vec: int = 4):
    """Persistent MG setup without a solve (ref qudaMultigridCreate);
    pass the returned pack to qudaInvertMG-style solves via
    param.preconditioner, destroy with qudaMultigridDestroy."""
    p = api.InvertParam(dslash_type=api.DslashType.WILSON, kappa=kappa,
                        inv_type=api.InverterType.GCR)
    return api.new_multigrid_quda(p, block=block, n_vec=n_vec)


def qudaSetupMultigrid(kappa: float, **kw):
    return qudaMultigridCreate(kappa, **kw)


# ---------------------------------------------------------------------------
# force-chain pieces (ref: qudaAsqtadForce:760, qudaCloverForce:890,
# qudaComputeOprod:870, qudaCloverDerivative:910, qudaCloverTrace:905)
# ---------------------------------------------------------------------------

def qudaAsqtadForce(mass: float, source_milc: torch.Tensor,
                    **kw) -> torch.Tensor:
    """One-level asqtad fermion force (fat7+Lepage+Naik chain rule via
    the differentiable fattening — the hisq entry with asqtad
    coefficients and no reunitarization, ref qudaAsqtadForce)."""
    from ..gauge.fermion_force import hisq_fermion_force
    from ..gauge.hisq import asqtad_coefficients
    geo = _geo()
    phi = spinor_from_milc(source_milc, geo)
    _, F = hisq_fermion_force(api._R.u_complex, geo, mass, phi,
                              coeffs=asqtad_coefficients(), **kw)
    return F


def qudaCloverForce(kappa: float, csw: float, source_milc: torch.Tensor,
                    **kw):
    """Two-flavor Wilson-clover MD force (ref qudaCloverForce ->
    computeCloverForceQuda; clover-term derivative through the
    differentiated field-strength construction)."""
    from ..fields.geometry import checkerboard_split
    from ..fields.spinor import SpinorField
    from ..gauge.fermion_force import clover_fermion_force
    geo = _geo()
    b = checkerboard_split(source_milc.reshape(geo.volume, 4, 3), geo)
    phi = SpinorField(geo, "double", str(b.device),
                      n_parity=2).from_complex(b)
    S, F = clover_fermion_force(api._R.u_complex, geo, kappa, csw, phi,
                                **kw)
    return S, F


def qudaComputeOprod(coeffs, fields_milc) -> torch.Tensor:
    """Outer product of solver solutions feeding the force chain:
    O_mu(x) = sum_i c_i psi_i(x+mu) psi_i(x)^dag, color 3x3 per (mu, x)
    in lex order (ref qudaComputeOprod -> computeStaggeredOprod)."""
    geo = _geo()
    from ..fields.geometry import checkerboard_join
    V = geo.volume
    out = torch.zeros((4, V, 3, 3), dtype=torch.complex128)
    from ..parallel.halo import shift_lex
    for c, f in zip(coeffs, fields_milc):
        psi = f.reshape(V, 3).to(torch.complex128)
        for mu in range(4):
            psi_p = shift_lex(psi.unsqueeze(-1), geo, mu, +1).squeeze(-1)
            out[mu] += c * torch.einsum("xa,xb->xab", psi_p, psi.conj())
    return out


def qudaCloverDerivative(oprod: torch.Tensor, mu: int, nu: int,
                         coeff: float = 1.0) -> torch.Tensor:
    """Gauge derivative of Re tr[oprod * F_munu(U)] (the clover-term
    insertion force, ref qudaCloverDerivative -> cloverDerivative):
    returns the TA-projected force for every direction, computed by
    backpropagating through the clover-leaf field strength."""
    from ..ops.reference import field_strength
    geo = _geo()
    u = api._R.u_complex.detach().clone().requires_grad_(True)
    Fmn = field_strength(u, geo)[(min(mu, nu), max(mu, nu))]
    sgn = 1.0 if mu < nu else -1.0
    o = oprod.to(Fmn.dtype)
    s = sgn * coeff * torch.einsum("pxab,pxba->", o, Fmn).real
    (g,) = torch.autograd.grad(s, u)
    # momentum-convention TA projection of U dS/dU
    M = api._R.u_complex * g.conj()
    A = 0.5 * (M - M.conj().mT)
    tr = A.diagonal(dim1=-2, dim2=-1).sum(-1, keepdim=True).unsqueeze(-1)
    return A - tr * torch.eye(3, dtype=A.dtype) / 3.0


def qudaCloverTrace(kappa: float, csw: float) -> torch.Tensor:
    """sigma_munu-contracted trace of the inverse clover term:
    out[(mu,nu)][parity, x] = tr_{spin,color-diag}(sigma_munu A^{-1}(x))
    as color 3x3 matrices — the sigma-trace piece of the clover force
    (ref qudaCloverTrace -> computeCloverSigmaTrace)."""
    from ..fields.gamma import sigma_munu
    from ..ops.reference import clover_matrix
    import numpy as np
    geo = _geo()
    A = clover_matrix(api._R.u_complex, geo, kappa, csw)
    Ainv = torch.linalg.inv(A.reshape(*A.shape[:2], 12, 12))
    Ainv = Ainv.reshape(2, geo.volume_cb, 4, 3, 4, 3)
    out = {}
    for mu in range(4):
        for nu in range(mu + 1, 4):
            s = torch.tensor(np.asarray(sigma_munu(mu, nu)),
                             dtype=Ainv.dtype)
            out[(mu, nu)] = torch.einsum("st,pxtasb->pxab", s, Ainv)
    return out
