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
