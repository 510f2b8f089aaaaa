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
