"""quda.h-parity entry points (ref: include/quda.h + lib/interface_quda.cpp
— the same param-struct + entry-point surface, expressed as Python
dataclasses/functions over the MI355X-native engine; field layouts accepted
at the boundary are the oracle complex layouts).

Covers the reference's resident-field model: load_gauge_quda/
load_clover_quda populate a module-level cache (gaugePrecise + sloppy /
precondition precision copies, interface_quda.cpp:537), invert_quda builds
the Dirac operator quadruple and runs the selected solver with mixed
precision, dslash_quda/mat_quda apply operators, plus observables,
smearing, eigensolver and multigrid hooks.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from enum import Enum
from typing import List, Optional

import torch

from .fields.clover import CloverField, pack_clover
from .fields.gauge import GaugeField
from .fields.geometry import LatticeGeometry
from .fields.spinor import SpinorField
from .models import (DiracClover, DiracCloverPC, DiracStaggered,
                     DiracStaggeredPC, DiracTwistedClover,
                     DiracTwistedCloverPC, DiracTwistedMass,
                     DiracTwistedMassPC, DiracWilson, DiracWilsonPC)
from .models.dwf import (DiracDomainWall, DiracDomainWallPC, DiracMobius,
                         DiracMobiusPC)
from .ops import blas
from .ops import reference as ref
from .solvers import (SolverStats, bicgstab_solve, bicgstabl_solve,
                      ca_cg_solve, ca_gcr_solve, cg_solve, cgne_solve,
                      cgnr_solve, gcr_solve, mr_solve, multishift_cg_solve)


class DslashType(Enum):
    WILSON = "wilson"
    CLOVER = "clover"
    TWISTED_MASS = "twisted-mass"
    TWISTED_CLOVER = "twisted-clover"
    STAGGERED = "staggered"
    ASQTAD = "asqtad"               # improved staggered (fat+long links)
    DOMAIN_WALL = "domain-wall"
    DOMAIN_WALL_4D = "domain-wall-4d"
    MOBIUS = "mobius"
    ZMOBIUS = "zmobius"
    MOBIUS_EOFA = "mobius-eofa"
    NDEG_TWISTED_MASS = "ndeg-twisted-mass"
    NDEG_TWISTED_CLOVER = "ndeg-twisted-clover"
    CLOVER_HASENBUSCH_TWIST = "clover-hasenbusch-twist"
    LAPLACE = "laplace"


class InverterType(Enum):
    CG = "cg"
    CGNE = "cgne"
    CGNR = "cgnr"
    BICGSTAB = "bicgstab"
    BICGSTABL = "bicgstab-l"
    GCR = "gcr"
    MR = "mr"
    CA_CG = "ca-cg"
    CA_GCR = "ca-gcr"
    GMRESDR = "gmresdr"
    EIGCG = "eigcg"


class SolutionType(Enum):
    MAT = "mat"          # solve M x = b on the full lattice
    MATPC = "matpc"      # solve the even-odd system directly


@dataclass
class GaugeParam:
    """ref: QudaGaugeParam quda.h:31 (the fields the engine consumes)."""
    X: tuple = (8, 8, 8, 8)
    cuda_prec: str = "double"
    cuda_prec_sloppy: str = "half"
    reconstruct: str = "none"
    reconstruct_sloppy: str = "twelve"
    anisotropy: float = 1.0
    t_boundary: str = "periodic"  # or "anti" (ref: QudaTboundary)
    staggered_phase_applied: bool = False  # input links carry eta(x)
    device: str = "cuda" if torch.cuda.is_available() else "cpu"


@dataclass
class InvertParam:
    """ref: QudaInvertParam quda.h:100 (consumed subset)."""
    dslash_type: DslashType = DslashType.WILSON
    inv_type: InverterType = InverterType.CG
    solution_type: SolutionType = SolutionType.MAT
    # NORMALIZATION: the stencil applies true projectors P = (1-+g)/2,
    # so M = 1 - kappa sum_mu P U psi and the FREE kappa_c is 1/4 —
    # kappa here = 2 x the standard Wilson convention (and csw here =
    # standard csw / 2 at matched operator). The C ABI converts.
    kappa: float = 0.135
    mass: float = 0.05          # staggered
    mu: float = 0.0             # twisted
    epsilon: float = 0.0        # non-degenerate doublet splitting
    b5_z: Optional[list] = None  # zMobius complex per-slice b5
    c5_z: Optional[list] = None  # zMobius complex per-slice c5
    eofa_pm: int = 1
    eofa_shift: float = 0.0
    mq1: float = 0.01
    mq2: float = 0.08
    mq3: float = 0.08
    m5: float = 1.8             # DWF height
    Ls: int = 8
    b5: float = 1.5
    c5: float = 0.5
    clover_csw: float = 0.0
    distance_pc_alpha0: float = 0.0  # ref quda.h:464 (arXiv:1006.4028)
    distance_pc_t0: int = 0
    tol: float = 1e-8
    maxiter: int = 1000
    reliable_delta: float = 0.1
    cuda_prec: str = "double"
    cuda_prec_sloppy: str = "half"
    preconditioner: Optional[object] = None  # e.g. MG.precond
    deflation: Optional[object] = None  # Deflation (newDeflationQuda)
    chrono_index: int = -1      # >=0: use/extend the resident chrono basis
    chrono_max_dim: int = 8
    # outputs (ref: out-fields of QudaInvertParam)
    iter: int = 0
    true_res: float = 0.0
    secs: float = 0.0
    gflops: float = 0.0


@dataclass
class EigParam:
    """ref: QudaEigParam quda.h:471 (consumed subset)."""
    n_ev: int = 8
    n_kr: int = 32
    tol: float = 1e-8
    max_restarts: int = 100
    use_poly_acc: bool = False
    poly_deg: int = 8
    a_min: float = 0.1
    a_max: float = 10.0
    spectrum: str = "smallest"   # or "largest"
    use_norm_op: bool = True     # TRLM on MdagM vs IRAM on M


# -- resident state (ref: interface_quda.cpp:537 static gauge cache) --------

class _Resident:
    geo: Optional[LatticeGeometry] = None
    gauge: Optional[GaugeField] = None
    fat_gauge: Optional[GaugeField] = None
    long_gauge: Optional[GaugeField] = None
    gauge_sloppy: Optional[GaugeField] = None
    clover: Optional[CloverField] = None
    clover_sloppy: Optional[CloverField] = None
    gauge_param: Optional[GaugeParam] = None
    u_complex: Optional[torch.Tensor] = None


_R = _Resident()


def init_quda(device: int = 0) -> None:
    """ref: initQuda (device init + comms; comms come from torchrun env)."""
    if torch.cuda.is_available():
        torch.cuda.set_device(device)
    from .parallel import comms
    comms.init_comms()


def end_quda() -> None:
    free_gauge_quda()
    _R.fat_gauge = None
    _R.long_gauge = None
    _MOM["p"] = None
    _CHRONO.clear()


def free_gauge_quda() -> None:
    _R.geo = _R.gauge = _R.gauge_sloppy = None
    _R.clover = _R.clover_sloppy = None
    _R.u_complex = None


def load_gauge_quda(u: torch.Tensor, param: GaugeParam) -> None:
    """u: [4, 2, V_cb, 3, 3] complex links (oracle layout) or a GaugeField.

    Creates the precise + sloppy resident copies (ref: loadGaugeQuda
    interface_quda.cpp:571)."""
    geo = LatticeGeometry(param.X)
    if isinstance(u, GaugeField):
        u = u.to_complex()
    u = u.to(param.device)
    if param.staggered_phase_applied:
        # interop: the caller's links already contain the staggered eta
        # phases (MILC convention). The stencil kernels generate eta
        # in-kernel, so STRIP them at load (eta^2 = 1: multiply again)
        # (ref: QudaStaggeredPhase / applyStaggeredPhase).
        from .ops.reference import staggered_phases
        u = u.clone()
        for par in (0, 1):
            ph = staggered_phases(geo, par)  # [Vcb, 4]
            for mu in range(4):
                u[mu, par] = u[mu, par] * ph[:, mu].to(u.dtype).reshape(
                    -1, 1, 1)
    if param.anisotropy != 1.0:
        # fold the bare anisotropy into the SPATIAL links (xi_0
        # convention: U_i -> U_i / xi; ref QudaGaugeParam.anisotropy).
        # Scaled links are no longer unitary, so reconstruction must be
        # off.
        assert param.reconstruct == "none", \
            "anisotropy needs reconstruct='none'"
        param.reconstruct_sloppy = "none"  # scaled links are not unitary
        u = u.clone()
        u[0:3] = u[0:3] / param.anisotropy
    if param.t_boundary == "anti":
        # fold the anti-periodic fermion boundary into the links: negate
        # U_t on the last timeslice (ref: QudaTboundary / the gauge-fix
        # phase application at load in gauge_field.cpp). Applied on a
        # copy so the caller's field is untouched; on a partitioned T
        # grid only the LAST rank's boundary slice carries the phase.
        # Negated SU(3) links cannot be represented by recon-12 (row2 =
        # conj(r0 x r1) always rebuilds +u2, never -u2), so reconstruction
        # must be off for BOTH residencies until the codec applies the
        # boundary phase at decode time (QUDA does this inside
        # gauge_field_order.h's Reconstruct functors).
        param.reconstruct = "none"
        param.reconstruct_sloppy = "none"
        from .parallel import comms
        u = u.clone()
        gt = comms.grid_dims()[3]
        ct = comms.grid_coords()[3]
        if ct == gt - 1:
            t_hi = geo.dims[3] - 1
            for p in (0, 1):
                idx = geo.face_index_cb(p, 3, t_hi).to(u.device)
                u[3, p, idx] = -u[3, p, idx]
    _R.geo = geo
    _R.gauge_param = param
    _R.u_complex = u
    _R.gauge = GaugeField(geo, param.cuda_prec, param.device,
                          reconstruct=param.reconstruct).from_complex(u)
    _R.gauge_sloppy = GaugeField(
        geo, param.cuda_prec_sloppy, param.device,
        reconstruct=param.reconstruct_sloppy).from_complex(u)


def load_clover_quda(inv_param: InvertParam,
                     clover: Optional[torch.Tensor] = None) -> None:
    """Build (or accept) the clover term for the resident gauge
    (ref: loadCloverQuda — with clover=None computes it from the resident
    field like QUDA's compute_clover path)."""
    assert _R.geo is not None, "load_gauge_quda first"
    A = clover if clover is not None else ref.clover_matrix(
        _R.u_complex, _R.geo, inv_param.kappa, inv_param.clover_csw)
    _R.clover = CloverField(_R.geo, _R.gauge.precision,
                            _R.gauge.device).from_matrices(A)
    sp = _R.gauge_sloppy.precision
    cls = CloverField(_R.geo, sp, _R.gauge.device)
    cls.data.copy_(cls._to_native(pack_clover(A)))
    inv = _R.clover.to_complex(inverse=True).to(_R.gauge.device)
    cls.inv_data.copy_(cls._to_native(pack_clover(inv)))
    _R.clover_sloppy = cls


def _make_dirac(p: InvertParam, sloppy: bool = False):
    g = _R.gauge_sloppy if sloppy else _R.gauge
    cl = _R.clover_sloppy if sloppy else _R.clover
    pc = p.solution_type == SolutionType.MATPC
    t = p.dslash_type
    if p.distance_pc_alpha0 != 0.0 and t in (DslashType.WILSON,
                                             DslashType.CLOVER):
        # distance preconditioning (ref: lib/dslash_wilson_distance.cu)
        from .models.distance import (DiracCloverDistance,
                                      DiracCloverDistancePC,
                                      DiracWilsonDistance,
                                      DiracWilsonDistancePC)
        kw = dict(alpha0=p.distance_pc_alpha0, t0=p.distance_pc_t0)
        if t == DslashType.WILSON:
            return (DiracWilsonDistancePC(g, p.kappa, **kw) if pc
                    else DiracWilsonDistance(g, p.kappa, **kw))
        assert cl is not None, "load_clover_quda first"
        return (DiracCloverDistancePC(g, cl, p.kappa, **kw) if pc
                else DiracCloverDistance(g, cl, p.kappa, **kw))
    if t == DslashType.WILSON:
        return DiracWilsonPC(g, p.kappa) if pc else DiracWilson(g, p.kappa)
    if t == DslashType.CLOVER:
        assert cl is not None, "load_clover_quda first"
        return (DiracCloverPC(g, cl, p.kappa) if pc
                else DiracClover(g, cl, p.kappa))
    if t == DslashType.TWISTED_MASS:
        return (DiracTwistedMassPC(g, p.kappa, p.mu) if pc
                else DiracTwistedMass(g, p.kappa, p.mu))
    if t == DslashType.TWISTED_CLOVER:
        assert cl is not None, "load_clover_quda first"
        return (DiracTwistedCloverPC(g, cl, p.kappa, p.mu) if pc
                else DiracTwistedClover(g, cl, p.kappa, p.mu))
    if t == DslashType.STAGGERED:
        return (DiracStaggeredPC(g, p.mass) if pc
                else DiracStaggered(g, p.mass))
    if t == DslashType.DOMAIN_WALL:
        return (DiracDomainWallPC(g, p.mass, p.m5, p.Ls) if pc
                else DiracDomainWall(g, p.mass, p.m5, p.Ls))
    if t == DslashType.MOBIUS:
        return (DiracMobiusPC(g, p.mass, p.m5, p.Ls, p.b5, p.c5) if pc
                else DiracMobius(g, p.mass, p.m5, p.Ls, p.b5, p.c5))
    if t == DslashType.DOMAIN_WALL_4D:
        from .models import DiracDomainWall4D, DiracDomainWall4DPC
        return (DiracDomainWall4DPC(g, p.mass, p.m5, p.Ls) if pc
                else DiracDomainWall4D(g, p.mass, p.m5, p.Ls))
    if t == DslashType.ZMOBIUS:
        from .models import DiracZMobius, DiracZMobiusPC
        b5 = p.b5_z or [p.b5] * p.Ls
        c5 = p.c5_z or [p.c5] * p.Ls
        return (DiracZMobiusPC(g, p.mass, p.m5, p.Ls, b5, c5) if pc
                else DiracZMobius(g, p.mass, p.m5, p.Ls, b5, c5))
    if t == DslashType.MOBIUS_EOFA:
        from .models import DiracMobiusEofa, DiracMobiusEofaPC
        kw = dict(mq1=p.mq1, mq2=p.mq2, mq3=p.mq3, eofa_pm=p.eofa_pm,
                  eofa_shift=p.eofa_shift)
        return (DiracMobiusEofaPC(g, p.m5, p.Ls, p.b5, p.c5, **kw) if pc
                else DiracMobiusEofa(g, p.m5, p.Ls, p.b5, p.c5, **kw))
    if t == DslashType.NDEG_TWISTED_MASS:
        from .models import DiracNdegTwistedMass, DiracNdegTwistedMassPC
        return (DiracNdegTwistedMassPC(g, p.kappa, p.mu, p.epsilon) if pc
                else DiracNdegTwistedMass(g, p.kappa, p.mu, p.epsilon))
    if t == DslashType.NDEG_TWISTED_CLOVER:
        assert cl is not None, "load_clover_quda first"
        from .models import (DiracNdegTwistedClover,
                             DiracNdegTwistedCloverPC)
        return (DiracNdegTwistedCloverPC(g, cl, p.kappa, p.mu, p.epsilon)
                if pc else
                DiracNdegTwistedClover(g, cl, p.kappa, p.mu, p.epsilon))
    if t == DslashType.CLOVER_HASENBUSCH_TWIST:
        assert cl is not None, "load_clover_quda first"
        from .models import (DiracCloverHasenbuschTwist,
                             DiracCloverHasenbuschTwistPC)
        return (DiracCloverHasenbuschTwistPC(g, cl, p.kappa, p.mu) if pc
                else DiracCloverHasenbuschTwist(g, cl, p.kappa, p.mu))
    if t == DslashType.LAPLACE:
        from .models.laplace import GaugeLaplace
        return GaugeLaplace(g, m2=p.mass, ndim=3)
    if t == DslashType.ASQTAD:
        from .models import (DiracImprovedStaggered,
                             DiracImprovedStaggeredPC)
        assert _R.fat_gauge is not None, "compute/load KS links first"
        return (DiracImprovedStaggeredPC(_R.fat_gauge, _R.long_gauge,
                                         p.mass) if pc
                else DiracImprovedStaggered(_R.fat_gauge, _R.long_gauge,
                                            p.mass))
    raise ValueError(t)


def _wrap(v: torch.Tensor, p: InvertParam, n_parity: int) -> SpinorField:
    geo = _R.geo
    nspin = 1 if p.dslash_type in (DslashType.STAGGERED,
                                   DslashType.ASQTAD) else 4
    ls = p.Ls if p.dslash_type in (
        DslashType.DOMAIN_WALL, DslashType.DOMAIN_WALL_4D,
        DslashType.MOBIUS, DslashType.ZMOBIUS,
        DslashType.MOBIUS_EOFA) else (
        2 if p.dslash_type in (DslashType.NDEG_TWISTED_MASS,
                               DslashType.NDEG_TWISTED_CLOVER) else 1)
    f = SpinorField(geo, p.cuda_prec, _R.gauge.device, n_parity,
                    nspin=nspin, ls=ls)
    f.from_complex(v.to(_R.gauge.device))
    return f


def dslash_quda(inp: torch.Tensor, p: InvertParam, parity: int) -> torch.Tensor:
    """Apply the parity-hopping dslash (ref: dslashQuda
    interface_quda.cpp:1709). inp at parity 1-parity ([Vcb, 4, 3] for
    Wilson-family, [Vcb, 3] for staggered; a leading parity dim of 1 is
    also accepted)."""
    d = _make_dirac(p)
    nsite_dims = 1 if p.dslash_type in (DslashType.STAGGERED,
                                        DslashType.ASQTAD) else 2
    src = _wrap(inp.unsqueeze(0) if inp.dim() == 1 + nsite_dims else inp,
                p, 1)
    out = src.clone_empty()
    d.dslash(out, src, parity)
    return out.to_complex().cpu()[0]


def mat_quda(inp: torch.Tensor, p: InvertParam,
             dagger: bool = False) -> torch.Tensor:
    """Apply the full operator M (ref: MatQuda)."""
    d = _make_dirac(p)
    src = _wrap(inp, p, 2)
    out = src.clone_empty()
    d.M(out, src, dagger=dagger)
    return out.to_complex().cpu()


def invert_quda(b: torch.Tensor, p: InvertParam,
                x0: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Solve M x = b (ref: invertQuda interface_quda.cpp:2986): builds the
    precise/sloppy operator pair, runs the selected solver (PC solves go
    through prepare/reconstruct), fills p.iter/true_res/secs/gflops.
    x0 (optional) is the initial guess (use_init_guess role) for
    non-PC solves."""
    t0 = time.perf_counter()
    d = _make_dirac(p)
    b_f = _wrap(b, p, 2)
    pc_capable = hasattr(d, "prepare")
    distw = getattr(d, "distance", None)

    if p.solution_type == SolutionType.MATPC and pc_capable:
        # distance ops fold the P/P^-1 weights into prepare/reconstruct
        src = d.prepare(b_f)
        x_e = src.clone_empty()
        stats = _run_solver(d, x_e, src, p, sloppy_pair=True)
        x_f = b_f.clone_empty()
        d.reconstruct(x_f, x_e, b_f)
    else:
        x_f = (_wrap(x0, p, 2) if x0 is not None
               else b_f.clone_empty())
        if distw is not None:
            distw.apply(b_f, inverse=True)   # solve M_dist x' = P^-1 b
        stats = _run_solver(d, x_f, b_f, p, sloppy_pair=False)
        if distw is not None:
            distw.apply(x_f)                 # x = P x'
            distw.apply(b_f)                 # restore

    # true residual on the requested (PHYSICAL) system: the plain operator
    r = b_f.clone_empty()
    plain = {**p.__dict__, "distance_pc_alpha0": 0.0}
    full = _make_dirac(InvertParam(**{**plain,
                                      "solution_type": SolutionType.MAT})) \
        if p.solution_type == SolutionType.MATPC else (
            _make_dirac(InvertParam(**plain)) if distw is not None else d)
    full_b = b_f
    full.M(r, x_f)
    import math
    p.true_res = math.sqrt(blas.xmy_norm2(full_b, r) / blas.norm2(full_b))
    p.iter = stats.iters
    p.secs = time.perf_counter() - t0
    vol_factor = _R.geo.volume * (getattr(d, "Ls", 1))
    p.gflops = (stats.iters * 2 * d.flops_per_site() * vol_factor / 1e9
                / max(p.secs, 1e-12)) if hasattr(d, "flops_per_site") else 0.0
    return x_f.to_complex().cpu()


def _run_solver(d, x, b, p: InvertParam, sloppy_pair: bool):
    inv = p.inv_type
    if inv == InverterType.CG:
        # CG delivers M x = b via the normal equations (ref solve-type
        # matrix, lib/solve.cpp: NORMOP solves): rhs <- Mdag b, except for
        # the staggered PC operator which is hermitian PD itself (its
        # MdagM hook applies M_pc once).
        rhs = b
        if not isinstance(d, DiracStaggeredPC):
            rhs = b.clone_empty()
            d.M(rhs, b, dagger=True)
        if p.deflation is not None:
            # deflated initial guess (ref: deflated_invert_test /
            # solver.cpp eig-deflation hookup)
            p.deflation.guess(x, rhs)
        if p.chrono_index >= 0:
            # chronological initial-guess projection over past solutions
            # (ref: inv_param.chrono_make_resident/use_resident_chrono)
            ch = chrono_forecaster(p.chrono_index, p.chrono_max_dim)
            ch.forecast(d, x, rhs)
        if sloppy_pair and p.cuda_prec_sloppy != p.cuda_prec:
            ds = _make_dirac(p, sloppy=True)
            st = cg_solve(d, x, rhs, op_sloppy=ds,
                          sloppy=p.cuda_prec_sloppy, tol=p.tol,
                          maxiter=p.maxiter, delta=p.reliable_delta)
        else:
            st = cg_solve(d, x, rhs, tol=p.tol, maxiter=p.maxiter)
        if p.chrono_index >= 0:
            chrono_forecaster(p.chrono_index, p.chrono_max_dim).append(x)
        return st
    if inv == InverterType.CGNR:
        return cgnr_solve(d, x, b, tol=p.tol, maxiter=p.maxiter)
    if inv == InverterType.CGNE:
        return cgne_solve(d, x, b, tol=p.tol, maxiter=p.maxiter)
    if inv == InverterType.BICGSTAB:
        return bicgstab_solve(d, x, b, tol=p.tol, maxiter=p.maxiter)
    if inv == InverterType.BICGSTABL:
        return bicgstabl_solve(d, x, b, tol=p.tol, maxiter=p.maxiter)
    if inv == InverterType.GCR:
        return gcr_solve(d, x, b, tol=p.tol, maxiter=p.maxiter,
                         precond=p.preconditioner)
    if inv == InverterType.GMRESDR:
        from .solvers.gmresdr import gmresdr_solve
        return gmresdr_solve(d, x, b, tol=p.tol, maxiter=p.maxiter)
    if inv == InverterType.EIGCG:
        from .solvers.eigcg import eigcg_solve
        rhs = b
        if not isinstance(d, DiracStaggeredPC):
            rhs = b.clone_empty()
            d.M(rhs, b, dagger=True)
        return eigcg_solve(d, x, rhs, tol=p.tol, maxiter=p.maxiter)
    if inv == InverterType.MR:
        return mr_solve(d, x, b, tol=p.tol, maxiter=p.maxiter)
    if inv == InverterType.CA_CG:
        return ca_cg_solve(d, x, b, tol=p.tol, maxiter=p.maxiter)
    if inv == InverterType.CA_GCR:
        return ca_gcr_solve(d, x, b, tol=p.tol, maxiter=p.maxiter)
    raise ValueError(inv)


def invert_multishift_quda(b: torch.Tensor, p: InvertParam,
                           shifts: List[float]) -> List[torch.Tensor]:
    """ref: invertMultiShiftQuda interface_quda.cpp:3405 — (MdagM+s)x=b on
    the even-odd system."""
    assert p.solution_type == SolutionType.MATPC
    d = _make_dirac(p)
    # accept the source with or without the leading parity axis
    # (Wilson site shape (4,3), staggered (3,))
    wilson_site = b.shape[-2:] == (4, 3) if b.dim() >= 2 else False
    has_par = b.dim() == (4 if wilson_site else 3)
    b_f = _wrap(b if has_par else b.unsqueeze(0), p, 1)
    xs = [b_f.clone_empty() for _ in shifts]
    st = multishift_cg_solve(d, xs, b_f, shifts, tol=p.tol,
                             maxiter=p.maxiter)
    p.iter = st.iters
    return [x.to_complex().cpu() for x in xs]


def eigensolve_quda(p: InvertParam, e: EigParam):
    """ref: eigensolveQuda interface_quda.cpp:2524."""
    from .solvers.eigen import ChebyshevOp, iram_solve, trlm_solve
    d = _make_dirac(p)
    npar = 1 if p.solution_type == SolutionType.MATPC else 2
    nspin = 1 if p.dslash_type == DslashType.STAGGERED else 4
    x0 = SpinorField(_R.geo, p.cuda_prec, _R.gauge.device, npar, nspin=nspin)
    if e.use_norm_op:
        poly = (ChebyshevOp(d, e.a_min, e.a_max, e.poly_deg)
                if e.use_poly_acc else None)
        res = trlm_solve(d, e.n_ev, e.n_kr, x0, tol=e.tol,
                         max_restarts=e.max_restarts, poly=poly,
                         which=e.spectrum)
    else:
        res = iram_solve(d, e.n_ev, e.n_kr, x0, tol=e.tol,
                         max_restarts=e.max_restarts,
                         which=("smallest_abs" if e.spectrum == "smallest"
                                else "largest_abs"))
    return res.evals, [v.to_complex().cpu() for v in res.evecs]


def new_multigrid_quda(p: InvertParam, block=(2, 2, 2, 2), n_vec: int = 4,
                       **kw):
    """ref: newMultigridQuda interface_quda.cpp:2772. Returns an MG object
    whose .precond plugs into InvertParam.preconditioner."""
    from .mg import MG, MGParam
    d = _make_dirac(InvertParam(**{**p.__dict__,
                                   "solution_type": SolutionType.MAT}))
    return MG(d, MGParam(block=block, n_vec=n_vec, **kw))


# -- observables / smearing (ref: gauge_observable.cpp entry points) --------

def plaq_quda() -> tuple:
    from .gauge import plaquette
    return plaquette(_R.u_complex, _R.geo)


def gauge_observables_quda() -> dict:
    from .gauge import plaquette, polyakov_loop, topological_charge
    from .gauge import det_trace, energy_density
    tot, sp, tm = plaquette(_R.u_complex, _R.geo)
    ep, ec = energy_density(_R.u_complex, _R.geo)
    d, t = det_trace(_R.u_complex, _R.geo)
    return {
        "plaquette": (tot, sp, tm),
        "polyakov_loop": polyakov_loop(_R.u_complex, _R.geo),
        "qcharge": topological_charge(_R.u_complex, _R.geo),
        "energy": (ep, ec),
        "det": d,
        "trace": t,
    }


def perform_gauge_smear_quda(kind: str, n_steps: int, coeff: float,
                             measure: bool = False):
    """ref: performGaugeSmearQuda / performWFlowQuda — smears the resident
    field in place; measure=True (wilson_flow only) also returns the
    per-step (t, E_plaq, E_clover, t^2 E) history (the
    gaugeSmearParam.meas_interval role)."""
    from .gauge import ape_smear, stout_smear, wilson_flow
    u = _R.u_complex
    if kind == "wilson_flow" and measure:
        from .gauge import wilson_flow_measure
        u, hist = wilson_flow_measure(u, _R.geo, coeff, n_steps)
        load_gauge_quda(u, _R.gauge_param)
        return hist
    if kind == "ape":
        u = ape_smear(u, _R.geo, coeff, n_steps)
    elif kind == "stout":
        u = stout_smear(u, _R.geo, coeff, n_steps)
    elif kind == "wilson_flow":
        u = wilson_flow(u, _R.geo, coeff, n_steps)
    elif kind == "hyp":
        from .gauge import hyp_smear
        u = hyp_smear(u, _R.geo, n_iter=n_steps)
    else:
        raise ValueError(kind)
    load_gauge_quda(u, _R.gauge_param)


# -- HMC entry points (ref: computeGaugeForceQuda, updateGaugeFieldQuda,
# momActionQuda, momResidentQuda) ------------------------------------------

def compute_gauge_force_quda(beta: float) -> torch.Tensor:
    from .gauge import gauge_force
    return gauge_force(_R.u_complex, _R.geo, beta)


def update_gauge_field_quda(mom: torch.Tensor, dt: float) -> None:
    """U <- exp(dt P) U on the resident field."""
    from .gauge.ops import _from_lex, _to_lex, exp_su3
    U = _to_lex(_R.u_complex, _R.geo)
    P = _to_lex(mom, _R.geo)
    load_gauge_quda(_from_lex(exp_su3(P, dt) @ U, _R.geo), _R.gauge_param)


def mom_action_quda(mom: torch.Tensor) -> float:
    from .gauge import mom_action
    return mom_action(mom)


_MOM = {"p": None}


def mom_resident_quda(mom: Optional[torch.Tensor] = None) -> torch.Tensor:
    """ref: momResidentQuda — set (mom given) or get the resident
    momentum field."""
    if mom is not None:
        _MOM["p"] = mom
    return _MOM["p"]


def compute_gauge_path_force_quda(paths, coeffs, beta: float = 1.0
                                  ) -> torch.Tensor:
    """ref: computeGaugeForceQuda with explicit input paths — the force
    of S = beta * sum_i c_i sum_x Re tr[1 - P_i(x)/3] for arbitrary
    signed-direction paths, by reverse-mode differentiation of the loop
    traces (validated against the analytic plaquette force and FD in
    the tests). Single-rank (autograd path); the plaquette action has
    the analytic multi-rank force."""
    from .gauge.ops import path_product, project_ta
    from .parallel import comms
    assert comms.comm_size() == 1, "autograd path force is single-rank"
    geo = _R.geo
    u_req = _R.u_complex.detach().clone().requires_grad_(True)
    s = None
    for c, p in zip(coeffs, paths):
        P = path_product(u_req, geo, tuple(p))
        tr = torch.diagonal(P, dim1=-2, dim2=-1).sum(-1).real.sum() / 3.0
        term = beta * c * (geo.volume - tr)
        s = term if s is None else s + term
    s.backward()
    g = u_req.grad
    u = _R.u_complex
    F = torch.empty_like(u)
    for mu in range(4):
        for par in (0, 1):
            F[mu, par] = 0.5 * project_ta(u[mu, par]
                                          @ g[mu, par].conj().mT)
    return F


def gauss_mom_quda(seed: int) -> torch.Tensor:
    from .gauge import random_momentum
    return random_momentum(_R.geo, _R.gauge.device, seed)


def compute_gauge_fixing_ovr_quda(gauge: str = "landau", **kw) -> None:
    """Fix the resident field (ref: computeGaugeFixingOVRQuda)."""
    from .gauge.fix import gauge_fix_ovr
    load_gauge_quda(gauge_fix_ovr(_R.u_complex, _R.geo, gauge=gauge, **kw),
                    _R.gauge_param)


def compute_ks_link_quda(coeffs=None):
    """Fat + long links of the resident field (ref: computeKSLinkQuda)."""
    from .gauge.hisq import asqtad_coefficients, fat_links, naik_links
    c = coeffs or asqtad_coefficients()
    fat = fat_links(_R.u_complex, _R.geo, c)
    lng = naik_links(_R.u_complex, _R.geo)
    gp = _R.gauge_param
    _R.fat_gauge = GaugeField(_R.geo, gp.cuda_prec, gp.device).from_complex(fat)
    _R.long_gauge = GaugeField(_R.geo, gp.cuda_prec, gp.device,
                               shift=3).from_complex(lng)
    return fat, lng


def mat_dag_mat_quda(inp: torch.Tensor, p: InvertParam) -> torch.Tensor:
    """ref: MatDagMatQuda interface_quda.cpp — applies M^dag M of the
    resident operator."""
    d = _make_dirac(p)
    n_parity = 1 if "PC" in type(d).__name__ else 2
    b = _wrap(inp, p, n_parity)
    out = _wrap(torch.zeros_like(inp), p, n_parity)
    tmp = _wrap(torch.zeros_like(inp), p, n_parity)
    d.MdagM(out, b, tmp)
    return out.to_complex()


def save_gauge_quda(path: Optional[str] = None) -> torch.Tensor:
    """ref: saveGaugeQuda — returns the resident gauge field as
    [4,2,Vcb,3,3] complex (the inverse of load_gauge_quda); with `path`
    also writes the checksummed on-disk record (utils.io.save_gauge)."""
    assert _R.u_complex is not None, "no resident gauge"
    if path is not None:
        from .utils.io import save_gauge
        save_gauge(path, _R.u_complex, _R.geo)
    return _R.u_complex.clone()


def covdev_quda(inp: torch.Tensor, p: InvertParam, mu: int,
                forward: bool = True) -> torch.Tensor:
    """ref: the GaugeCovDev dslash type (covariant displacement, used
    for sequential sources and derivative operators)."""
    from .models.laplace import covdev_apply
    f = _wrap(inp, p, 2)
    return covdev_apply(_R.gauge, f, mu, forward).to_complex()


def invert_multi_src_quda(bs, p: InvertParam, *, splits=None):
    """ref: invertMultiSrcQuda + the split-grid comm key: solves all
    sources; with `splits` (per-dim sub-grid factors) the sources are
    distributed over split-grid sub-grids with field redistribution
    (parallel.split_grid.split_grid_solve); otherwise each source is
    solved in order on the current grid."""
    from .parallel import comms
    if splits is not None and comms.is_distributed():
        from .parallel.split_grid import split_grid_solve
        gp0 = _R.gauge_param

        def solve_one(u_big_cb, b_big_cb, geo_big):
            saved = (_R.gauge, _R.gauge_sloppy, _R.u_complex, _R.geo,
                     _R.gauge_param)
            try:
                gp = GaugeParam(X=geo_big.dims, cuda_prec=gp0.cuda_prec,
                                cuda_prec_sloppy=gp0.cuda_prec_sloppy,
                                device=gp0.device)
                load_gauge_quda(u_big_cb, gp)
                return invert_quda(b_big_cb, p)
            finally:
                (_R.gauge, _R.gauge_sloppy, _R.u_complex, _R.geo,
                 _R.gauge_param) = saved

        return split_grid_solve(_R.u_complex, list(bs), _R.geo, splits,
                                solve_one)
    return [invert_quda(b, p) for b in bs]


def blas_gemm_quda(a: torch.Tensor, b: torch.Tensor,
                   alpha=1.0, beta=0.0, c: torch.Tensor = None):
    """ref: blasGEMMQuda (batched strided complex GEMM; rocBLAS via
    torch.matmul on device tensors)."""
    r = alpha * (a @ b)
    if c is not None and beta != 0.0:
        r = r + beta * c
    return r


def compute_clover_force_quda(kappa: float, csw: float, x_sol: torch.Tensor,
                              seed: int = 0) -> torch.Tensor:
    """ref: computeCloverForceQuda — clover-fermion force on the resident
    gauge for pseudofermion phi (autograd-exact; see
    gauge.fermion_force.clover_fermion_force)."""
    from .gauge.fermion_force import clover_fermion_force
    phi = _wrap(x_sol, InvertParam(kappa=kappa), 2)
    S, F = clover_fermion_force(_R.u_complex, _R.geo, kappa, csw, phi)
    return F


def contract_quda(x: torch.Tensor, y: torch.Tensor, p: InvertParam,
                  mode: str = "open") -> torch.Tensor:
    """ref: contractQuda — open-spin / DeGrand-Rossi contraction of two
    propagator fields."""
    from .ops import contract
    xf = _wrap(x, p, 2)
    yf = _wrap(y, p, 2)
    if mode == "open":
        return contract.contract_open_spin(xf, yf)
    return contract.contract_dr(xf, yf)


def contract_ft_quda(x: torch.Tensor, y: torch.Tensor, p: InvertParam,
                     momenta, reduct_dim: int = 3) -> torch.Tensor:
    """ref: contractFTQuda — momentum-projected timeslice contraction."""
    from .ops import contract
    return contract.contract_ft(_wrap(x, p, 2), _wrap(y, p, 2),
                                list(momenta), reduct_dim)


def new_deflation_quda(p: InvertParam, e: EigParam):
    """ref: newDeflationQuda — eigensolve the resident operator and
    return a Deflation object (plugs into solver initial guesses)."""
    from .solvers.eigen import Deflation
    evals, evecs = eigensolve_quda(p, e)
    npar = 1 if p.solution_type == SolutionType.MATPC else 2
    fields = [_wrap(v, p, npar) for v in evecs]
    return Deflation([float(complex(v).real) for v in evals], fields)


_CHRONO = {}


def chrono_forecaster(index: int = 0, max_dim: int = 8):
    """Resident chrono basis per index (ref: inv_param.chrono_index +
    use_resident_chrono)."""
    from .solvers.mre import ChronoForecaster
    if index not in _CHRONO:
        _CHRONO[index] = ChronoForecaster(max_dim)
    return _CHRONO[index]


def flush_chrono_quda(index: int = -1) -> None:
    """ref: flushChronoQuda interface_quda.cpp — drop resident chrono
    bases (index -1 = all)."""
    if index < 0:
        _CHRONO.clear()
    else:
        _CHRONO.pop(index, None)


def perform_fermion_smear_quda(src: torch.Tensor, p: InvertParam,
                               n_steps: int, width: float) -> torch.Tensor:
    """ref: performFermionSmearQuda — Gaussian/Wuppertal smearing of a
    source with the resident gauge."""
    from .models.laplace import wuppertal_smear
    f = _wrap(src, p, 2)
    out = wuppertal_smear(_R.gauge, f, alpha=width, n_steps=n_steps)
    return out.to_complex()


def update_multigrid_quda(mg, p: InvertParam) -> None:
    """ref: updateMultigridQuda — re-setup the MG hierarchy after the
    resident gauge changed (re-generates null vectors + coarse ops on the
    CURRENT resident links)."""
    from .mg import MG
    d = _make_dirac(InvertParam(**{**p.__dict__,
                                   "solution_type": SolutionType.MAT}))
    mg.__init__(d, mg.param)


# -- verbosity / logging (ref: logQuda/printfQuda/warningQuda, util_quda.h)

_VERBOSITY = 1  # 0 silent, 1 summarize, 2 verbose, 3 debug


def set_verbosity_quda(level: int) -> None:
    global _VERBOSITY
    _VERBOSITY = int(level)


def log_quda(level: int, msg: str) -> None:
    if level <= _VERBOSITY:
        print(f"QUDA-AMD: {msg}", flush=True)


def warning_quda(msg: str) -> None:
    import sys
    print(f"QUDA-AMD WARNING: {msg}", file=sys.stderr, flush=True)
