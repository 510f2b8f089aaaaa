"""Param validation + pretty-printing (role of the reference's
lib/check_params.h X-macro file, which is included in CHECK/PRINT modes
to generate checkQudaInvertParam / printQudaInvertParam etc.).

check_*_param raises ValueError with EVERY violated constraint listed
(the reference errors on the first); print_*_param returns the aligned
field dump the reference writes through printfQuda, and logs it through
utils.trace at verbosity >= 2.
"""

from __future__ import annotations

from dataclasses import fields


def _dump(p, title: str) -> str:
    lines = [f"{title}:"]
    for f in fields(p):
        v = getattr(p, f.name)
        if hasattr(v, "value"):
            v = v.value
        lines.append(f"  {f.name:24s} = {v}")
    return "\n".join(lines)


def print_gauge_param(p) -> str:
    return _dump(p, "QudaAmdGaugeParam")


def print_invert_param(p) -> str:
    return _dump(p, "QudaAmdInvertParam")


def print_eig_param(p) -> str:
    return _dump(p, "QudaAmdEigParam")


def check_gauge_param(p) -> None:
    errs = []
    if len(p.X) != 4:
        errs.append(f"X must have 4 extents, got {p.X}")
    elif any(int(x) <= 0 or int(x) % 2 for x in p.X):
        errs.append(f"every extent must be positive and even: {p.X}")
    for fld in ("cuda_prec", "cuda_prec_sloppy"):
        if getattr(p, fld) not in ("double", "single", "half", "quarter"):
            errs.append(f"{fld}: unknown precision {getattr(p, fld)!r}")
    for fld in ("reconstruct", "reconstruct_sloppy"):
        if getattr(p, fld) not in ("none", "twelve", "eight"):
            errs.append(f"{fld}: unknown reconstruct {getattr(p, fld)!r}")
    if p.t_boundary not in ("periodic", "anti"):
        errs.append(f"t_boundary must be periodic|anti, got {p.t_boundary!r}")
    if p.anisotropy <= 0:
        errs.append(f"anisotropy must be > 0, got {p.anisotropy}")
    if (p.anisotropy != 1.0 or p.t_boundary == "anti") and \
            p.reconstruct_sloppy != "none":
        # folded phases/scales break the unitarity the codecs assume
        # (api.load_gauge_quda enforces the same guard at load time)
        errs.append("anisotropy/anti-boundary require reconstruct 'none'")
    if errs:
        raise ValueError("GaugeParam check failed:\n  " + "\n  ".join(errs))


def check_invert_param(p) -> None:
    errs = []
    if p.tol <= 0:
        errs.append(f"tol must be > 0, got {p.tol}")
    if p.maxiter <= 0:
        errs.append(f"maxiter must be > 0, got {p.maxiter}")
    if not (0 < p.reliable_delta <= 1):
        errs.append(f"reliable_delta must be in (0,1], got {p.reliable_delta}")
    if p.cuda_prec not in ("double", "single"):
        errs.append(f"cuda_prec (outer) must be double|single, got {p.cuda_prec!r}")
    if p.cuda_prec_sloppy not in ("double", "single", "half", "quarter"):
        errs.append(f"cuda_prec_sloppy: unknown precision {p.cuda_prec_sloppy!r}")
    from ..api import DslashType
    dwf = p.dslash_type in (DslashType.DOMAIN_WALL, DslashType.DOMAIN_WALL_4D,
                            DslashType.MOBIUS, DslashType.ZMOBIUS,
                            DslashType.MOBIUS_EOFA) \
        if hasattr(DslashType, "DOMAIN_WALL_4D") else \
        p.dslash_type in (DslashType.DOMAIN_WALL, DslashType.MOBIUS)
    if dwf and p.Ls < 2:
        errs.append(f"domain-wall actions need Ls >= 2, got {p.Ls}")
    if p.dslash_type in (DslashType.CLOVER, DslashType.TWISTED_CLOVER) and \
            p.clover_csw == 0.0:
        errs.append("clover actions need clover_csw != 0")
    if p.distance_pc_alpha0 < 0:
        errs.append(f"distance_pc_alpha0 must be >= 0, got {p.distance_pc_alpha0}")
    if errs:
        raise ValueError("InvertParam check failed:\n  " + "\n  ".join(errs))


def check_eig_param(e) -> None:
    errs = []
    if e.n_ev <= 0:
        errs.append(f"n_ev must be > 0, got {e.n_ev}")
    if e.n_kr <= e.n_ev:
        errs.append(f"n_kr ({e.n_kr}) must exceed n_ev ({e.n_ev})")
    if e.spectrum not in ("smallest", "largest"):
        errs.append(f"spectrum must be smallest|largest, got {e.spectrum!r}")
    if e.use_poly_acc and e.a_min >= e.a_max:
        errs.append(f"poly acceleration needs a_min < a_max, got [{e.a_min}, {e.a_max}]")
    if errs:
        raise ValueError("EigParam check failed:\n  " + "\n  ".join(errs))
