"""Solver suite (ref: lib/solver.cpp Solver::create factory + the
inv_*_quda.cpp implementations)."""

from .bicgstab import bicgstab_solve, bicgstabl_solve
from .block_cg import block_cg_solve
from .ca import ca_cg_solve, ca_gcr_solve
from .cg import SolverStats, cg_solve
from .gcr import gcr_solve, mr_solve
from .madwf import TransferLs, madwf_solve, train_transfer
from .eigcg import IncrementalDeflation, eigcg_solve, inc_eigcg_solve
from .gmresdr import gmresdr_solve
from .mspcg import dd_gcr_solve, mspcg_solve, schwarz_precond
from .mre import ChronoForecaster
from .rational import (RationalApprox, rational_approx, rational_apply,
                       rhmc_pseudofermion_action)
from .multishift import multishift_cg_solve
from .variants import (cg3_solve, cgne_solve, cgnr_solve, pcg_solve,
                       sd_solve)

SOLVERS = {
    "cg": cg_solve,
    "cgne": cgne_solve,
    "cgnr": cgnr_solve,
    "cg3": cg3_solve,
    "sd": sd_solve,
    "pcg": pcg_solve,
    "bicgstab": bicgstab_solve,
    "bicgstab-l": bicgstabl_solve,
    "gcr": gcr_solve,
    "mr": mr_solve,
    "ca-cg": ca_cg_solve,
    "ca-gcr": ca_gcr_solve,
    "multishift-cg": multishift_cg_solve,
    "block-cg": block_cg_solve,
    "gmresdr": gmresdr_solve,
    "eigcg": eigcg_solve,
    "mspcg": mspcg_solve,
    "dd-gcr": dd_gcr_solve,
}


def create_solver(name: str):
    """Solver factory (ref: lib/solver.cpp:47 Solver::create)."""
    try:
        return SOLVERS[name]
    except KeyError:
        raise ValueError(f"unknown solver '{name}'; have {sorted(SOLVERS)}")


__all__ = ["cg_solve", "SolverStats", "bicgstab_solve", "bicgstabl_solve",
           "gcr_solve", "mr_solve", "ca_cg_solve", "ca_gcr_solve",
           "multishift_cg_solve", "cgne_solve", "cgnr_solve", "cg3_solve",
           "sd_solve", "pcg_solve", "ChronoForecaster", "create_solver",
           "SOLVERS", "block_cg_solve", "RationalApprox", "rational_approx",
           "rational_apply", "rhmc_pseudofermion_action", "gmresdr_solve",
           "eigcg_solve", "inc_eigcg_solve", "IncrementalDeflation",
           "TransferLs", "madwf_solve", "train_transfer", "mspcg_solve",
           "schwarz_precond", "dd_gcr_solve"]
