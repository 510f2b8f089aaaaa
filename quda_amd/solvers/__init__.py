from .cg import cg_solve, SolverStats

__all__ = ["cg_solve", "SolverStats"]
