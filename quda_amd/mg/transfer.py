"""Multigrid transfer operators (ref: lib/transfer.cpp +
kernels/block_orthogonalize.cuh, restrictor.cuh, prolongator.cuh —
re-derived MI355X-first: aggregation is einsum/batched-GEMM shaped, which
torch dispatches to rocBLAS on the GPU).

Aggregates: geometric blocks of the fine lattice x 2 chiral halves
(DeGrand-Rossi g5 = diag(1,1,-1,-1): chirality 0 = spins 0,1).
Coarse dof per site: Ns_c=2 (chirality) x Nvec "colors".

The fine<->coarse map works on the ORACLE layout [2, Vcb, 4, 3] flattened
to lex [V, 4, 3]; fine fields enter/leave as SpinorFields.
"""

from __future__ import annotations

from typing import List, Tuple

import torch

from ..fields.geometry import LatticeGeometry, checkerboard_join, checkerboard_split
from ..fields.spinor import SpinorField
from ..ops import blas


class Transfer:
    """Prolongator/restrictor over geometric blocks + chiral blocking."""

    def __init__(self, geo: LatticeGeometry, block: Tuple[int, int, int, int],
                 vectors: List[SpinorField]):
        self.geo = geo
        self.block = tuple(block)
        for i in range(4):
            assert geo.dims[i] % block[i] == 0, (geo.dims, block)
        self.coarse_dims = tuple(geo.dims[i] // block[i] for i in range(4))
        self.nvec = len(vectors)
        self.device = vectors[0].device
        self.precision = vectors[0].precision
        # aggregate index of every fine lex site
        cd = self.coarse_dims
        c = geo.coords.to(torch.int64)
        bc = [c[:, i] // block[i] for i in range(4)]
        self.agg_of_lex = (((bc[3] * cd[2] + bc[2]) * cd[1] + bc[1]) * cd[0]
                           + bc[0]).to(self.device)  # [V]
        self.n_agg = cd[0] * cd[1] * cd[2] * cd[3]
        self.block_vol = geo.volume // self.n_agg
        # site order within aggregates: argsort by aggregate
        order = torch.argsort(self.agg_of_lex, stable=True)
        self.sites_by_agg = order.reshape(self.n_agg, self.block_vol)  # [Na, B]
        # V tensor: [Na, B, 4, 3, Nvec] from block-orthonormalized vectors
        self.V = self._pack(vectors)
        self.V = block_orthonormalize(self.V)

    def _pack(self, vectors: List[SpinorField]) -> torch.Tensor:
        vs = []
        for v in vectors:
            lex = checkerboard_join(v.to_complex(), self.geo)  # [V,4,3]
            vs.append(lex[self.sites_by_agg])                  # [Na,B,4,3]
        return torch.stack(vs, dim=-1)  # [Na,B,4,3,Nvec]

    # -- fine -> coarse -----------------------------------------------------
    def restrict(self, fine: SpinorField) -> torch.Tensor:
        """[Na, 2, Nvec] complex coarse vector: c[a,chi,v] =
        sum_{x in a, s in chi} conj(V[a,x,s,c,v]) psi(x,s,c)."""
        lex = checkerboard_join(fine.to_complex(), self.geo)[self.sites_by_agg]
        return self.restrict_lex(lex)

    def restrict_lex(self, lex: torch.Tensor) -> torch.Tensor:
        out = torch.empty((self.n_agg, 2, self.nvec), dtype=lex.dtype,
                          device=lex.device)
        for chi, sl in ((0, slice(0, 2)), (1, slice(2, 4))):
            out[:, chi, :] = torch.einsum("abscv,absc->av",
                                          self.V[:, :, sl].conj(), lex[:, :, sl])
        return out

    # -- coarse -> fine -----------------------------------------------------
    def prolong_lex(self, coarse: torch.Tensor) -> torch.Tensor:
        """[Na,2,Nvec] -> aggregate-ordered fine [Na,B,4,3]."""
        out = torch.zeros((self.n_agg, self.block_vol, 4, 3),
                          dtype=coarse.dtype, device=coarse.device)
        for chi, sl in ((0, slice(0, 2)), (1, slice(2, 4))):
            out[:, :, sl] = torch.einsum("abscv,av->absc", self.V[:, :, sl],
                                         coarse[:, chi, :])
        return out

    def prolong(self, coarse: torch.Tensor, out: SpinorField) -> SpinorField:
        lex_a = self.prolong_lex(coarse)
        V = self.geo.volume
        lex = torch.empty((V, 4, 3), dtype=lex_a.dtype, device=lex_a.device)
        lex[self.sites_by_agg.reshape(-1)] = lex_a.reshape(V, 4, 3)
        out.from_complex(checkerboard_split(lex, self.geo))
        return out


def block_orthonormalize(V: torch.Tensor, passes: int = 2) -> torch.Tensor:
    """Per-(aggregate, chirality) modified Gram-Schmidt of the Nvec columns
    (ref: kernels/block_orthogonalize.cuh, 2-pass)."""
    Na, B, _, _, Nv = V.shape
    out = V.clone()
    for chi, sl in ((0, slice(0, 2)), (1, slice(2, 4))):
        W = out[:, :, sl].reshape(Na, B * 6, Nv).clone()
        for _ in range(passes):
            for j in range(Nv):
                for i in range(j):
                    c = torch.einsum("ab,ab->a", W[:, :, i].conj(), W[:, :, j])
                    W[:, :, j] -= c.unsqueeze(-1) * W[:, :, i]
                nrm = W[:, :, j].norm(dim=-1, keepdim=True).clamp_min(1e-30)
                W[:, :, j] = W[:, :, j] / nrm
        out[:, :, sl] = W.reshape(Na, B, 2, 3, Nv)
    return out


def generate_null_vectors(op, n_vec: int, *, tol: float = 5e-5,
                          maxiter: int = 500, seed: int = 500,
                          precision: str = "double") -> List[SpinorField]:
    """Smooth random vectors against the fine operator: approximately solve
    MdagM x = 0 from a random start (ref: multigrid.cpp:71
    generateNullVectors — BiCGStab/CG on random sources)."""
    from ..solvers import cg_solve
    vecs = []
    for k in range(n_vec):
        x = op.new_spinor(precision, n_parity=2)
        x.gaussian_(seed=seed + k)
        # inverse-iteration smoothing: solve MdagM y = x loosely
        y = op.new_spinor(precision, n_parity=2)
        cg_solve(op, y, x, tol=tol, maxiter=maxiter)
        vecs.append(y)
    return vecs
