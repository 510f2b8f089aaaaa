"""Second-level coarsening: transfer + Galerkin operator built ON a
CoarseOp (ref: lib/coarsecoarse_op.in.cpp + dirac_coarse.cpp recursion —
the coarse dof keep their 2-chirality x Nvec structure; level-2
aggregation blocks geometrically over coarse sites with plain (non-chiral)
Nvec2 vectors, which keeps the machinery exact Galerkin)."""

from __future__ import annotations

from typing import List, Tuple

import numpy as np
import torch

from .coarse import CoarseOp, _coarse_neighbors, coarse_bicgstab


class CoarseTransfer:
    """Aggregation over coarse sites: vectors are [Na, Nc] tensors; the
    level-2 field is [Na2, Nv2]."""

    def __init__(self, cd: Tuple[int, int, int, int],
                 block: Tuple[int, int, int, int],
                 vectors: List[torch.Tensor]):
        self.cd = tuple(cd)
        self.block = tuple(block)
        for i in range(4):
            assert cd[i] % block[i] == 0, (cd, block)
        self.cd2 = tuple(cd[i] // block[i] for i in range(4))
        self.nvec = len(vectors)
        cx, cy, cz, ct = cd
        Na = cx * cy * cz * ct
        idx = torch.arange(Na)
        x = idx % cx
        y = (idx // cx) % cy
        z = (idx // (cx * cy)) % cz
        t = idx // (cx * cy * cz)
        bc = [x // block[0], y // block[1], z // block[2], t // block[3]]
        c2 = self.cd2
        agg = (((bc[3] * c2[2] + bc[2]) * c2[1] + bc[1]) * c2[0] + bc[0])
        self.n_agg = c2[0] * c2[1] * c2[2] * c2[3]
        order = torch.argsort(agg, stable=True)
        self.block_vol = Na // self.n_agg
        self.sites_by_agg = order.reshape(self.n_agg, self.block_vol)
        dev = vectors[0].device
        self.sites_by_agg = self.sites_by_agg.to(dev)
        Nc = vectors[0].shape[1]
        # V: [Na2, B, Nc, Nv2], block-orthonormalized per aggregate
        V = torch.stack([v[self.sites_by_agg] for v in vectors], dim=-1)
        self.V = self._orthonormalize(V)

    @staticmethod
    def _orthonormalize(V: torch.Tensor, passes: int = 2) -> torch.Tensor:
        Na2, B, Nc, Nv = V.shape
        W = V.reshape(Na2, B * Nc, Nv).clone()
        for _ in range(passes):
            for j in range(Nv):
                for i in range(j):
                    c = torch.einsum("ab,ab->a", W[:, :, i].conj(), W[:, :, j])
                    W[:, :, j] -= c.unsqueeze(-1) * W[:, :, i]
                nrm = W[:, :, j].norm(dim=-1, keepdim=True).clamp_min(1e-30)
                W[:, :, j] = W[:, :, j] / nrm
        return W.reshape(Na2, B, Nc, Nv)

    def restrict(self, c: torch.Tensor) -> torch.Tensor:
        """[Na, Nc] -> [Na2, Nv2]."""
        ca = c[self.sites_by_agg]  # [Na2, B, Nc]
        return torch.einsum("abcv,abc->av", self.V.conj(), ca)

    def prolong(self, c2: torch.Tensor) -> torch.Tensor:
        out_a = torch.einsum("abcv,av->abc", self.V, c2)
        Na = self.sites_by_agg.numel()
        Nc = out_a.shape[-1]
        out = torch.empty((Na, Nc), dtype=out_a.dtype, device=out_a.device)
        out[self.sites_by_agg.reshape(-1)] = out_a.reshape(Na, Nc)
        return out


def generate_coarse_null_vectors(co: CoarseOp, n_vec: int, *, tol=5e-2,
                                 maxiter=100, seed=700) -> List[torch.Tensor]:
    vecs = []
    for k in range(n_vec):
        gen = torch.Generator().manual_seed(seed + k)
        r = torch.view_as_complex(torch.randn((co.Na, co.Nc, 2), generator=gen,
                                              dtype=torch.float64)).to(co.X.device)
        x = coarse_bicgstab(co, r, tol=tol, maxiter=maxiter)
        vecs.append(x)
    return vecs


def build_coarse2_op(co: CoarseOp, t2: CoarseTransfer) -> CoarseOp:
    """Direction-separated Galerkin of a CoarseOp: X2 collects the self
    coupling + interior hops, Y2[d] the aggregate-boundary hops."""
    cd = co.cd
    Na, Nc = co.Na, co.Nc
    Nv = t2.nvec
    dev = co.X.device
    # coordinates of coarse sites (for boundary masks)
    cx, cy, cz, ct = cd
    idx = torch.arange(Na)
    coords = [idx % cx, (idx // cx) % cy, (idx // (cx * cy)) % cz,
              idx // (cx * cy * cz)]
    bnd = {}
    for mu in range(4):
        blk = t2.block[mu]
        c = coords[mu].to(dev)
        bnd[(mu, 1)] = ((c % blk) == blk - 1)
        bnd[(mu, 0)] = ((c % blk) == 0)

    X2 = torch.zeros((t2.n_agg, Nv, Nv), dtype=co.X.dtype, device=dev)
    Y2 = [torch.zeros((t2.n_agg, Nv, Nv), dtype=co.X.dtype, device=dev)
          for _ in range(8)]
    for v in range(Nv):
        # column family v: the v-th basis vector over all aggregates
        col = torch.zeros((Na, Nc), dtype=co.X.dtype, device=dev)
        col[t2.sites_by_agg.reshape(-1)] = \
            t2.V[:, :, :, v].reshape(Na, Nc)
        # multi-rank: hops crossing a rank boundary read the neighbor
        # rank's column family through the CoarseOp's own face machinery
        ghosts = co._exchange_c(col)
        # self/X coupling
        WX = torch.einsum("aij,aj->ai", co.X, col)
        X2[:, :, v] += t2.restrict(WX)
        # hops: out[a] += Y[d][a] col[nbr(a,d)] — source aggregate is
        # nbr's; crossing iff a is on the d-boundary of its aggregate
        for d in range(8):
            mu, fwd = d // 2, d % 2
            W = torch.einsum("aij,aj->ai", co.Y[d],
                             co._nbr_c(col, d, ghosts))
            m = bnd[(mu, fwd)].unsqueeze(-1)
            W_int = torch.where(m, torch.zeros_like(W), W)
            W_bnd = torch.where(m, W, torch.zeros_like(W))
            X2[:, :, v] += t2.restrict(W_int)
            Y2[d][:, :, v] += t2.restrict(W_bnd)
    return CoarseOp(X2, Y2, t2.cd2, mask=co.mask)


class CoarseMG:
    """Two-grid solver FOR the coarse system (making the overall MG
    3-level): bicgstab smoothing + level-2 Galerkin correction
    (ref: recursive MG::operator(), multigrid.cpp:1145)."""

    def __init__(self, co: CoarseOp, block2=(2, 2, 2, 2), n_vec2: int = 4,
                 null_tol: float = 5e-2, null_maxiter: int = 100,
                 seed: int = 700, vectors=None):
        self.co = co
        vecs = vectors if vectors is not None else generate_coarse_null_vectors(
            co, n_vec2, tol=null_tol, maxiter=null_maxiter, seed=seed)
        self.t2 = CoarseTransfer(co.cd, block2, vecs)
        self.co2 = build_coarse2_op(co, self.t2)

    def solve(self, b: torch.Tensor, *, tol: float = 5e-2,
              maxiter: int = 50, nu: int = 2,
              bottom_tol: float = 1e-2, bottom_maxiter: int = 200
              ) -> torch.Tensor:
        from .coarse import gnorm2
        x = torch.zeros_like(b)
        b2 = gnorm2(b)
        if b2 == 0:
            return x
        for _ in range(maxiter):
            r = b - self.co.apply(x)
            r2 = gnorm2(r)
            if r2 <= tol * tol * b2:
                break
            # pre-smooth: a couple of bicgstab steps on the residual eq
            e = coarse_bicgstab(self.co, r, tol=1e-10, maxiter=nu)
            x = x + e
            r = b - self.co.apply(x)
            # level-2 correction
            rc = self.t2.restrict(r)
            ec = coarse_bicgstab(self.co2, rc, tol=bottom_tol,
                                 maxiter=bottom_maxiter)
            x = x + self.t2.prolong(ec)
        return x
