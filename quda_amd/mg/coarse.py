"""Coarse-grid operator construction and application
(ref: lib/coarse_op.in.cpp + kernels/coarse_op_kernel.cuh CalculateY/VUV,
lib/dirac_coarse.cpp, kernels/dslash_coarse.cuh — re-derived: the coarse
links Y and coarse clover X come from direction-separated Galerkin
triple products R D P; application is batched complex GEMM, which torch
dispatches to rocBLAS — the GEMM-shaped path MFMA wants on MI355X).

Coarse fields are plain tensors [Na, Nc] (Nc = 2*Nvec), lex-ordered
coarse sites; the operator stores
  X      [Na, Nc, Nc]      self coupling (coarse clover)
  Y[d]   [Na, Nc, Nc]      d = 2*mu+fwd: coupling to the +/-mu neighbor
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from ..fields.geometry import LatticeGeometry, checkerboard_join
from ..ops.reference import _gamma_tensors
from .transfer import Transfer


def _coarse_neighbors(cd) -> torch.Tensor:
    """[Na, 8] lex neighbor table on the coarse grid (dir = 2*mu+fwd),
    periodic (coarse extents may be 1 or odd)."""
    cx, cy, cz, ct = cd
    Na = cx * cy * cz * ct
    idx = torch.arange(Na)
    x = idx % cx
    y = (idx // cx) % cy
    z = (idx // (cx * cy)) % cz
    t = idx // (cx * cy * cz)
    coords = [x, y, z, t]
    dims = [cx, cy, cz, ct]
    out = torch.empty((Na, 8), dtype=torch.int64)
    for mu in range(4):
        for fwd in (0, 1):
            c = [cc.clone() for cc in coords]
            c[mu] = (c[mu] + (1 if fwd else -1)) % dims[mu]
            out[:, 2 * mu + fwd] = (((c[3] * cz + c[2]) * cy + c[1]) * cx
                                    + c[0])
    return out


def _hop_lex(u_lex, geo, psi_lex, mu, fwd: bool, dagger: bool, P):
    """One directed Wilson hop on lex fields:
    fwd: U_mu(x) P(-mu) psi(x+mu); bwd: U_mu(x-mu)^d P(+mu) psi(x-mu)."""
    sgn = 1 if not dagger else 0
    if fwd:
        idx = geo.neighbor_lex(mu, +1).to(psi_lex.device)
        proj = torch.einsum("st,vtc->vsc", P[mu, 1 - sgn], psi_lex[idx])
        return torch.einsum("vij,vsj->vsi", u_lex[mu], proj)
    idx = geo.neighbor_lex(mu, -1).to(psi_lex.device)
    proj = torch.einsum("st,vtc->vsc", P[mu, sgn], psi_lex[idx])
    return torch.einsum("vji,vsj->vsi", u_lex[mu][idx].conj(), proj)


class CoarseOp:
    """Explicit coarse operator with X/Y tensors; non-hermitian M plus
    Mdag; presents new_vec/apply for the coarse-level solvers."""

    def __init__(self, X: torch.Tensor, Y: List[torch.Tensor], cd):
        self.X = X
        self.Y = Y  # list of 8: dir = 2*mu+fwd
        self.cd = tuple(cd)
        self.Na, self.Nc = X.shape[0], X.shape[1]
        self.nbr = _coarse_neighbors(cd).to(X.device)

    def apply(self, c: torch.Tensor, dagger: bool = False) -> torch.Tensor:
        if not dagger:
            out = torch.einsum("aij,aj->ai", self.X, c)
            for d in range(8):
                src = self.nbr[:, d]
                out += torch.einsum("aij,aj->ai", self.Y[d], c[src])
            return out
        out = torch.einsum("aji,aj->ai", self.X.conj(), c)
        for d in range(8):
            # coupling a' -> (a' + dir); adjoint scatters: out[nbr] += Y^d c
            src = self.nbr[:, d]
            contrib = torch.einsum("aji,aj->ai", self.Y[d].conj(), c)
            out.index_add_(0, src, contrib)
        return out

    def dense(self) -> np.ndarray:
        """Dense [Na*Nc, Na*Nc] matrix (tests / exact coarse solves)."""
        n = self.Na * self.Nc
        A = np.zeros((n, n), dtype=complex)
        for a in range(self.Na):
            r = slice(a * self.Nc, (a + 1) * self.Nc)
            A[r, r] += self.X[a].cpu().numpy()
            for d in range(8):
                s = int(self.nbr[a, d])
                A[r, s * self.Nc:(s + 1) * self.Nc] += self.Y[d][a].cpu().numpy()
        return A


def build_coarse_op(op, transfer: Transfer) -> CoarseOp:
    """Galerkin coarse operator of the FULL fine operator `op` (must expose
    M on full-parity SpinorFields and its piecewise structure via
    kappa/clover attributes): direction-separated triple products.

    The fine op is assumed of the form  M = Diag - kappa * sum_hops
    (Wilson: Diag = 1; clover: Diag = A)."""
    geo = transfer.geo
    dev = transfer.device
    u_cb = op.gauge.to_complex()
    lo = geo.lex_of_cb.to(dev)
    u_lex = torch.empty((4, geo.volume, 3, 3), dtype=u_cb.dtype, device=dev)
    u_lex[:, lo[0]] = u_cb[:, 0]
    u_lex[:, lo[1]] = u_cb[:, 1]
    P = _gamma_tensors(dev, u_cb.dtype)
    kappa = op.kappa
    clover = getattr(op, "clover", None)
    A_lex = None
    if clover is not None:
        A_cb = clover.to_complex()  # [2, Vcb, 12, 12]
        A_lex = torch.empty((geo.volume, 12, 12), dtype=u_cb.dtype, device=dev)
        A_lex[lo[0]] = A_cb[0]
        A_lex[lo[1]] = A_cb[1]

    Nv = transfer.nvec
    Nc = 2 * Nv
    Na = transfer.n_agg
    X = torch.zeros((Na, Nc, Nc), dtype=u_cb.dtype, device=dev)
    Y = [torch.zeros((Na, Nc, Nc), dtype=u_cb.dtype, device=dev)
         for _ in range(8)]

    # boundary masks in aggregate-ordered layout [Na, B]
    c = geo.coords.to(torch.int64)
    sb = transfer.sites_by_agg
    bnd = {}
    for mu in range(4):
        blk = transfer.block[mu]
        cc = c[:, mu].to(dev)
        bnd[(mu, 1)] = ((cc % blk) == blk - 1)[sb]  # +mu crosses
        bnd[(mu, 0)] = ((cc % blk) == 0)[sb]        # -mu crosses

    V_full = torch.zeros((geo.volume, 4, 3), dtype=u_cb.dtype, device=dev)

    def col_index(chi, v):
        return chi * Nv + v

    for v in range(Nv):
        for chi, sl in ((0, slice(0, 2)), (1, slice(2, 4))):
            # column family: V_v restricted to chirality chi (all aggregates)
            V_full.zero_()
            col_a = torch.zeros((Na, transfer.block_vol, 4, 3),
                                dtype=u_cb.dtype, device=dev)
            col_a[:, :, sl] = transfer.V[:, :, sl, :, v]
            V_full[transfer.sites_by_agg.reshape(-1)] = col_a.reshape(-1, 4, 3)
            j = col_index(chi, v)
            # diagonal term
            if A_lex is not None:
                diag = torch.einsum("vab,vb->va",
                                    A_lex, V_full.reshape(-1, 12)).reshape(-1, 4, 3)
            else:
                diag = V_full
            r = _restrict_cols(transfer, diag)      # [Na, Nc]
            X[:, :, j] += r
            # hop terms
            for mu in range(4):
                for fwd in (0, 1):
                    W = -kappa * _hop_lex(u_lex, geo, V_full, mu, bool(fwd),
                                          False, P)
                    Wa = W[sb]                      # [Na, B, 4, 3]
                    m = bnd[(mu, fwd)].unsqueeze(-1).unsqueeze(-1)
                    W_int = torch.where(m, torch.zeros_like(Wa), Wa)
                    W_bnd = torch.where(m, Wa, torch.zeros_like(Wa))
                    X[:, :, j] += _restrict_agg(transfer, W_int)
                    Y[2 * mu + fwd][:, :, j] += _restrict_agg(transfer, W_bnd)
    return CoarseOp(X, Y, transfer.coarse_dims)


def _restrict_agg(transfer: Transfer, Wa: torch.Tensor) -> torch.Tensor:
    """[Na,B,4,3] aggregate-ordered -> [Na, Nc] coefficients."""
    Na, Nv = transfer.n_agg, transfer.nvec
    out = torch.empty((Na, 2 * Nv), dtype=Wa.dtype, device=Wa.device)
    for chi, sl in ((0, slice(0, 2)), (1, slice(2, 4))):
        out[:, chi * Nv:(chi + 1) * Nv] = torch.einsum(
            "abscv,absc->av", transfer.V[:, :, sl].conj(), Wa[:, :, sl])
    return out


def _restrict_cols(transfer: Transfer, lex: torch.Tensor) -> torch.Tensor:
    return _restrict_agg(transfer, lex[transfer.sites_by_agg])


def coarse_bicgstab(op: CoarseOp, b: torch.Tensor, *, tol: float = 1e-8,
                    maxiter: int = 1000) -> torch.Tensor:
    """BiCGStab on the coarse tensors (all torch ops — rocBLAS batched
    GEMM on GPU; role of the CA-GCR coarse solver, lib/multigrid.cpp)."""
    x = torch.zeros_like(b)
    r = b.clone()
    r0 = r.clone()
    p = r.clone()
    b2 = (r.conj() * r).sum().real.item()
    if b2 == 0:
        return x
    stop = tol * tol * b2
    rho = (r0.conj() * r).sum()
    for _ in range(maxiter):
        v = op.apply(p)
        r0v = (r0.conj() * v).sum()
        if r0v.abs().item() == 0:
            break
        alpha = rho / r0v
        s = r - alpha * v
        t = op.apply(s)
        t2 = (t.conj() * t).sum().real
        if t2.item() == 0:
            x = x + alpha * p
            break
        omega = (t.conj() * s).sum() / t2
        x = x + alpha * p + omega * s
        r = s - omega * t
        r2 = (r.conj() * r).sum().real.item()
        if r2 < stop:
            break
        rho_new = (r0.conj() * r).sum()
        beta = (rho_new / rho) * (alpha / omega)
        rho = rho_new
        p = r + beta * (p - omega * v)
    return x
