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


def _lex_face(dims, mu: int, coord: int) -> torch.Tensor:
    """Lex indices of the x_mu == coord sites of a lex-ordered lattice
    `dims`, in lex order of the remaining coords (the canonical face
    order both sides of an exchange agree on)."""
    V = dims[0] * dims[1] * dims[2] * dims[3]
    idx = torch.arange(V)
    stride = 1
    for i in range(mu):
        stride *= dims[i]
    return idx[(idx // stride) % dims[mu] == coord]


def _exchange_lex_faces(c: torch.Tensor, dims, mask: int,
                        faces: Optional[dict] = None) -> dict:
    """Exchange the ±mu faces of a lex-ordered per-site tensor c
    ([V, ...]). Returns {(mu, 1): from +mu neighbor (its x=0 face),
    (mu, 0): from -mu neighbor (its x=hi face)} — exactly the data the
    fwd/bwd hops at the x=hi / x=0 faces consume."""
    from ..parallel.halo import active_dims, exchange_tensors
    sends, recvs = {}, {}
    for mu in active_dims(mask):
        lo = faces[(mu, 0)] if faces else _lex_face(dims, mu, 0).to(c.device)
        hi = (faces[(mu, 1)] if faces
              else _lex_face(dims, mu, dims[mu] - 1).to(c.device))
        sends[(mu, 0)] = c[lo].contiguous()
        sends[(mu, 1)] = c[hi].contiguous()
        recvs[(mu, 0)] = torch.empty_like(sends[(mu, 1)])
        recvs[(mu, 1)] = torch.empty_like(sends[(mu, 0)])
    exchange_tensors(sends, recvs)
    return recvs


def _hop_lex(u_lex, geo, psi_lex, mu, fwd: bool, dagger: bool, P,
             halo: Optional[dict] = None):
    """One directed Wilson hop on lex fields:
    fwd: U_mu(x) P(-mu) psi(x+mu); bwd: U_mu(x-mu)^d P(+mu) psi(x-mu).
    halo (multi-rank): {"mask", "face" {(mu,edge): lex idx},
    "psi" {(mu,dir): ghost}, "u" {(mu,0): link ghost}} — boundary sites
    of partitioned dims read the neighbor RANK's psi/U instead of the
    local periodic wrap."""
    sgn = 1 if not dagger else 0
    cut = halo is not None and (halo["mask"] >> mu) & 1
    if fwd:
        idx = geo.neighbor_lex(mu, +1).to(psi_lex.device)
        psi_n = psi_lex[idx]
        if cut:
            psi_n[halo["face"][(mu, 1)]] = halo["psi"][(mu, 1)]
        proj = torch.einsum("st,vtc->vsc", P[mu, 1 - sgn], psi_n)
        return torch.einsum("vij,vsj->vsi", u_lex[mu], proj)
    idx = geo.neighbor_lex(mu, -1).to(psi_lex.device)
    psi_n = psi_lex[idx]
    u_n = u_lex[mu][idx]
    if cut:
        f0 = halo["face"][(mu, 0)]
        psi_n[f0] = halo["psi"][(mu, 0)]
        u_n = u_n.clone()
        u_n[f0] = halo["u"][(mu, 0)]
    proj = torch.einsum("st,vtc->vsc", P[mu, sgn], psi_n)
    return torch.einsum("vji,vsj->vsi", u_n.conj(), proj)


class CoarseOp:
    """Explicit coarse operator with X/Y tensors; non-hermitian M plus
    Mdag; presents new_vec/apply for the coarse-level solvers.

    Multi-rank (mask != 0): the local tensors cover this rank's coarse
    sub-lattice; apply() exchanges the c faces of partitioned dims (the
    coarse ghost machinery of ref lib/dslash_coarse.hpp:30 /
    dirac_coarse.cpp), and the dagger runs in GATHER form using the
    neighbor rank's Y faces (exchanged once at construction)."""

    def __init__(self, X: torch.Tensor, Y: List[torch.Tensor], cd,
                 mask: int = 0):
        self.X = X
        self.Y = Y  # list of 8: dir = 2*mu+fwd
        self.cd = tuple(cd)
        self.Na, self.Nc = X.shape[0], X.shape[1]
        self.nbr = _coarse_neighbors(cd).to(X.device)
        self.mask = mask
        self.face = {}
        self.Yg = {}
        if mask:
            from ..parallel.halo import active_dims, exchange_tensors
            for mu in active_dims(mask):
                self.face[(mu, 0)] = _lex_face(cd, mu, 0).to(X.device)
                self.face[(mu, 1)] = _lex_face(cd, mu, cd[mu] - 1).to(X.device)
            # Y ghosts for the dagger gather: the fwd gather at x=hi needs
            # the +mu neighbor's Y[2mu+0] on its x=0 face; the bwd gather
            # at x=0 needs the -mu neighbor's Y[2mu+1] on its x=hi face.
            sends, recvs = {}, {}
            for mu in active_dims(mask):
                sends[(mu, 0)] = Y[2 * mu + 0][self.face[(mu, 0)]].contiguous()
                sends[(mu, 1)] = Y[2 * mu + 1][self.face[(mu, 1)]].contiguous()
                recvs[(mu, 0)] = torch.empty_like(sends[(mu, 1)])
                recvs[(mu, 1)] = torch.empty_like(sends[(mu, 0)])
            exchange_tensors(sends, recvs)
            self.Yg = recvs

    def _exchange_c(self, c: torch.Tensor) -> Optional[dict]:
        if not self.mask:
            return None
        return _exchange_lex_faces(c, self.cd, self.mask, self.face)

    def _nbr_c(self, c: torch.Tensor, d: int, ghosts) -> torch.Tensor:
        """c at the d-neighbor of every site; partitioned-boundary rows
        come from the ghost faces (cn is a fresh gather, safe to patch)."""
        cn = c[self.nbr[:, d]]
        mu, fwd = d // 2, d % 2
        if self.mask and (self.mask >> mu) & 1:
            cn[self.face[(mu, fwd)]] = ghosts[(mu, fwd)]
        return cn

    # -- MFMA fast path (GPU): one fused 9-matrix kernel ------------------
    def _hip_setup(self):
        """Lazy: stacked [9,Na,Nc,Nc] c64 matrices + ghost-extended nbr
        table for k_coarse_dslash_mfma (csrc/coarse.hip)."""
        if getattr(self, "_m9", None) is not None:
            return
        dev = self.X.device
        self._m9 = torch.stack([self.X] + list(self.Y)).to(
            torch.complex64).contiguous()
        nbr9 = torch.empty((self.Na, 9), dtype=torch.int64, device=dev)
        nbr9[:, 0] = torch.arange(self.Na, device=dev)
        nbr9[:, 1:] = self.nbr
        self._ghost_order = []  # [(mu, fwd, face_len)] in c_ext append order
        off = self.Na
        for d in range(8):
            mu, fwd = d // 2, d % 2
            if self.mask and (self.mask >> mu) & 1:
                f = self.face[(mu, fwd)]
                nbr9[f, 1 + d] = off + torch.arange(len(f), device=dev)
                self._ghost_order.append((mu, fwd, len(f)))
                off += len(f)
        self._n_ghost = off - self.Na
        self._nbr9 = nbr9.contiguous()

    def _apply_hip(self, c: torch.Tensor, ghosts) -> torch.Tensor:
        from ..ops.dispatch import hip_ext
        self._hip_setup()
        nr = 1 if c.dim() == 2 else c.shape[2]
        cb = c.reshape(self.Na, self.Nc, nr).to(torch.complex64)
        if self._n_ghost:
            gs = [ghosts[(mu, fwd)].reshape(n, self.Nc, nr)
                  for mu, fwd, n in self._ghost_order]
            cb = torch.cat([cb] + [g.to(torch.complex64) for g in gs])
        cb = cb.contiguous()
        out = torch.empty((self.Na, self.Nc, nr), dtype=torch.complex64,
                          device=c.device)
        hip_ext().coarse_dslash_mfma(self._m9, self._nbr9, cb, out,
                                     self.Na, self.Nc, nr)
        out = out.to(c.dtype)
        return out.reshape(self.Na, self.Nc) if c.dim() == 2 else out

    def apply_block(self, C: torch.Tensor) -> torch.Tensor:
        """Forward apply on an RHS block [Na, Nc, NR] (multi-RHS coarse
        dslash — the reference's dslash_coarse_mma shape)."""
        ghosts = self._exchange_c(C)
        if (C.device.type == "cuda" and self.Nc % 16 == 0
                and C.shape[2] <= 16 and getattr(self, "use_hip", True)):
            return self._apply_hip(C, ghosts)
        out = torch.einsum("aij,ajn->ain", self.X, C)
        for d in range(8):
            out += torch.einsum("aij,ajn->ain", self.Y[d],
                                self._nbr_c(C, d, ghosts))
        return out

    def apply(self, c: torch.Tensor, dagger: bool = False) -> torch.Tensor:
        if c.dim() == 3 and not dagger:
            return self.apply_block(c)
        ghosts = self._exchange_c(c)
        if (not dagger and c.device.type == "cuda" and self.Nc % 16 == 0
                and getattr(self, "use_hip", True)):
            return self._apply_hip(c, ghosts)
        if not dagger:
            out = torch.einsum("aij,aj->ai", self.X, c)
            for d in range(8):
                out += torch.einsum("aij,aj->ai", self.Y[d],
                                    self._nbr_c(c, d, ghosts))
            return out
        # dagger in gather form: out[a] = X(a)^d c(a)
        #   + sum_d Y[opp(d)](nbr(a,d))^d c(nbr(a,d))   (opp(d) = d^1)
        out = torch.einsum("aji,aj->ai", self.X.conj(), c)
        for d in range(8):
            mu, fwd = d // 2, d % 2
            src = self.nbr[:, d]
            Yn = self.Y[d ^ 1][src]
            cn = c[src]
            if self.mask and (self.mask >> mu) & 1:
                f = self.face[(mu, fwd)]
                cn[f] = ghosts[(mu, fwd)]
                Yn = Yn.clone()
                Yn[f] = self.Yg[(mu, fwd)]
            out += torch.einsum("aji,aj->ai", Yn.conj(), cn)
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
    from ..parallel import comms
    geo = transfer.geo
    dev = transfer.device
    u_cb = op.gauge.to_complex()
    lo = geo.lex_of_cb.to(dev)
    u_lex = torch.empty((4, geo.volume, 3, 3), dtype=u_cb.dtype, device=dev)
    u_lex[:, lo[0]] = u_cb[:, 0]
    u_lex[:, lo[1]] = u_cb[:, 1]
    P = _gamma_tensors(dev, u_cb.dtype)
    mask = comms.comm_mask()
    halo = None
    if mask:
        # rank-boundary hop data: the -mu neighbor's U_mu (its x=hi face)
        # for bwd hops at x=0; psi ghosts are refreshed per column family.
        from ..parallel.halo import active_dims
        faces = {}
        for mu in active_dims(mask):
            faces[(mu, 0)] = _lex_face(geo.dims, mu, 0).to(dev)
            faces[(mu, 1)] = _lex_face(geo.dims, mu, geo.dims[mu] - 1).to(dev)
        ug = {}
        for mu in active_dims(mask):
            g = _exchange_lex_faces(u_lex[mu], geo.dims, mask, faces)
            ug[(mu, 0)] = g[(mu, 0)]
        halo = {"mask": mask, "face": faces, "u": ug, "psi": None}
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
            if halo is not None:
                halo["psi"] = _exchange_lex_faces(V_full, geo.dims, mask,
                                                  halo["face"])
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
                                          False, P, halo=halo)
                    Wa = W[sb]                      # [Na, B, 4, 3]
                    m = bnd[(mu, fwd)].unsqueeze(-1).unsqueeze(-1)
                    W_int = torch.where(m, torch.zeros_like(Wa), Wa)
                    W_bnd = torch.where(m, Wa, torch.zeros_like(Wa))
                    X[:, :, j] += _restrict_agg(transfer, W_int)
                    Y[2 * mu + fwd][:, :, j] += _restrict_agg(transfer, W_bnd)
    return CoarseOp(X, Y, transfer.coarse_dims, mask=mask)


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


def gdot(a: torch.Tensor, b: torch.Tensor):
    """Globally-reduced <a, b> (complex scalar tensor) — rank-local sum
    followed by an allreduce when distributed."""
    from ..parallel import comms
    s = (a.conj() * b).sum()
    if comms.is_distributed():
        comms.allreduce_tensor(torch.view_as_real(s))
    return s


def gnorm2(a: torch.Tensor) -> float:
    return gdot(a, a).real.item()


def coarse_bicgstab(op: CoarseOp, b: torch.Tensor, *, tol: float = 1e-8,
                    maxiter: int = 1000) -> torch.Tensor:
    """BiCGStab on the coarse tensors (all torch ops — rocBLAS batched
    GEMM on GPU; role of the CA-GCR coarse solver, lib/multigrid.cpp).
    All inner products are GLOBAL (multi-rank coarse grids)."""
    x = torch.zeros_like(b)
    r = b.clone()
    r0 = r.clone()
    p = r.clone()
    b2 = gnorm2(r)
    if b2 == 0:
        return x
    stop = tol * tol * b2
    rho = gdot(r0, r)
    for _ in range(maxiter):
        v = op.apply(p)
        r0v = gdot(r0, v)
        if r0v.abs().item() == 0:
            break
        alpha = rho / r0v
        s = r - alpha * v
        t = op.apply(s)
        t2 = gdot(t, t).real
        if t2.item() == 0:
            x = x + alpha * p
            break
        omega = gdot(t, s) / t2
        x = x + alpha * p + omega * s
        r = s - omega * t
        r2 = gnorm2(r)
        if r2 < stop:
            break
        rho_new = gdot(r0, r)
        beta = (rho_new / rho) * (alpha / omega)
        rho = rho_new
        p = r + beta * (p - omega * v)
    return x
