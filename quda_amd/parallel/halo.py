"""Halo (ghost-zone) exchange engine (ref: lib/dslash_policy.hpp +
include/kernels/dslash_pack.cuh + lattice_field.cpp createComms — rebuilt
MI355X-first: spin-projected face buffers packed by HIP kernels, exchanged
with torch.distributed point-to-point ops — RCCL send/recv over xGMI on a
GPU node, gloo for CPU multi-process tests; self-wraparound when a
partitioned dim has grid size 1, which exercises the full comms code path
on a single process exactly like the reference's --partition flags).

Buffer/key conventions (mirror csrc/halo.h):
  key (mu, dir): dir=1 ghost arrives from the +mu neighbor (feeds the
  forward hop), dir=0 from -mu (backward hop). send[(mu,0)] is packed from
  the x_mu=0 face and travels to the -mu neighbor; send[(mu,1)] from the
  x_mu=L-1 face to +mu.

Deterministic op ordering: every rank issues its isend/irecv ops sorted by
(mu, travel-direction) where travel=0 is the +mu-going message
(send[(mu,1)] on the sender, recv (mu,0)... on the receiver recv[(mu,0)]
arrives from -mu, i.e. it is the +mu-going message) — this keeps NCCL's
per-peer order-based matching consistent even when the +mu and -mu
neighbor are the same rank (grid extent 2).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from ..fields.geometry import LatticeGeometry
from . import comms

Key = Tuple[int, int]  # (mu, dir)


def active_dims(mask: int) -> List[int]:
    return [mu for mu in range(4) if (mask >> mu) & 1]


def _travel_key(mu: int, travel: int, extra: int = 0):
    return (mu, travel, extra)


def exchange_tensors(sends: Dict[Key, torch.Tensor],
                     recvs: Dict[Key, torch.Tensor]) -> None:
    """Blocking exchange (see exchange_tensors_start)."""
    for r in exchange_tensors_start(sends, recvs):
        r.wait()


def exchange_tensors_start(sends: Dict[Key, torch.Tensor],
                           recvs: Dict[Key, torch.Tensor]) -> list:
    """Start exchanging face tensors: recv[(mu,1)] <- +mu neighbor's
    send[(mu,0)], recv[(mu,0)] <- -mu neighbor's send[(mu,1)]. Complex
    tensors are viewed as real. Returns the request list; self-wraparound
    copies run immediately (stream-ordered). wait() on each request before
    touching recv buffers (on NCCL that makes the CURRENT stream wait — the
    host does not block, so an interior kernel launched in between overlaps
    with the transfer)."""
    ops = []  # (order_key, is_send, tensor, peer)
    # self-wraparound check must use the GLOBAL rank: neighbor_rank returns
    # global ranks, and inside a split-grid sub-communicator comm_rank() is
    # group-local
    my_rank = dist.get_rank() if comms.is_distributed() else 0
    local_copies = []
    for (mu, d), s in sends.items():
        peer = comms.neighbor_rank(mu, -1 if d == 0 else +1)
        # send (mu,0) goes -mu => it is the "-mu-going" message: travel=1
        travel = 1 if d == 0 else 0
        if peer == my_rank:
            # self-wraparound: recv[(mu, 1-d)] = send[(mu, d)]
            local_copies.append(((mu, 1 - d), s))
            continue
        ops.append((_travel_key(mu, travel, 0), True, s, peer))
    for (mu, d), r in recvs.items():
        peer = comms.neighbor_rank(mu, -1 if d == 0 else +1)
        if peer == my_rank:
            continue
        # recv (mu,0) arrives from -mu travelling +mu: travel=0
        travel = 0 if d == 0 else 1
        ops.append((_travel_key(mu, travel, 1), False, r, peer))
    for key, s in local_copies:
        recvs[key].copy_(s)
    if not ops:
        return []
    ops.sort(key=lambda o: (o[0], not o[1]))  # sends before recvs per key
    reqs = []
    for _, is_send, t, peer in ops:
        tt = torch.view_as_real(t) if t.is_complex() else t
        if is_send:
            reqs.append(dist.isend(tt.contiguous(), peer))
        else:
            assert tt.is_contiguous()
            reqs.append(dist.irecv(tt, peer))
    return reqs


# ---------------------------------------------------------------------------
# native (device-layout) spinor halo: persistent buffers + HIP pack kernels
# ---------------------------------------------------------------------------

GHOST_W0 = {"double": 2, "single": 4, "half": 4, "quarter": 4}  # max reals per ghost chunk


def ghost_width(ncomp: int, precision: str) -> int:
    """Mirrors csrc/halo.h GhostAcc<> chunk width."""
    w = GHOST_W0[precision]
    while ncomp % w:
        w //= 2
    return w


class SpinorHalo:
    """Persistent send/recv ghost buffers for one (geometry, precision,
    device, mask, ncomp) signature (role of the reference's static ghost
    arenas, lattice_field.h:250). ncomp = 12 (spin-projected Wilson) or
    6 (staggered full site)."""

    def __init__(self, geo: LatticeGeometry, precision: str, device,
                 mask: int, ncomp: int = 12, depth: int = 1):
        from ..fields.layout import DTYPE_OF
        self.geo = geo
        self.precision = precision
        self.mask = mask
        self.ncomp = ncomp
        self.depth = depth  # ghost layers (3 for the staggered Naik term)
        self.device = torch.device(device)
        self.send: Dict[Key, torch.Tensor] = {}
        self.recv: Dict[Key, torch.Tensor] = {}
        self.send_nrm: Dict[Key, torch.Tensor] = {}
        self.recv_nrm: Dict[Key, torch.Tensor] = {}
        gw = ghost_width(ncomp, precision)
        dt = DTYPE_OF[precision]
        for mu in active_dims(mask):
            fcb = geo.face_volume_cb(mu) * depth
            for d in (0, 1):
                shape = (ncomp // gw, fcb, gw)
                self.send[(mu, d)] = torch.empty(shape, dtype=dt, device=device)
                self.recv[(mu, d)] = torch.empty(shape, dtype=dt, device=device)
                if precision in ("half", "quarter"):
                    self.send_nrm[(mu, d)] = torch.empty(fcb, dtype=torch.float32,
                                                         device=device)
                    self.recv_nrm[(mu, d)] = torch.empty(fcb, dtype=torch.float32,
                                                         device=device)

    def pack(self, ext, inp, parity: int, dagger: bool) -> None:
        """Pack all active faces of `inp` (the dslash input, at `parity`):
        Wilson ghosts carry the projector the consuming hop applies,
        staggered ghosts the full site."""
        geo = self.geo
        empty = torch.empty(0, dtype=torch.float32, device=self.device)
        from ..ops.dispatch import norm_or_empty
        for mu in active_dims(self.mask):
            fcb = geo.face_volume_cb(mu)
            for d in (0, 1):
                edge = 0 if d == 0 else 1
                if self.ncomp == 6:
                    ext.pack_face_stag(self.send[(mu, d)],
                                       self.send_nrm.get((mu, d), empty),
                                       inp.data, norm_or_empty(inp),
                                       list(geo.dims), geo.parity_offset,
                                       geo.volume_cb, parity, mu, edge, fcb,
                                       self.depth)
                else:
                    s01 = d ^ (1 if dagger else 0)
                    ext.pack_face(self.send[(mu, d)],
                                  self.send_nrm.get((mu, d), empty),
                                  inp.data, norm_or_empty(inp),
                                  list(geo.dims), geo.parity_offset,
                                  geo.volume_cb, parity, mu, s01, edge, fcb)

    def exchange(self) -> None:
        for r in self.exchange_start():
            r.wait()

    def exchange_start(self) -> list:
        reqs = exchange_tensors_start(self.send, self.recv)
        if self.precision in ("half", "quarter"):
            reqs += exchange_tensors_start(self.send_nrm, self.recv_nrm)
        return reqs

    def ghost_args(self):
        """(ghost[8], ghost_nrm[8], face_cb[4]) lists for ext.dslash_*."""
        empty = torch.empty(0, dtype=self.recv[next(iter(self.recv))].dtype,
                            device=self.device) if self.recv else None
        empty_n = torch.empty(0, dtype=torch.float32, device=self.device)
        ghosts, nrms = [], []
        for mu in range(4):
            for d in (0, 1):
                g = self.recv.get((mu, d))
                ghosts.append(g if g is not None else empty)
                n = self.recv_nrm.get((mu, d))
                nrms.append(n if n is not None else empty_n)
        face_cb = [self.geo.face_volume_cb(mu) for mu in range(4)]
        return ghosts, nrms, face_cb


_HALO_CACHE: Dict[tuple, SpinorHalo] = {}


def get_spinor_halo(geo: LatticeGeometry, precision: str, device,
                    mask: int, ncomp: int = 12, depth: int = 1) -> SpinorHalo:
    key = (geo.dims, geo.parity_offset, precision, str(device), mask, ncomp,
           depth)
    h = _HALO_CACHE.get(key)
    if h is None:
        h = SpinorHalo(geo, precision, device, mask, ncomp, depth)
        _HALO_CACHE[key] = h
    return h


# ---------------------------------------------------------------------------
# oracle (complex-layout) halo for the CPU reference path
# ---------------------------------------------------------------------------

def exchange_psi_oracle(psi: torch.Tensor, geo: LatticeGeometry,
                        parity_in: int, mask: int,
                        depth: int = 1) -> Dict[Key, torch.Tensor]:
    """Exchange full (unprojected) spinor faces of `psi` ([V_cb,4,3] or
    [V_cb,3] complex at parity_in). Returns {(mu,dir): faces} in
    ghost-index order; with depth > 1 each face is [depth, Fcb, ...] where
    layer l of (mu,1) holds the +mu neighbor's x_mu = l sites and layer l
    of (mu,0) the -mu neighbor's x_mu = X-1-l sites (Naik nFace=3)."""
    sends, recvs = {}, {}
    for mu in active_dims(mask):
        hi = geo.dims[mu] - 1
        lo_layers = [psi[geo.face_index_cb(parity_in, mu, l)]
                     for l in range(depth)]
        hi_layers = [psi[geo.face_index_cb(parity_in, mu, hi - l)]
                     for l in range(depth)]
        sends[(mu, 0)] = torch.stack(lo_layers).contiguous()
        sends[(mu, 1)] = torch.stack(hi_layers).contiguous()
        recvs[(mu, 0)] = torch.empty_like(sends[(mu, 1)])
        recvs[(mu, 1)] = torch.empty_like(sends[(mu, 0)])
    exchange_tensors(sends, recvs)
    if depth == 1:
        recvs = {k: v[0] for k, v in recvs.items()}
    return recvs


# ---------------------------------------------------------------------------
# 5-d (domain-wall) halo: per-slice packs into one slab buffer, ONE exchange
# ---------------------------------------------------------------------------

class DwfHalo:
    """Ghost buffers for 5-d fields: one slab per (mu,dir) shaped
    [Ls, 12/gw, Fcb, gw]; each s-slice packs into its slab and the whole
    buffer ships in a single message per (mu,dir)."""

    def __init__(self, geo: LatticeGeometry, precision: str, device,
                 mask: int, ls: int):
        from ..fields.layout import DTYPE_OF
        self.geo = geo
        self.precision = precision
        self.mask = mask
        self.ls = ls
        self.device = torch.device(device)
        gw = ghost_width(12, precision)
        dt = DTYPE_OF[precision]
        self.send, self.recv = {}, {}
        self.send_nrm, self.recv_nrm = {}, {}
        for mu in active_dims(mask):
            fcb = geo.face_volume_cb(mu)
            for d in (0, 1):
                shape = (ls, 12 // gw, fcb, gw)
                self.send[(mu, d)] = torch.empty(shape, dtype=dt, device=device)
                self.recv[(mu, d)] = torch.empty(shape, dtype=dt, device=device)
                if precision in ("half", "quarter"):
                    self.send_nrm[(mu, d)] = torch.empty((ls, fcb),
                                                         dtype=torch.float32,
                                                         device=device)
                    self.recv_nrm[(mu, d)] = torch.empty((ls, fcb),
                                                         dtype=torch.float32,
                                                         device=device)

    def pack_exchange(self, ext, inp, parity: int, dagger: bool) -> None:
        geo = self.geo
        Vcb = geo.volume_cb
        empty = torch.empty(0, dtype=torch.float32, device=self.device)
        for mu in active_dims(self.mask):
            fcb = geo.face_volume_cb(mu)
            for d in (0, 1):
                s01 = d ^ (1 if dagger else 0)
                edge = 0 if d == 0 else 1
                for s in range(self.ls):
                    nrm = (self.send_nrm[(mu, d)][s]
                           if self.precision in ("half", "quarter") else empty)
                    ext.pack_face(self.send[(mu, d)][s], nrm,
                                  inp.data, _norm_or_empty(inp),
                                  list(geo.dims), geo.parity_offset, Vcb,
                                  parity, mu, s01, edge, fcb,
                                  v_stride=self.ls * Vcb, s_offset=s * Vcb)
        exchange_tensors(self.send, self.recv)
        if self.precision in ("half", "quarter"):
            exchange_tensors(self.send_nrm, self.recv_nrm)

    def ghost_args(self, s: int):
        empty_n = torch.empty(0, dtype=torch.float32, device=self.device)
        ghosts, nrms = [], []
        for mu in range(4):
            for d in (0, 1):
                g = self.recv.get((mu, d))
                if g is None:
                    ghosts.append(torch.empty(0, dtype=next(iter(self.recv.values())).dtype,
                                              device=self.device))
                    nrms.append(empty_n)
                else:
                    ghosts.append(g[s])
                    nrms.append(self.recv_nrm[(mu, d)][s]
                                if self.precision in ("half", "quarter") else empty_n)
        face_cb = [self.geo.face_volume_cb(mu) for mu in range(4)]
        return ghosts, nrms, face_cb


def _norm_or_empty(f):
    from ..ops.dispatch import norm_or_empty
    return norm_or_empty(f)


_DWF_HALO_CACHE: Dict[tuple, DwfHalo] = {}


def get_dwf_halo(geo: LatticeGeometry, precision: str, device, mask: int,
                 ls: int) -> DwfHalo:
    key = (geo.dims, precision, str(device), mask, ls)
    h = _DWF_HALO_CACHE.get(key)
    if h is None:
        h = DwfHalo(geo, precision, device, mask, ls)
        _DWF_HALO_CACHE[key] = h
    return h


def exchange_psi5_oracle(psi5: torch.Tensor, geo: LatticeGeometry,
                         parity_in: int, mask: int, ls: int):
    """Oracle 5-d spinor face exchange: psi5 [Ls*V,4,3] -> ghosts
    {(mu,dir): [Ls, Fcb, 4, 3]}."""
    V = geo.volume_cb
    v = psi5.reshape(ls, V, 4, 3)
    sends, recvs = {}, {}
    for mu in active_dims(mask):
        hi = geo.dims[mu] - 1
        idx0 = geo.face_index_cb(parity_in, mu, 0).to(psi5.device)
        idx1 = geo.face_index_cb(parity_in, mu, hi).to(psi5.device)
        sends[(mu, 0)] = v[:, idx0].contiguous()
        sends[(mu, 1)] = v[:, idx1].contiguous()
        recvs[(mu, 0)] = torch.empty_like(sends[(mu, 1)])
        recvs[(mu, 1)] = torch.empty_like(sends[(mu, 0)])
    exchange_tensors(sends, recvs)
    return recvs


# ---------------------------------------------------------------------------
# distributed lexicographic-field shift (gauge-sector halo: staples, field
# strength, smearing, forces become multi-rank correct through this)
# ---------------------------------------------------------------------------

def shift_lex(f: torch.Tensor, geo: LatticeGeometry, mu: int, disp: int
              ) -> torch.Tensor:
    """f: [Vlex, ...] -> f(x + disp*mu). |disp| must be 1 (compose for
    more). On partitioned dims the wrapped face comes from the neighbor
    rank (ref: the gauge exchangeGhost/exchangeExtendedGhost role)."""
    assert disp in (1, -1)
    from . import comms
    idx = geo.neighbor_lex(mu, disp).to(f.device)
    out = f[idx]
    if not ((comms.comm_mask() >> mu) & 1):
        return out
    hi = geo.dims[mu] - 1
    if disp == 1:
        # my x_mu = hi sites need the +mu neighbor's x_mu = 0 slab
        send = {(mu, 0): f[_face_lex(geo, mu, 0).to(f.device)].contiguous()}
        recv = {(mu, 1): torch.empty_like(send[(mu, 0)])}
        exchange_tensors(send, recv)
        out[_face_lex(geo, mu, hi).to(f.device)] = recv[(mu, 1)]
    else:
        send = {(mu, 1): f[_face_lex(geo, mu, hi).to(f.device)].contiguous()}
        recv = {(mu, 0): torch.empty_like(send[(mu, 1)])}
        exchange_tensors(send, recv)
        out[_face_lex(geo, mu, 0).to(f.device)] = recv[(mu, 0)]
    return out


def _face_lex(geo: LatticeGeometry, mu: int, edge: int) -> torch.Tensor:
    """Lex indices of the face x_mu == edge, in natural lex order (both
    ends agree: transverse coords are identical across the boundary)."""
    key = ("face_lex", mu, edge)
    cache = geo.__dict__.setdefault("_nbr_cache", {})
    if key not in cache:
        c = geo.coords[:, mu]
        cache[key] = (c == edge).nonzero(as_tuple=True)[0].contiguous()
    return cache[key]


class BatchSpinorHalo:
    """Multi-RHS ghost buffers: per (mu,dir) ONE contiguous tensor
    [n_rhs, ncomp/gw, depth*Fcb, gw] so the whole batch ships as a single
    message per face (ref: dslash create_comms_batch, the merged-halo
    multi-RHS path — the per-RHS kernels then consume their slice)."""

    def __init__(self, geo: LatticeGeometry, precision: str, device,
                 mask: int, n_rhs: int, ncomp: int = 12):
        from ..fields.layout import DTYPE_OF
        self.geo = geo
        self.precision = precision
        self.mask = mask
        self.n_rhs = n_rhs
        self.ncomp = ncomp
        self.device = torch.device(device)
        gw = ghost_width(ncomp, precision)
        dt = DTYPE_OF[precision]
        self.send, self.recv = {}, {}
        self.send_nrm, self.recv_nrm = {}, {}
        for mu in active_dims(mask):
            fcb = geo.face_volume_cb(mu)
            for d in (0, 1):
                shape = (n_rhs, ncomp // gw, fcb, gw)
                self.send[(mu, d)] = torch.empty(shape, dtype=dt,
                                                 device=device)
                self.recv[(mu, d)] = torch.empty(shape, dtype=dt,
                                                 device=device)
                if precision in ("half", "quarter"):
                    self.send_nrm[(mu, d)] = torch.empty(
                        (n_rhs, fcb), dtype=torch.float32, device=device)
                    self.recv_nrm[(mu, d)] = torch.empty(
                        (n_rhs, fcb), dtype=torch.float32, device=device)

    def pack_one(self, ext, i: int, inp, parity: int, dagger: bool) -> None:
        geo = self.geo
        empty = torch.empty(0, dtype=torch.float32, device=self.device)
        from ..ops.dispatch import norm_or_empty
        for mu in active_dims(self.mask):
            fcb = geo.face_volume_cb(mu)
            for d in (0, 1):
                s01 = d ^ (1 if dagger else 0)
                nrm = self.send_nrm.get((mu, d))
                ext.pack_face(self.send[(mu, d)][i],
                              nrm[i] if nrm is not None else empty,
                              inp.data, norm_or_empty(inp),
                              list(geo.dims), geo.parity_offset,
                              geo.volume_cb, parity, mu, s01, d, fcb)

    def exchange(self) -> None:
        reqs = exchange_tensors_start(self.send, self.recv)
        if self.precision in ("half", "quarter"):
            reqs += exchange_tensors_start(self.send_nrm, self.recv_nrm)
        for r in reqs:
            r.wait()

    def exchange_start(self) -> list:
        reqs = exchange_tensors_start(self.send, self.recv)
        if self.precision in ("half", "quarter"):
            reqs += exchange_tensors_start(self.send_nrm, self.recv_nrm)
        return reqs

    def ghost_args(self, i: int):
        """Per-RHS (ghost[8], ghost_nrm[8], face_cb[4]) slice views."""
        empty = torch.empty(0, dtype=next(iter(self.recv.values())).dtype,
                            device=self.device) if self.recv else None
        empty_n = torch.empty(0, dtype=torch.float32, device=self.device)
        ghosts, nrms = [], []
        for mu in range(4):
            for d in (0, 1):
                g = self.recv.get((mu, d))
                ghosts.append(g[i] if g is not None else empty)
                n = self.recv_nrm.get((mu, d))
                nrms.append(n[i] if n is not None else empty_n)
        face_cb = [self.geo.face_volume_cb(mu) for mu in range(4)]
        return ghosts, nrms, face_cb


_BATCH_HALO_CACHE: Dict[tuple, "BatchSpinorHalo"] = {}


def get_batch_halo(geo: LatticeGeometry, precision: str, device, mask: int,
                   n_rhs: int, ncomp: int = 12) -> BatchSpinorHalo:
    key = (geo.dims, geo.parity_offset, precision, str(device), mask,
           n_rhs, ncomp)
    h = _BATCH_HALO_CACHE.get(key)
    if h is None:
        h = BatchSpinorHalo(geo, precision, device, mask, n_rhs, ncomp)
        _BATCH_HALO_CACHE[key] = h
    return h
