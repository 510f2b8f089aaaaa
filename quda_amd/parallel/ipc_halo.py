"""HIP-IPC remote-write halo exchange (role of the reference's P2P
remote-write transport, lib/targets/cuda/comm_target.cpp:41-134 and the
QUDA_ENABLE_P2P remote-write policy): the pack kernel writes the face
DIRECTLY into the ±mu neighbor rank's recv buffer over xGMI — no send
buffer, no RCCL message, one write stream per (dim, dir) so every xGMI
link carries its own face concurrently (SURVEY.md §2.11 MI355X mapping).

Setup (once per signature): every rank exports hipIpcMemHandle_t of its
recv buffers (+ fp norm buffers) and all-gathers them over the control
backend (gloo or RCCL); each rank opens its neighbors' handles. Data
path per dslash: pack_face_ptr into the peer pointer; synchronization in
this first implementation is device-drain + barrier (pack kernels are
stream-ordered; `torch.cuda.synchronize` guarantees the remote writes
landed, the barrier that EVERY rank's did) — the flag-write/wait variant
is the next refinement. Self-wraparound dims pack into the own recv
buffer directly (stream-ordered, no sync needed).
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch
import torch.distributed as dist

from . import comms
from .halo import SpinorHalo, active_dims

Key = Tuple[int, int]


class IpcSpinorHalo(SpinorHalo):
    """SpinorHalo whose exchange is a remote WRITE into the peer."""

    def __init__(self, geo, precision: str, device, mask: int,
                 ncomp: int = 12, depth: int = 1):
        super().__init__(geo, precision, device, mask, ncomp, depth)
        from ..ops.dispatch import hip_ext
        ext = hip_ext()
        my = dist.get_rank() if comms.is_distributed() else 0
        # export handles of MY recv buffers
        handles = {}
        for key, t in self.recv.items():
            handles[key] = ext.ipc_get_handle(t)
        for key, t in self.recv_nrm.items():
            handles[("n", *key)] = ext.ipc_get_handle(t)
        world = dist.get_world_size() if comms.is_distributed() else 1
        gathered = [None] * world
        if comms.is_distributed():
            dist.all_gather_object(gathered, handles)
        else:
            gathered[0] = handles
        # open the peers I write into: my (mu,0)-face goes to the -mu
        # neighbor's recv[(mu,1)]; my (mu,1)-face to +mu's recv[(mu,0)]
        # (exchange_tensors semantics, halo.py)
        self.peer_ptr: Dict[Key, int] = {}      # keyed by MY send key
        self.peer_nrm_ptr: Dict[Key, int] = {}
        self._opened = []
        self.self_wrap: Dict[Key, bool] = {}
        for mu in active_dims(mask):
            for d in (0, 1):
                peer = comms.neighbor_rank(mu, -1 if d == 0 else +1)
                dst_key = (mu, 1 - d)
                if peer == my:
                    self.self_wrap[(mu, d)] = True
                    continue
                self.self_wrap[(mu, d)] = False
                h = gathered[peer][dst_key]
                p, pbase = ext.ipc_open_handle(h)
                self.peer_ptr[(mu, d)] = p
                self._opened.append(pbase)
                hn = gathered[peer].get(("n", *dst_key))
                if hn is not None:
                    pn, pnb = ext.ipc_open_handle(hn)
                    self.peer_nrm_ptr[(mu, d)] = pn
                    self._opened.append(pnb)
        if comms.is_distributed():
            dist.barrier()  # all mappings open before anyone packs

    def close(self):
        from ..ops.dispatch import hip_ext
        ext = hip_ext()
        for p in self._opened:
            ext.ipc_close_handle(p)
        self._opened = []

    def pack_remote(self, ext, inp, parity: int, dagger: bool) -> None:
        """Pack every active face straight into its consumer's memory."""
        geo = self.geo
        from ..ops.dispatch import norm_or_empty
        empty = torch.empty(0, dtype=torch.float32, device=self.device)
        for mu in active_dims(self.mask):
            fcb = geo.face_volume_cb(mu)
            for d in (0, 1):
                edge = 0 if d == 0 else 1
                s01 = d ^ (1 if dagger else 0)
                if self.self_wrap[(mu, d)]:
                    dst = self.recv[(mu, 1 - d)]
                    dn = self.recv_nrm.get((mu, 1 - d), empty)
                    ext.pack_face(dst, dn, inp.data, norm_or_empty(inp),
                                  list(geo.dims), geo.parity_offset,
                                  geo.volume_cb, parity, mu, s01, edge, fcb)
                else:
                    pn = self.peer_nrm_ptr.get((mu, d), 0)
                    ext.pack_face_ptr(self.peer_ptr[(mu, d)], pn,
                                      inp.data, norm_or_empty(inp),
                                      list(geo.dims), geo.parity_offset,
                                      geo.volume_cb, parity, mu, s01, edge,
                                      fcb, 0, 0)

    def complete(self) -> None:
        """All remote writes landed everywhere (drain + barrier)."""
        if comms.is_distributed():
            torch.cuda.synchronize()
            dist.barrier()


_IPC_CACHE: Dict[tuple, IpcSpinorHalo] = {}


def get_ipc_halo(geo, precision: str, device, mask: int,
                 ncomp: int = 12) -> IpcSpinorHalo:
    key = (geo.dims, precision, str(device), mask, ncomp,
           comms.grid_dims(), comms.grid_coords())
    h = _IPC_CACHE.get(key)
    if h is None:
        h = IpcSpinorHalo(geo, precision, device, mask, ncomp)
        _IPC_CACHE[key] = h
    return h
