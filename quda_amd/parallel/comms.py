"""Process-grid communication layer (ref: include/comm_quda.h +
lib/communicator_mpi.cpp — rebuilt over torch.distributed: the "nccl"
backend IS RCCL on ROCm and rides xGMI on one node; "gloo" covers CPU
multi-process tests).

A 4-D grid topology maps ranks onto (Gx,Gy,Gz,Gt) with x fastest (same
convention as the reference's default lex rank order,
interface_quda.cpp:330). Default for N ranks: partition the T dimension
only — MI355X's 288 GB HBM favors few, fat local volumes.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional, Tuple

import torch
import torch.distributed as dist

_STATE = {"grid": (1, 1, 1, 1), "coords": (0, 0, 0, 0), "initialized": False,
          "forced_mask": 0, "group": None, "group_ranks": None,
          "group_rank": 0}


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def comm_rank() -> int:
    if _STATE.get("group_ranks") is not None:
        return _STATE["group_rank"]
    return dist.get_rank() if is_distributed() else 0


def comm_size() -> int:
    if _STATE.get("group_ranks") is not None:
        return len(_STATE["group_ranks"])
    return dist.get_world_size() if is_distributed() else 1


def to_global_rank(local: int) -> int:
    """Translate a topology-local rank to the global torch.distributed
    rank (identity outside split-grid mode)."""
    gr = _STATE.get("group_ranks")
    return gr[local] if gr is not None else local


def grid_dims() -> Tuple[int, int, int, int]:
    return _STATE["grid"]


def grid_coords() -> Tuple[int, int, int, int]:
    return _STATE["coords"]


def default_grid(world: int) -> Tuple[int, int, int, int]:
    """Partition T only (ranks on one node share xGMI; halos in one dim keep
    each exchange on a single point-to-point link pair)."""
    return (1, 1, 1, world)


def init_comms(grid: Optional[Tuple[int, int, int, int]] = None,
               backend: Optional[str] = None) -> Tuple[int, int]:
    """Initialize torch.distributed (if launched under torchrun) + topology.

    Returns (rank, world_size). Safe to call when single-process (no env).
    Analogue of initCommsGridQuda (quda.h:981).
    """
    if not dist.is_initialized() and os.environ.get("WORLD_SIZE"):
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            local = int(os.environ.get("LOCAL_RANK", 0))
            torch.cuda.set_device(local)
        dist.init_process_group(backend=backend)
    world = comm_size()
    g = tuple(grid) if grid is not None else default_grid(world)
    if g[0] * g[1] * g[2] * g[3] != world:
        raise ValueError(f"grid {g} != world size {world}")
    r = comm_rank()
    gx, gy, gz, gt = g
    coords = (r % gx, (r // gx) % gy, (r // (gx * gy)) % gz, r // (gx * gy * gz))
    _STATE.update(grid=g, coords=coords, initialized=True)
    return r, world


def rank_of_coords(c: Tuple[int, int, int, int]) -> int:
    gx, gy, gz, gt = _STATE["grid"]
    return ((c[3] % gt * gz + c[2] % gz) * gy + c[1] % gy) * gx + c[0] % gx


def neighbor_rank(dim: int, displacement: int) -> int:
    """GLOBAL rank of the grid neighbor (p2p ops take global ranks even
    inside a split-grid sub-communicator)."""
    c = list(_STATE["coords"])
    c[dim] += displacement
    return to_global_rank(rank_of_coords(tuple(c)))


def is_partitioned(dim: int) -> bool:
    return _STATE["grid"][dim] > 1 or (_STATE["forced_mask"] >> dim) & 1


def comm_mask() -> int:
    """Bitmask of partitioned dims (halo-exchange dims). Includes forced
    self-wraparound dims (the reference's --partition test mode,
    tests/CMakeLists.txt partition matrix)."""
    m = _STATE["forced_mask"]
    for d in range(4):
        if _STATE["grid"][d] > 1:
            m |= 1 << d
    return m


def set_forced_partition(mask: int) -> None:
    """Force halo-exchange code paths for dims in `mask` even at grid
    extent 1 (self-wraparound; results must be identical)."""
    _STATE["forced_mask"] = int(mask)


def parity_offset_of_rank(local_dims) -> int:
    """Global parity offset for this rank's sub-lattice."""
    c = _STATE["coords"]
    return sum(c[i] * local_dims[i] for i in range(4)) % 2


# -- collectives ------------------------------------------------------------

def allreduce_sum(x):
    """Global sum of a python scalar (float or complex-as-2-floats caller
    side). The latency-critical per-iteration collective."""
    if not is_distributed() or _solo():
        return x
    dev = "cuda" if dist.get_backend() == "nccl" else "cpu"
    t = torch.tensor([x], dtype=torch.float64, device=dev)
    dist.all_reduce(t, group=_STATE.get("group"))
    return t.item()


def allreduce_tensor(t: torch.Tensor) -> torch.Tensor:
    if is_distributed() and not _solo():
        dist.all_reduce(t, group=_STATE.get("group"))
    return t


def barrier():
    if is_distributed() and not _solo():
        dist.barrier(group=_STATE.get("group"))


# -- communicator stack (ref: lib/communicator_stack.cpp push_communicator:
# split-grid repartitions ranks into independent sub-grids) ----------------

from contextlib import contextmanager


@contextmanager
def solo_mode():
    """Push a single-rank communicator: collectives and halo exchange
    become local no-ops while each rank works on an independent problem
    (the sub-grid-size-1 split-grid case; general sub-grids are a tracked
    gap). Restores the outer topology on exit."""
    saved = dict(_STATE)
    saved_solo = _STATE.get("solo", False)
    _STATE.update(grid=(1, 1, 1, 1), coords=(0, 0, 0, 0), forced_mask=0,
                  solo=True)
    try:
        yield
    finally:
        _STATE.clear()
        _STATE.update(saved)
        _STATE["solo"] = saved_solo


def _solo() -> bool:
    return bool(_STATE.get("solo", False))


# -- general split-grid sub-communicators ----------------------------------

_GROUP_CACHE = {}


@contextmanager
def split_grid_mode(splits):
    """Partition the process grid into prod(splits) independent sub-grids
    (ref: lib/communicator_stack.cpp push_communicator with a non-trivial
    split key). Each sub-grid has process grid g_i/splits_i and spans the
    WHOLE lattice (fields must be redistributed — see
    split_grid.split_grid_solve). Yields (subgrid_index, n_subgrids).
    Topology queries, collectives and halo exchange all act within the
    sub-grid while inside the context."""
    g = _STATE["grid"]
    c = _STATE["coords"]
    for i in range(4):
        assert g[i] % splits[i] == 0, (g, splits)
    sub = tuple(g[i] // splits[i] for i in range(4))
    n_sub = splits[0] * splits[1] * splits[2] * splits[3]
    block = tuple(c[i] // sub[i] for i in range(4))
    newc = tuple(c[i] % sub[i] for i in range(4))
    sub_idx = ((block[3] * splits[2] + block[2]) * splits[1]
               + block[1]) * splits[0] + block[0]
    key = (g, tuple(splits))
    if n_sub == 1 or not is_distributed():
        groups = {0: (None, list(range(comm_size())))}
    elif key in _GROUP_CACHE:
        groups = _GROUP_CACHE[key]
    else:
        # build ALL sub-groups in the same (lex) order on every rank
        groups = {}
        idx = 0
        for bt in range(splits[3]):
            for bz in range(splits[2]):
                for by in range(splits[1]):
                    for bx in range(splits[0]):
                        b = (bx, by, bz, bt)
                        ranks = []
                        for t in range(sub[3]):
                            for z in range(sub[2]):
                                for y in range(sub[1]):
                                    for x in range(sub[0]):
                                        cc = (b[0] * sub[0] + x,
                                              b[1] * sub[1] + y,
                                              b[2] * sub[2] + z,
                                              b[3] * sub[3] + t)
                                        gx, gy, gz, gt = g
                                        ranks.append(((cc[3] * gz + cc[2])
                                                      * gy + cc[1]) * gx
                                                     + cc[0])
                        grp = dist.new_group(ranks)
                        groups[idx] = (grp, ranks)
                        idx += 1
        _GROUP_CACHE[key] = groups
    grp, ranks = groups[sub_idx if n_sub > 1 else 0]
    saved = dict(_STATE)
    me = dist.get_rank() if is_distributed() else 0
    _STATE.update(grid=sub, coords=newc, forced_mask=0, group=grp,
                  group_ranks=ranks, group_rank=ranks.index(me))
    try:
        yield sub_idx, n_sub
    finally:
        _STATE.clear()
        _STATE.update(saved)
