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
          "forced_mask": 0}


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def comm_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def comm_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


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
    c = list(_STATE["coords"])
    c[dim] += displacement
    return rank_of_coords(tuple(c))


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
    dist.all_reduce(t)
    return t.item()


def allreduce_tensor(t: torch.Tensor) -> torch.Tensor:
    if is_distributed() and not _solo():
        dist.all_reduce(t)
    return t


def barrier():
    if is_distributed() and not _solo():
        dist.barrier()


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
