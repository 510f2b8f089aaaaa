"""Split-grid multi-source solves (ref: include/split_grid.h
split_field/join_field + callMultiSrcQuda interface_quda.cpp:3064 — the
pure data-parallel repartition of the process grid over right-hand
sides).

Implemented case: sub-grid size 1 (each rank holds the full lattice and
solves its share of the sources with collectives switched off via
comms.solo_mode). General sub-grid repartitioning of a distributed
lattice is a tracked gap."""

from __future__ import annotations

from typing import Callable, List

import torch
import torch.distributed as dist

from ..fields.spinor import SpinorField
from . import comms


def multi_src_solve(sources: List[SpinorField],
                    solve_one: Callable[[SpinorField, SpinorField], object]
                    ) -> List[SpinorField]:
    """Solve len(sources) independent systems, distributing them
    round-robin over the ranks; every rank returns ALL solutions.
    solve_one(x, b) fills x for one source (run inside solo_mode, so any
    solver/operator stack works unchanged)."""
    rank, world = comms.comm_rank(), comms.comm_size()
    n = len(sources)
    xs: List[SpinorField] = [s.clone_empty() for s in sources]
    with comms.solo_mode():
        for i in range(rank, n, max(world, 1)):
            solve_one(xs[i], sources[i])
    if world > 1:
        for i in range(n):
            owner = i % world
            t = xs[i].data
            dist.broadcast(t, src=owner)
            if xs[i].norm is not None:
                dist.broadcast(xs[i].norm, src=owner)
    return xs
