"""Foreign (application) field orders (ref: gauge_field_order.h /
color_spinor_field_order.h QDP/MILC/CPS order classes + the host-side
copy kernels lib/copy_gauge*.cu — converters between our chunked-SoA
device layout's oracle view and the interop layouts client apps hand in).

Conventions implemented:
- QDP:  per-direction arrays; sites in even-odd blocks, cb index = our cb
        index; color matrices row-major. gauge: list of 4 tensors
        [2, Vcb, 3, 3]; spinor: [2, Vcb, 4, 3].
- MILC: one array, sites in even-odd blocks; gauge [V, 4, 3, 3] (direction
        slowest-varying per site), spinor [V, 4, 3] (or [V, 3] staggered).
- CPS:  lexicographic site order, direction-major gauge [4, V, 3, 3],
        spinor [V, 4, 3] lex.
"""

from __future__ import annotations

from typing import List

import torch

from .geometry import LatticeGeometry, checkerboard_join, checkerboard_split


# -- QDP --------------------------------------------------------------------

def gauge_to_qdp(u: torch.Tensor, geo: LatticeGeometry) -> List[torch.Tensor]:
    """[4,2,V,3,3] -> list of 4 [2,V,3,3] (shared-storage views)."""
    return [u[mu] for mu in range(4)]


def gauge_from_qdp(qdp: List[torch.Tensor], geo: LatticeGeometry) -> torch.Tensor:
    return torch.stack(list(qdp), dim=0)


def spinor_to_qdp(c: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """oracle [2,V,4,3] IS the QDP order."""
    return c


def spinor_from_qdp(c: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    return c


# -- MILC -------------------------------------------------------------------

def gauge_to_milc(u: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """[4,2,V,3,3] -> [2V, 4, 3, 3] even-block-then-odd site-major."""
    V = geo.volume_cb
    out = torch.empty((2 * V, 4, 3, 3), dtype=u.dtype, device=u.device)
    for p in (0, 1):
        out[p * V:(p + 1) * V] = u[:, p].permute(1, 0, 2, 3)
    return out


def gauge_from_milc(m: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    V = geo.volume_cb
    u = torch.empty((4, 2, V, 3, 3), dtype=m.dtype, device=m.device)
    for p in (0, 1):
        u[:, p] = m[p * V:(p + 1) * V].permute(1, 0, 2, 3)
    return u


def spinor_to_milc(c: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """[2,V,...] -> [2V, ...] even block then odd."""
    return c.reshape(2 * geo.volume_cb, *c.shape[2:])


def spinor_from_milc(m: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    return m.reshape(2, geo.volume_cb, *m.shape[1:])


# -- CPS (lexicographic) ----------------------------------------------------

def gauge_to_cps(u: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """[4,2,V,3,3] -> [4, Vlex, 3, 3] lex site order."""
    lo = geo.lex_of_cb.to(u.device)
    out = torch.empty((4, geo.volume, 3, 3), dtype=u.dtype, device=u.device)
    out[:, lo[0]] = u[:, 0]
    out[:, lo[1]] = u[:, 1]
    return out


def gauge_from_cps(c: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    lo = geo.lex_of_cb.to(c.device)
    return torch.stack([c[:, lo[0]], c[:, lo[1]]], dim=1)


def spinor_to_cps(c: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    return checkerboard_join(c, geo)


def spinor_from_cps(lex: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    return checkerboard_split(lex, geo)
