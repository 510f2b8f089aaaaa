"""Lattice geometry: dimensions, even-odd checkerboarding, site indexing.

Conventions (chosen to match the reference semantics, not its code;
ref: /root/reference/include/index_helper.cuh:575 and lattice_field.h):

- 4 dimensions ordered (x, y, z, t); x runs fastest in memory.
- Lexicographic index  lex = ((t*Z + z)*Y + y)*X + x.
- Site parity          p   = (x + y + z + t) & 1   (plus a global offset for
  multi-rank grids so parity is globally consistent).
- Checkerboard index   cb  = lex >> 1 within a parity; full-lattice storage is
  [parity][cb].

All index math here is vectorized torch/numpy for the CPU oracle path; the
HIP kernels recompute coordinates arithmetically in-register (no tables).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from functools import cached_property
from typing import Sequence, Tuple

import torch


@dataclass(frozen=True)
class LatticeGeometry:
    """Local (per-rank) lattice geometry.

    Parameters
    ----------
    dims : (X, Y, Z, T) local lattice extents. Each must be even so that the
        checkerboard split is exact (the reference has the same constraint).
    parity_offset : sum of this rank's global coordinate offsets mod 2, so
        that site parity is globally consistent across a partitioned grid.
    """

    dims: Tuple[int, int, int, int]
    parity_offset: int = 0

    def __post_init__(self):
        if len(self.dims) != 4:
            raise ValueError(f"need 4 dims, got {self.dims}")
        for d in self.dims:
            if d % 2 != 0 or d < 2:
                raise ValueError(f"all local dims must be even and >=2: {self.dims}")
        object.__setattr__(self, "dims", tuple(int(d) for d in self.dims))

    # -- basic volumes ------------------------------------------------------
    @property
    def X(self) -> int:
        return self.dims[0]

    @property
    def Y(self) -> int:
        return self.dims[1]

    @property
    def Z(self) -> int:
        return self.dims[2]

    @property
    def T(self) -> int:
        return self.dims[3]

    @property
    def volume(self) -> int:
        x, y, z, t = self.dims
        return x * y * z * t

    @property
    def volume_cb(self) -> int:
        return self.volume // 2

    def face_volume(self, dim: int) -> int:
        """Number of sites on one face orthogonal to `dim` (full, not cb)."""
        v = 1
        for i, d in enumerate(self.dims):
            if i != dim:
                v *= d
        return v

    def face_volume_cb(self, dim: int) -> int:
        return self.face_volume(dim) // 2

    # -- coordinate <-> index maps (vectorized, torch) ----------------------
    @cached_property
    def coords(self) -> torch.Tensor:
        """[volume, 4] int32 tensor of (x,y,z,t) for each lex index."""
        X, Y, Z, T = self.dims
        lex = torch.arange(self.volume, dtype=torch.int64)
        x = lex % X
        y = (lex // X) % Y
        z = (lex // (X * Y)) % Z
        t = lex // (X * Y * Z)
        return torch.stack([x, y, z, t], dim=1).to(torch.int32)

    @cached_property
    def parity(self) -> torch.Tensor:
        """[volume] int8: parity of each lex site (including global offset)."""
        c = self.coords.to(torch.int64)
        return ((c.sum(dim=1) + self.parity_offset) % 2).to(torch.int8)

    @cached_property
    def lex_of_cb(self) -> torch.Tensor:
        """[2, volume_cb] int64: lex index of each (parity, cb) site."""
        p = self.parity
        lex = torch.arange(self.volume, dtype=torch.int64)
        even = lex[p == 0]
        odd = lex[p == 1]
        assert even.numel() == odd.numel() == self.volume_cb
        return torch.stack([even, odd], dim=0)

    @cached_property
    def cb_of_lex(self) -> torch.Tensor:
        """[volume] int64: cb index of each lex site (within its parity)."""
        out = torch.empty(self.volume, dtype=torch.int64)
        lo = self.lex_of_cb
        out[lo[0]] = torch.arange(self.volume_cb, dtype=torch.int64)
        out[lo[1]] = torch.arange(self.volume_cb, dtype=torch.int64)
        return out

    def coords_of_cb(self, parity: int) -> torch.Tensor:
        """[volume_cb, 4] coords of each cb site at given parity."""
        return self.coords[self.lex_of_cb[parity]]

    # -- neighbor tables (for CPU oracle / pack kernels) --------------------
    def neighbor_lex(self, dim: int, displacement: int) -> torch.Tensor:
        """[volume] lex index of the site displaced by +/-1 in `dim`
        (periodic wrap within the local lattice)."""
        c = self.coords.to(torch.int64).clone()
        d = self.dims[dim]
        c[:, dim] = (c[:, dim] + displacement) % d
        X, Y, Z, _ = self.dims
        return ((c[:, 3] * Z + c[:, 2]) * Y + c[:, 1]) * X + c[:, 0]

    def neighbor_cb(self, parity: int, dim: int, displacement: int) -> torch.Tensor:
        """[volume_cb] int64: for each cb site at `parity`, the cb index (in the
        opposite-parity array) of the site displaced by +-1 in `dim`, with
        periodic wrap inside the local lattice. Cached."""
        key = (parity, dim, displacement)
        cache = self.__dict__.setdefault("_nbr_cache", {})
        if key not in cache:
            nbr_lex = self.neighbor_lex(dim, displacement)  # [volume]
            my_lex = self.lex_of_cb[parity]
            cache[key] = self.cb_of_lex[nbr_lex[my_lex]].contiguous()
        return cache[key]

    def face_index_cb(self, parity: int, dim: int, edge: int) -> torch.Tensor:
        """[Fcb] int64 cb indices of the parity-`parity` sites on the face
        coords[dim]==edge, ordered by ghost index (transverse coords
        flattened lowest-dim-fastest, >>1) — the order csrc/halo.h
        ghost_idx/face_coords use, shared by pack and unpack."""
        key = ("face", parity, dim, edge)
        cache = self.__dict__.setdefault("_nbr_cache", {})
        if key not in cache:
            c = self.coords_of_cb(parity).to(torch.int64)
            sel = (c[:, dim] == edge).nonzero(as_tuple=True)[0]
            rd = [i for i in range(4) if i != dim]
            cc = c[sel]
            flat3 = ((cc[:, rd[2]] * self.dims[rd[1]] + cc[:, rd[1]])
                     * self.dims[rd[0]] + cc[:, rd[0]])
            order = torch.argsort(flat3)
            cache[key] = sel[order].contiguous()
        return cache[key]

    def boundary_mask_cb(self, parity: int, dim: int, displacement: int) -> torch.Tensor:
        """[volume_cb] bool: True where the displaced neighbor wraps around the
        local lattice boundary in `dim` (i.e. lives in the halo when `dim` is
        partitioned)."""
        c = self.coords_of_cb(parity)[:, dim].to(torch.int64)
        d = self.dims[dim]
        if displacement > 0:
            return c == d - 1
        return c == 0

    def __repr__(self):
        return f"LatticeGeometry(dims={self.dims}, parity_offset={self.parity_offset})"


def checkerboard_split(lex_field: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """[volume, ...] lex-ordered -> [2, volume_cb, ...] (parity, cb)."""
    lo = geo.lex_of_cb.to(lex_field.device)
    return torch.stack([lex_field[lo[0]], lex_field[lo[1]]], dim=0)


def checkerboard_join(cb_field: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """[2, volume_cb, ...] -> [volume, ...] lex-ordered."""
    out = torch.empty((geo.volume,) + tuple(cb_field.shape[2:]),
                      dtype=cb_field.dtype, device=cb_field.device)
    lo = geo.lex_of_cb.to(cb_field.device)
    out[lo[0]] = cb_field[0]
    out[lo[1]] = cb_field[1]
    return out
