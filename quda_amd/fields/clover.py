"""CloverField (ref: lib/clover_field.cpp, include/clover_field_order.h).

The clover site matrix A is hermitian and block-diagonal in chirality
(DeGrand-Rossi basis): two 6x6 hermitian blocks per site. Packed storage
(same information content as the reference's 72-real layout,
clover_field_order.h:587):

  per site, per chirality block b in {0,1}:
      6 real diagonal entries, then 15 lower-triangular complex entries
      (row-major, i > j)  -> 36 reals; both blocks -> 72 reals/site.

Chunked native tensor: [n_parity=2, 72/w, V_cb, w] (+ the inverse, same
shape) — the inverse is required by the even-odd preconditioned operator
(A_oo^-1) and is built here with batched 6x6 Cholesky-free torch inverse.
"""

from __future__ import annotations

from typing import Optional

import torch

from .geometry import LatticeGeometry
from .layout import DTYPE_OF, chunk_width, n_chunks

N_REALS = 72


def pack_clover(A: torch.Tensor) -> torch.Tensor:
    """[..., 12, 12] hermitian chirality-block-diagonal -> [..., 72] real."""
    lead = A.shape[:-2]
    out = torch.empty(*lead, N_REALS, dtype=A.real.dtype, device=A.device)
    tri = torch.tril_indices(6, 6, offset=-1)
    for b in range(2):
        blk = A[..., 6 * b:6 * b + 6, 6 * b:6 * b + 6]
        base = 36 * b
        out[..., base:base + 6] = torch.diagonal(blk, dim1=-2, dim2=-1).real
        off = blk[..., tri[0], tri[1]]  # [..., 15] complex
        out[..., base + 6:base + 36] = torch.view_as_real(off).reshape(*lead, 30)
    return out


def unpack_clover(p: torch.Tensor, dtype=torch.complex128) -> torch.Tensor:
    """[..., 72] real -> [..., 12, 12] complex hermitian block-diagonal."""
    lead = p.shape[:-1]
    rdt = torch.float64 if dtype == torch.complex128 else torch.float32
    p = p.to(rdt)
    A = torch.zeros(*lead, 12, 12, dtype=dtype, device=p.device)
    tri = torch.tril_indices(6, 6, offset=-1)
    for b in range(2):
        base = 36 * b
        blk = torch.zeros(*lead, 6, 6, dtype=dtype, device=p.device)
        diag = p[..., base:base + 6]
        off = torch.view_as_complex(
            p[..., base + 6:base + 36].reshape(*lead, 15, 2).contiguous())
        blk[..., tri[0], tri[1]] = off
        blk = blk + blk.conj().mT
        blk[..., range(6), range(6)] = diag.to(dtype)
        A[..., 6 * b:6 * b + 6, 6 * b:6 * b + 6] = blk
    return A


class CloverField:
    def __init__(self, geo: LatticeGeometry, precision: str = "double",
                 device="cpu"):
        self.geo = geo
        self.precision = precision
        w = chunk_width(N_REALS, precision)  # quarter: 16 !| 72 -> 8
        nch = n_chunks(N_REALS, precision)
        shape = (2, nch, geo.volume_cb, w)
        dt = DTYPE_OF[precision]
        self.data = torch.zeros(shape, dtype=dt, device=device)
        self.inv_data = torch.zeros(shape, dtype=dt, device=device)

    @property
    def device(self):
        return self.data.device

    def _to_native(self, packed: torch.Tensor) -> torch.Tensor:
        """[2, V, 72] real -> chunked [2, nch, V, w]."""
        w = chunk_width(N_REALS, self.precision)
        V = self.geo.volume_cb
        return (packed.reshape(2, V, -1, w).movedim(2, 1).contiguous()
                .to(DTYPE_OF[self.precision]))

    def _from_native(self, native: torch.Tensor) -> torch.Tensor:
        V = self.geo.volume_cb
        return native.movedim(1, 2).reshape(2, V, N_REALS)

    def from_matrices(self, A: torch.Tensor) -> "CloverField":
        """Build from [2, V, 12, 12] hermitian complex (and invert)."""
        V = self.geo.volume_cb
        assert A.shape == (2, V, 12, 12)
        self.data.copy_(self._to_native(pack_clover(A)))
        Ainv = torch.zeros_like(A)
        for b in range(2):
            blk = A[..., 6 * b:6 * b + 6, 6 * b:6 * b + 6]
            Ainv[..., 6 * b:6 * b + 6, 6 * b:6 * b + 6] = torch.linalg.inv(blk)
        self.inv_data.copy_(self._to_native(pack_clover(Ainv)))
        return self

    def to_complex(self, inverse: bool = False, dtype=torch.complex128):
        src = self.inv_data if inverse else self.data
        return unpack_clover(self._from_native(src), dtype)

    def __repr__(self):
        return f"CloverField({self.geo.dims}, {self.precision}, device={self.device})"
