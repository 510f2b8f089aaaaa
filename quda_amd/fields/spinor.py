"""ColorSpinorField equivalent (ref: lib/color_spinor_field.cpp).

Storage: torch tensor in the chunked SoA layout of layout.py,
shape [n_parity, n_chunk, V_cb, w]; HALF adds a per-site norm tensor
[n_parity, V_cb] (fp32).

The canonical *oracle* representation is `to_complex()`:
[n_parity, V_cb, 4, 3] complex — every CPU reference op (ops/reference.py)
and every GPU-vs-CPU numerics test works in that layout.
"""

from __future__ import annotations

from typing import Optional

import torch

from .geometry import LatticeGeometry
from .layout import (CHUNK_BYTES, DTYPE_OF, WIDTH_OF, chunked_to_complex,
                     complex_to_chunked, n_chunks)

FULL = 2  # both parities
PARITY = 1  # single-parity field (even-odd solves)


class SpinorField:
    """nspin=4 Wilson-type or nspin=1 staggered-type fermion field."""
    ncolor = 3

    def __init__(self, geo: LatticeGeometry, precision: str = "double",
                 device="cpu", n_parity: int = FULL,
                 data: Optional[torch.Tensor] = None,
                 norm: Optional[torch.Tensor] = None, nspin: int = 4,
                 ls: int = 1):
        self.geo = geo
        self.precision = precision
        self.n_parity = n_parity
        self.nspin = nspin
        self.ls = ls  # 5th (domain-wall) extent; site index = s*Vcb + x
        self.ncomp = self.nspin * self.ncolor * 2
        from .layout import chunk_width
        w = chunk_width(self.ncomp, precision)
        nch = n_chunks(self.ncomp, precision)
        shape = (n_parity, nch, ls * geo.volume_cb, w)
        if data is not None:
            assert tuple(data.shape) == shape, (data.shape, shape)
            self.data = data
        else:
            self.data = torch.zeros(shape, dtype=DTYPE_OF[precision], device=device)
        if precision in ("half", "quarter"):
            nshape = (n_parity, ls * geo.volume_cb)
            if norm is not None:
                assert tuple(norm.shape) == nshape
                self.norm = norm
            else:
                self.norm = torch.zeros(nshape, dtype=torch.float32, device=device)
        else:
            self.norm = None

    # ------------------------------------------------------------------
    @property
    def device(self):
        return self.data.device

    @property
    def volume_cb(self) -> int:
        return self.ls * self.geo.volume_cb

    @property
    def site_shape(self):
        return (self.nspin, self.ncolor) if self.nspin > 1 else (self.ncolor,)

    def clone_empty(self, precision: Optional[str] = None) -> "SpinorField":
        return SpinorField(self.geo, precision or self.precision,
                           self.device, self.n_parity, nspin=self.nspin,
                           ls=self.ls)

    def copy_(self, src: "SpinorField") -> "SpinorField":
        """Any-precision copy (ref: lib/copy_color_spinor_*.cu)."""
        if src.precision == self.precision:
            self.data.copy_(src.data)
            if self.norm is not None:
                self.norm.copy_(src.norm)
        else:
            self.from_complex(src.to_complex())
        return self

    # -- oracle layout conversions -------------------------------------
    def to_complex(self, dtype=torch.complex128) -> torch.Tensor:
        """-> [n_parity, V_cb, 4, 3] complex (denormalized for half)."""
        c = chunked_to_complex(self.data, dtype)  # [P, V, ncomp/2]
        if self.norm is not None:
            c = c * self.norm.to(c.real.dtype).unsqueeze(-1)
        return c.reshape(self.n_parity, self.volume_cb, *self.site_shape)

    def from_complex(self, c: torch.Tensor) -> "SpinorField":
        assert c.shape == (self.n_parity, self.volume_cb, *self.site_shape), \
            (c.shape, self.site_shape)
        flat = c.reshape(self.n_parity, self.volume_cb, self.ncomp // 2)
        if self.precision in ("half", "quarter"):
            mags = torch.view_as_real(flat).abs().amax(dim=(-1, -2))  # [P,V]
            self.norm.copy_(mags.to(torch.float32))
            scale = torch.where(mags > 0, 1.0 / mags, torch.zeros_like(mags))
            flat = flat * scale.to(flat.real.dtype).unsqueeze(-1)
        self.data.copy_(complex_to_chunked(flat, self.precision))
        return self

    # -- fills ----------------------------------------------------------
    def zero_(self) -> "SpinorField":
        self.data.zero_()
        if self.norm is not None:
            self.norm.zero_()
        return self

    def gaussian_(self, seed: Optional[int] = None) -> "SpinorField":
        g = torch.Generator(device="cpu")
        if seed is not None:
            g.manual_seed(seed)
        c = torch.randn((self.n_parity, self.volume_cb, *self.site_shape, 2),
                        generator=g, dtype=torch.float64)
        self.from_complex(torch.view_as_complex(c).to(self.device))
        return self

    # -- views -----------------------------------------------------------
    def to(self, device) -> "SpinorField":
        return SpinorField(self.geo, self.precision, device, self.n_parity,
                           data=self.data.to(device),
                           norm=None if self.norm is None else self.norm.to(device),
                           nspin=self.nspin, ls=self.ls)

    def parity_view(self, parity: int) -> "SpinorField":
        """Zero-copy single-parity view of a full field (ref Even()/Odd())."""
        assert self.n_parity == FULL
        return SpinorField(self.geo, self.precision, self.device, PARITY,
                           data=self.data[parity:parity + 1],
                           norm=None if self.norm is None else self.norm[parity:parity + 1],
                           nspin=self.nspin, ls=self.ls)

    def __repr__(self):
        return (f"SpinorField({self.geo.dims}, {self.precision}, "
                f"n_parity={self.n_parity}, device={self.device})")
