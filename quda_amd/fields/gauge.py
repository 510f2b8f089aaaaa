"""GaugeField equivalent (ref: lib/gauge_field.cpp).

Storage: [4 (mu), n_parity, n_chunk, V_cb, 2] in complex-pair chunks
(double2 = 16 B, float2 = 8 B, half2 = 4 B per lane — all coalesced).

reconstruct:
  "none"   : 18 reals/link (full 3x3 complex matrix)
  "twelve" : 12 reals/link (rows 0,1; row2 = conj(row0 x row1) in-kernel)
(ref: gauge_field_order.h:2369 reconstruct mappers; 13/9/8 variants are
future work)

Oracle layout: to_complex() -> [4, n_parity, V_cb, 3, 3] complex, U_mu(x).
"""

from __future__ import annotations

from typing import Optional

import torch

from .geometry import LatticeGeometry
from .layout import DTYPE_OF

RECON_COMPS = {"none": 18, "twelve": 12}


class GaugeField:
    def __init__(self, geo: LatticeGeometry, precision: str = "double",
                 device="cpu", reconstruct: str = "none",
                 data: Optional[torch.Tensor] = None):
        self.geo = geo
        self.precision = precision
        self.reconstruct = reconstruct
        ncomp = RECON_COMPS[reconstruct]
        shape = (4, 2, ncomp // 2, geo.volume_cb, 2)
        if data is not None:
            assert tuple(data.shape) == shape, (data.shape, shape)
            self.data = data
        else:
            self.data = torch.zeros(shape, dtype=DTYPE_OF[precision], device=device)

    @property
    def device(self):
        return self.data.device

    def to(self, device) -> "GaugeField":
        return GaugeField(self.geo, self.precision, device, self.reconstruct,
                          data=self.data.to(device))

    # ------------------------------------------------------------------
    def to_complex(self, dtype=torch.complex128) -> torch.Tensor:
        """-> [4, 2, V_cb, 3, 3] complex with row2 reconstructed if needed."""
        d = self.data.to(torch.float64 if dtype == torch.complex128 else torch.float32)
        nch = d.shape[2]
        V = self.geo.volume_cb
        c = torch.view_as_complex(d.movedim(2, 3).contiguous())  # [4,2,V,nch]
        if self.reconstruct == "none":
            return c.reshape(4, 2, V, 3, 3)
        rows01 = c.reshape(4, 2, V, 2, 3)
        row2 = torch.cross(rows01[..., 0, :], rows01[..., 1, :], dim=-1).conj()
        return torch.cat([rows01, row2.unsqueeze(-2)], dim=-2)

    def from_complex(self, u: torch.Tensor) -> "GaugeField":
        V = self.geo.volume_cb
        assert u.shape == (4, 2, V, 3, 3)
        if self.reconstruct == "twelve":
            u = u[..., 0:2, :]
        nch = RECON_COMPS[self.reconstruct] // 2
        flat = torch.view_as_real(u.reshape(4, 2, V, nch))  # [4,2,V,nch,2]
        self.data.copy_(flat.movedim(3, 2).contiguous().to(self.data.dtype))
        return self

    # -- fills ----------------------------------------------------------
    def unit_(self) -> "GaugeField":
        V = self.geo.volume_cb
        u = torch.eye(3, dtype=torch.complex128, device=self.device)
        u = u.expand(4, 2, V, 3, 3).contiguous()
        return self.from_complex(u)

    def random_su3_(self, seed: Optional[int] = None, sigma: float = 1.0) -> "GaugeField":
        """Random SU(3) links: Gaussian complex matrix -> Gram-Schmidt rows ->
        det-phase fix (same role as the reference tests'
        constructRandomGaugeField, tests/utils/host_utils.cpp:1022)."""
        g = torch.Generator(device="cpu")
        if seed is not None:
            g.manual_seed(seed)
        V = self.geo.volume_cb
        m = torch.randn((4, 2, V, 3, 3, 2), generator=g, dtype=torch.float64) * sigma
        # bias toward identity for small sigma (keeps links near unit gauge)
        u = torch.view_as_complex(m)
        u = project_su3(u)
        return self.from_complex(u.to(self.device))

    def __repr__(self):
        return (f"GaugeField({self.geo.dims}, {self.precision}, "
                f"recon={self.reconstruct}, device={self.device})")


def project_su3(u: torch.Tensor) -> torch.Tensor:
    """Project [..., 3, 3] complex onto SU(3): Gram-Schmidt rows + det fix."""
    r0 = u[..., 0, :]
    r0 = r0 / r0.norm(dim=-1, keepdim=True)
    r1 = u[..., 1, :]
    r1 = r1 - (r0.conj() * r1).sum(-1, keepdim=True) * r0
    r1 = r1 / r1.norm(dim=-1, keepdim=True)
    r2 = torch.cross(r0, r1, dim=-1).conj()
    out = torch.stack([r0, r1, r2], dim=-2)
    # det is now exactly +1 by construction (r2 = conj(r0 x r1))
    return out
