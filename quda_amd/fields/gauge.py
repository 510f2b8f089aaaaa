"""GaugeField (ref: lib/gauge_field.cpp, include/gauge_field_order.h) —
MI355X-first "stencil" layout.

Instead of the reference's per-direction FloatN arrays (which make the
backward-hop link loads neighbor-indexed), each checkerboard site stores ALL
8 links its dslash stencil needs, contiguously:

    site x, parity p:  [ U_0(x) .. U_3(x),  U_0(x-0) .. U_3(x-3) ]

i.e. the backward links are a pre-shifted copy. Every gauge load in the
kernel is then site-local and 16-byte vectorized for every precision x
reconstruct combination; only spinor loads remain neighbor-indexed. Costs
2x gauge memory — 288 GB HBM3E per GPU makes that free (SURVEY.md 2.11
rebuild note), and it removes 8 neighbor-index computations + all strided
sub-16B gauge loads per site.

Tensor shape: [2 (parity), 8*L/w, V_cb, w] with L = reals/link
(18 = full 3x3, 12 = rows 0,1 with row2 = conj(row0 x row1) reconstructed
in-kernel — ref gauge_field_order.h:2369), w = 16 bytes / itemsize.
Component order within a site: link slot q in 0..7 (0-3 fwd mu, 4-7 bwd mu),
comp = q*L + (row*3+col)*2 + reim.
"""

from __future__ import annotations

from typing import Optional

import torch

from .geometry import LatticeGeometry
from .layout import DTYPE_OF, WIDTH_OF

RECON_COMPS = {"none": 18, "twelve": 12, "eight": 8}


def pack_recon8(u: torch.Tensor) -> torch.Tensor:
    """[..., 3, 3] complex SU(3) -> [..., 8] real (the arXiv:0911.3191
    codec as the reference packs it, gauge_field_order.h Reconstruct<8>:
    phases of b1 and -c1 (units of pi) + b2, b3, a1 re/im; the matrix is
    compressed in the row-permuted form {{b},{a},{-c}} to avoid the
    unit-gauge singularity)."""
    import math
    sh = u.shape[:-2]
    out = torch.empty((*sh, 8), dtype=torch.float64)
    a1, b = u[..., 0, 0], u[..., 1, :]
    c1 = u[..., 2, 0]
    out[..., 0] = torch.atan2(b[..., 0].imag, b[..., 0].real) / math.pi
    out[..., 1] = torch.atan2(-c1.imag, -c1.real) / math.pi
    out[..., 2] = b[..., 1].real
    out[..., 3] = b[..., 1].imag
    out[..., 4] = b[..., 2].real
    out[..., 5] = b[..., 2].imag
    out[..., 6] = a1.real
    out[..., 7] = a1.imag
    return out.to(u.real.dtype)


def unpack_recon8(p: torch.Tensor, dtype=torch.complex128) -> torch.Tensor:
    """[..., 8] real -> [..., 3, 3] complex (mirrors the kernel decode in
    csrc/common.h GaugeAcc::load_base<RECON=8>; u0 = 1)."""
    import math
    rd = torch.float64 if dtype == torch.complex128 else torch.float32
    p = p.to(rd)
    M = torch.empty((*p.shape[:-1], 3, 3), dtype=dtype)
    b2 = torch.complex(p[..., 2], p[..., 3])
    b3 = torch.complex(p[..., 4], p[..., 5])
    a1 = torch.complex(p[..., 6], p[..., 7])
    # permuted M rows: row0 = b, row1 = a, row2 = -c
    row_sum = (b2.abs() ** 2 + b3.abs() ** 2)
    b1 = torch.polar((1.0 - row_sum).clamp_min(0).sqrt(),
                     p[..., 0] * math.pi)
    col_sum = b1.abs() ** 2 + a1.abs() ** 2
    mc1 = torch.polar((1.0 - col_sum).clamp_min(0).sqrt(),
                      p[..., 1] * math.pi)
    r_inv2 = 1.0 / row_sum
    A = b1.conj() * a1
    a2 = -(mc1.conj() * b3.conj() + A * b2) * r_inv2
    a3 = (mc1.conj() * b2.conj() - A * b3) * r_inv2
    B = b1.conj() * mc1
    mc2 = (a1.conj() * b3.conj() - B * b2) * r_inv2
    mc3 = -(a1.conj() * b2.conj() + B * b3) * r_inv2
    M[..., 0, 0], M[..., 0, 1], M[..., 0, 2] = a1, a2, a3
    M[..., 1, 0], M[..., 1, 1], M[..., 1, 2] = b1, b2, b3
    M[..., 2, 0], M[..., 2, 1], M[..., 2, 2] = -mc1, -mc2, -mc3
    return M


class GaugeField:
    def __init__(self, geo: LatticeGeometry, precision: str = "double",
                 device="cpu", reconstruct: str = "none",
                 data: Optional[torch.Tensor] = None, shift: int = 1):
        self.geo = geo
        self.precision = precision
        self.reconstruct = reconstruct
        self.shift = shift  # hop distance of the stencil (3 = long links)
        L = RECON_COMPS[reconstruct]
        w = WIDTH_OF[precision]
        assert (8 * L) % w == 0
        shape = (2, (8 * L) // w, geo.volume_cb, w)
        if data is not None:
            assert tuple(data.shape) == shape, (data.shape, shape)
            self.data = data
        else:
            self.data = torch.zeros(shape, dtype=DTYPE_OF[precision], device=device)

    @property
    def device(self):
        return self.data.device

    @property
    def L(self) -> int:
        return RECON_COMPS[self.reconstruct]

    def _nbr_shift(self, p: int, mu: int):
        """cb index of x - shift*mu (local periodic wrap)."""
        geo = self.geo
        if self.shift == 1:
            return geo.neighbor_cb(p, mu, -1)
        c = geo.coords_of_cb(p).to(torch.int64).clone()
        d = geo.dims[mu]
        c[:, mu] = (c[:, mu] - self.shift) % d
        X, Y, Z, _ = geo.dims
        lex = ((c[:, 3] * Z + c[:, 2]) * Y + c[:, 1]) * X + c[:, 0]
        return geo.cb_of_lex[lex]

    def to(self, device) -> "GaugeField":
        return GaugeField(self.geo, self.precision, device, self.reconstruct,
                          data=self.data.to(device), shift=self.shift)

    # ------------------------------------------------------------------
    def to_complex(self, dtype=torch.complex128) -> torch.Tensor:
        """-> [4, 2, V_cb, 3, 3] complex U_mu(x) (from the fwd slots)."""
        L = self.L
        V = self.geo.volume_cb
        d = self.data.to(torch.float64 if dtype == torch.complex128 else torch.float32)
        flat = d.movedim(1, 2).reshape(2, V, 8 * L)  # [p, V, comps]
        out = torch.empty((4, 2, V, 3, 3), dtype=dtype, device=d.device)
        for mu in range(4):
            comps = flat[:, :, mu * L:(mu + 1) * L]
            if self.reconstruct == "eight":
                out[mu] = unpack_recon8(comps, dtype).to(d.device)
                continue
            rows = torch.view_as_complex(
                comps.reshape(2, V, L // 2, 2).contiguous())
            if self.reconstruct == "none":
                out[mu] = rows.reshape(2, V, 3, 3)
            else:
                r01 = rows.reshape(2, V, 2, 3)
                r2 = torch.cross(r01[..., 0, :], r01[..., 1, :], dim=-1).conj()
                out[mu] = torch.cat([r01, r2.unsqueeze(-2)], dim=-2)
        return out

    def _encode(self, links: torch.Tensor) -> torch.Tensor:
        """[..., 3, 3] complex -> [..., L] real per the reconstruct codec."""
        if self.reconstruct == "eight":
            return pack_recon8(links).to(links.device)
        nrows = self.L // 6
        return torch.view_as_real(links[..., 0:nrows, :].contiguous()).reshape(
            *links.shape[:-2], self.L)

    def from_complex(self, u: torch.Tensor) -> "GaugeField":
        """u: [4, 2, V_cb, 3, 3] complex -> fill fwd + shifted bwd slots.

        When the process grid partitions a dim (comms.comm_mask), the bwd
        slots of the x_mu=0 face are U_mu links owned by the -mu neighbor:
        exchanged here once at load time (analogue of the reference's gauge
        exchangeGhost, lib/gauge_field.cpp:453 — our stencil layout folds
        the gauge ghost INTO the field, so the dslash needs no gauge ghost
        at apply time)."""
        L = self.L
        V = self.geo.volume_cb
        assert u.shape == (4, 2, V, 3, 3)
        if self.reconstruct != "none":
            # recon-12 rebuilds row 2 as conj(row0 x row1): only valid for
            # unitary links. Fat/smeared links MUST use reconstruct="none"
            # — catch the misuse at load time on a sample STRIDED across
            # the whole volume (a t=0-only sample misses boundary-phased
            # timeslices and partially-smeared regions).
            idx = torch.arange(0, V, max(1, V // 64), device=u.device)
            s = u[:, :, idx]
            r2 = torch.cross(s[..., 0, :], s[..., 1, :], dim=-1).conj()
            err = (r2 - s[..., 2, :]).abs().max().item()
            if err > 1e-3:
                raise ValueError(
                    "reconstruct='twelve' on non-unitary links (row-2 "
                    f"reconstruction error {err:.2e}); use "
                    "reconstruct='none' for fat/smeared fields")
        dev = u.device
        flat = torch.empty((2, V, 8 * L),
                           dtype=torch.float64 if u.dtype == torch.complex128 else torch.float32,
                           device=dev)
        for p in (0, 1):
            for mu in range(4):
                flat[p, :, mu * L:(mu + 1) * L] = self._encode(u[mu, p])
                bwd_idx = self._nbr_shift(p, mu).to(dev)
                bwd = u[mu, (1 - p) if self.shift % 2 else p][bwd_idx]
                flat[p, :, (4 + mu) * L:(5 + mu) * L] = self._encode(bwd)
        # fix up bwd slots on partitioned-dim boundary faces: the x_mu = l
        # (l < shift) sites need U_mu(x - shift*mu) owned by the -mu
        # neighbor at its coord X-shift+l (one exchange per layer; shift=3
        # long links exchange three layers).
        from ..parallel import comms
        from ..parallel.halo import active_dims, exchange_tensors
        mask = comms.comm_mask()
        for mu in active_dims(mask):
            geo = self.geo
            hi = geo.dims[mu] - 1
            fcb = geo.face_volume_cb(mu)
            for l in range(self.shift):
                c_src = hi - (self.shift - 1) + l
                send = torch.empty((2, fcb, 3, 3), dtype=u.dtype,
                                   device=dev)
                for q in (0, 1):
                    send[q] = u[mu, q][geo.face_index_cb(q, mu, c_src)
                                       .to(dev)]
                recv = torch.empty_like(send)
                exchange_tensors({(mu, 1): send}, {(mu, 0): recv})
                for p in (0, 1):
                    fidx = geo.face_index_cb(p, mu, l).to(dev)
                    q = (1 - p) if self.shift % 2 else p
                    flat[p, fidx, (4 + mu) * L:(5 + mu) * L] = \
                        self._encode(recv[q])
        w = WIDTH_OF[self.precision]
        native = flat.reshape(2, V, (8 * L) // w, w).movedim(2, 1).contiguous()
        self.data.copy_(native.to(self.data.dtype))
        self._bwd_ghost_cache = {}
        return self

    def bwd_ghost(self, mu: int, parity: int, coord: int = 0) -> torch.Tensor:
        """[Fcb, 3, 3] complex U_mu(x-shift*mu) for the x_mu=coord face
        sites of `parity`, in ghost order — decoded from the stored bwd
        slots (used by the CPU oracle on partitioned dims; exact same
        values the HIP kernel reads)."""
        cache = self.__dict__.setdefault("_bwd_ghost_cache", {})
        key = (mu, parity, coord)
        if key not in cache:
            L = self.L
            geo = self.geo
            V = geo.volume_cb
            fidx0 = geo.face_index_cb(parity, mu, coord)
            d = self.data[parity].to(torch.float64)          # [NCH, V, w]
            flat = d.movedim(0, 1).reshape(V, 8 * L)
            rows = torch.view_as_complex(
                flat[fidx0.to(d.device), (4 + mu) * L:(5 + mu) * L]
                .reshape(-1, L // 2, 2).contiguous())
            nrows = L // 6
            rows = rows.reshape(-1, nrows, 3)
            if self.reconstruct == "none":
                out = rows
            else:
                r2 = torch.cross(rows[:, 0, :], rows[:, 1, :], dim=-1).conj()
                out = torch.cat([rows, r2.unsqueeze(1)], dim=1)
            cache[key] = out
        return cache[key]

    # -- fills ----------------------------------------------------------
    def unit_(self) -> "GaugeField":
        V = self.geo.volume_cb
        u = torch.eye(3, dtype=torch.complex128, device=self.device)
        u = u.expand(4, 2, V, 3, 3).contiguous()
        return self.from_complex(u)

    def random_su3_(self, seed: Optional[int] = None, sigma: float = 1.0) -> "GaugeField":
        """Random SU(3) links (same role as the reference tests'
        constructRandomGaugeField, tests/utils/host_utils.cpp:1022)."""
        g = torch.Generator(device="cpu")
        if seed is not None:
            g.manual_seed(seed)
        V = self.geo.volume_cb
        m = torch.randn((4, 2, V, 3, 3, 2), generator=g, dtype=torch.float64) * sigma
        u = project_su3(torch.view_as_complex(m).to(self.device))
        return self.from_complex(u)

    def __repr__(self):
        return (f"GaugeField({self.geo.dims}, {self.precision}, "
                f"recon={self.reconstruct}, device={self.device})")


def project_su3_polar(u: torch.Tensor) -> torch.Tensor:
    """GAUGE-COVARIANT SU(3) projection: U(3) polar factor (SVD) with the
    det phase divided out (the reference's smearing-grade projection,
    su3_project.cuh; Gram-Schmidt below is NOT covariant — fine for
    random-field generation, wrong inside smearing)."""
    U_, _, Vh = torch.linalg.svd(u)
    W = U_ @ Vh
    ph = torch.linalg.det(W) ** (1.0 / 3.0)
    return W / ph[..., None, None]


def project_su3(u: torch.Tensor) -> torch.Tensor:
    """Project [..., 3, 3] complex onto SU(3): Gram-Schmidt rows + det fix."""
    r0 = u[..., 0, :]
    r0 = r0 / r0.norm(dim=-1, keepdim=True)
    r1 = u[..., 1, :]
    r1 = r1 - (r0.conj() * r1).sum(-1, keepdim=True) * r0
    r1 = r1 / r1.norm(dim=-1, keepdim=True)
    r2 = torch.cross(r0, r1, dim=-1).conj()
    return torch.stack([r0, r1, r2], dim=-2)
