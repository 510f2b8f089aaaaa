"""Device-native field layouts: 16-byte-chunked structure-of-arrays.

The reference's FloatN AoSoA orders (gauge_field_order.h:1516,
color_spinor_field_order.h:1191) exist to make consecutive GPU threads issue
contiguous vector loads. The MI355X-native equivalent: every field is stored
as [n_chunk, V_cb, w] where one chunk row holds w consecutive real components
of a site and w * itemsize == 16 bytes — so lane i of a wave64 loads a full
float4/double2/half8 at base + i*16B, the CDNA4 coalescing sweet spot
(cdna_hip_programming.md section 2).

Component flattening for a (spin, color) spinor:
    comp = (s * ncolor + c) * 2 + reim          (24 components for Ns=4,Nc=3)
for a color matrix (gauge link):
    comp = (row * 3 + col) * 2 + reim           (18, or 12 reconstructed)

HALF precision spinors additionally carry a per-site fp32 norm; stored values
are component/norm in [-1,1] ("block float", same idea as the reference's
fixed-point half with per-site norm, color_spinor_field_order.h:1426).
"""

from __future__ import annotations

import torch

CHUNK_BYTES = 16

DTYPE_OF = {
    "double": torch.float64,
    "single": torch.float32,
    "half": torch.float16,
    "quarter": torch.float8_e4m3fn,
}

WIDTH_OF = {  # reals per 16-byte chunk
    "double": 2,
    "single": 4,
    "half": 8,
    "quarter": 16,
}


def chunk_width(ncomp: int, prec: str) -> int:
    """Widest chunk (in reals) <= the precision's 16B width that divides
    ncomp (mirrors csrc/common.h chunk_w<>: 24 -> 16B chunks, 6 -> one
    complex per chunk)."""
    w = WIDTH_OF[prec]
    while ncomp % w:
        w //= 2
    return w


def n_chunks(ncomp: int, prec: str) -> int:
    return ncomp // chunk_width(ncomp, prec)


def complex_to_chunked(site_comp: torch.Tensor, prec: str) -> torch.Tensor:
    """[..., V, ncomp/2] complex  ->  [..., nchunk, V, w] real (native order).

    Leading dims (parity etc.) are preserved.
    """
    w = chunk_width(site_comp.shape[-1] * 2, prec)
    real = torch.view_as_real(site_comp)           # [..., V, ncomp/2, 2]
    flat = real.reshape(*site_comp.shape[:-1], -1)  # [..., V, ncomp]
    ncomp = flat.shape[-1]
    nch = ncomp // w
    # [., V, nch, w] -> [., nch, V, w]
    out = flat.reshape(*flat.shape[:-1], nch, w).movedim(-2, -3).contiguous()
    return out.to(DTYPE_OF[prec])


def chunked_to_complex(native: torch.Tensor, out_dtype=torch.complex128) -> torch.Tensor:
    """[..., nchunk, V, w] real -> [..., V, ncomp/2] complex."""
    nch, V, w = native.shape[-3:]
    flat = native.movedim(-3, -2).reshape(*native.shape[:-3], V, nch * w)
    real_dtype = torch.float64 if out_dtype == torch.complex128 else torch.float32
    flat = flat.to(real_dtype)
    return torch.view_as_complex(flat.reshape(*flat.shape[:-1], (nch * w) // 2, 2).contiguous())
