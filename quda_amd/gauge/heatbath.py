"""Pure-gauge SU(3) heatbath + overrelaxation via SU(2) subgroups
(ref: lib/pgauge_heatbath.cu, kernels/gauge_heatbath.cuh —
Cabibbo-Marinari / Kennedy-Pendleton, re-derived)."""

from __future__ import annotations

from typing import Optional

import torch

from ..fields.geometry import LatticeGeometry
from .ops import _from_lex, _to_lex, staple_sum

_SUBGROUPS = [(0, 1), (0, 2), (1, 2)]


def _su2_extract(W: torch.Tensor, i: int, j: int):
    """SU(2)-projected (a0,a1,a2,a3) of the (i,j) 2x2 submatrix of W."""
    a0 = (W[..., i, i].real + W[..., j, j].real) / 2
    a1 = (W[..., i, j].imag + W[..., j, i].imag) / 2
    a2 = (W[..., i, j].real - W[..., j, i].real) / 2
    a3 = (W[..., i, i].imag - W[..., j, j].imag) / 2
    return a0, a1, a2, a3


def _su2_embed_mul(U: torch.Tensor, r0, r1, r2, r3, i: int, j: int):
    """Left-multiply U by the SU(2) matrix r embedded in rows/cols (i,j)."""
    out = U.clone()
    a = r0 + 1j * r3
    b = r2 + 1j * r1
    out[..., i, :] = (a * U[..., i, :].mT).mT + (b * U[..., j, :].mT).mT
    out[..., j, :] = (-torch.conj(b) * U[..., i, :].mT).mT + (torch.conj(a) * U[..., j, :].mT).mT
    return out


_DEVICE_RNG = False


def set_device_rng(v: bool) -> None:
    """Draw Monte-Carlo randoms directly on the field's device (fast path
    for GPU thermalization; host RNG with a seeded generator is the
    deterministic default — the host path proved PCIe-bound at scale)."""
    global _DEVICE_RNG
    _DEVICE_RNG = bool(v)


def _rand(n, gen, device, dtype=torch.float64):
    if _DEVICE_RNG and str(device) != "cpu":
        return torch.rand(n, dtype=dtype, device=device)
    return torch.rand(n, generator=gen, dtype=dtype).to(device)


def _kp_sample(k: torch.Tensor, beta_eff: float, gen) -> torch.Tensor:
    """Kennedy-Pendleton sampling of a0 with density
    ~ sqrt(1-a0^2) exp(beta_eff k a0) (vectorized rejection; RNG on host,
    sampling math on the field's device)."""
    n = k.shape
    dev = k.device
    a0 = torch.empty_like(k)
    todo = torch.ones_like(k, dtype=torch.bool)
    alpha = beta_eff * k
    it = 0
    import math
    while todo.any() and it < 100:
        r1 = _rand(n, gen, dev).clamp_min(1e-12)
        r2 = _rand(n, gen, dev)
        r3 = _rand(n, gen, dev).clamp_min(1e-12)
        x = -(torch.log(r1) + (torch.cos(2 * math.pi * r2)) ** 2 * torch.log(r3)) / alpha
        acc = (_rand(n, gen, dev) ** 2 <= 1 - 0.5 * x) & (x <= 2.0)
        upd = todo & acc
        a0[upd] = (1 - x)[upd]
        todo = todo & ~acc
        it += 1
    if todo.any():
        a0[todo] = 2 * _rand((int(todo.sum().item()),), gen, dev) - 1  # fallback
    return a0


def _sweep(u, geo, beta, gen, mode: str):
    U = _to_lex(u, geo)
    parity_lex = geo.parity.to(u.device)
    for mu in range(4):
        for p in (0, 1):
            sel = parity_lex == p
            S = staple_sum(U, geo, mu)
            for (i, j) in _SUBGROUPS:
                W = U[mu] @ S.conj().mT
                a0, a1, a2, a3 = _su2_extract(W, i, j)
                k = torch.sqrt(a0 ** 2 + a1 ** 2 + a2 ** 2 + a3 ** 2).clamp_min(1e-30)
                v0, v1, v2, v3 = a0 / k, a1 / k, a2 / k, a3 / k
                if mode == "heatbath":
                    b0 = _kp_sample(k.to(torch.float64), 2 * beta / 3,
                                    gen).to(a0.dtype)
                    rho = torch.sqrt((1 - b0 ** 2).clamp_min(0))
                    import math
                    ct = 2 * _rand(a0.shape, gen, a0.device, a0.dtype) - 1
                    st = torch.sqrt((1 - ct ** 2).clamp_min(0))
                    ph = 2 * math.pi * _rand(a0.shape, gen, a0.device, a0.dtype)
                    b1 = rho * st * torch.cos(ph)
                    b2 = rho * st * torch.sin(ph)
                    b3 = rho * ct
                    # new su2 g = b . v^-1
                    r0 = b0 * v0 + b1 * v1 + b2 * v2 + b3 * v3
                    r1 = -b0 * v1 + b1 * v0 - b2 * v3 + b3 * v2
                    r2 = -b0 * v2 + b2 * v0 - b3 * v1 + b1 * v3
                    r3 = -b0 * v3 + b3 * v0 - b1 * v2 + b2 * v1
                else:  # overrelax: g = v^-2 (reflects W)
                    r0 = 2 * v0 * v0 - 1
                    r1 = -2 * v0 * v1
                    r2 = -2 * v0 * v2
                    r3 = -2 * v0 * v3
                Unew = _su2_embed_mul(U[mu], r0, r1, r2, r3, i, j)
                U[mu][sel] = Unew[sel]
    return _from_lex(U, geo)


def _native_sweep(u, geo, beta, seed, mode: int):
    """One sweep through the hand-written HIP kernel (csrc/heatbath.hip):
    8 launches (mu x parity), staples + all 3 SU(2) subgroup hits fused
    per site, counter-based in-kernel RNG."""
    from ..ops.dispatch import hip_ext
    ext = hip_ext()
    uc = u if u.is_contiguous() else u.contiguous()
    for mu in range(4):
        for p in (0, 1):
            ext.heatbath_sweep_dir(uc, list(geo.dims), geo.parity_offset,
                                   geo.volume_cb, p, mu, 2.0 * beta / 3.0,
                                   int(seed or 0), mode)
    return uc


def _use_native(u, geo) -> bool:
    from ..parallel import comms
    import os
    if os.environ.get("QUDA_AMD_NATIVE_HEATBATH", "1") == "0":
        return False
    return (u.device.type == "cuda" and u.dtype == torch.complex128
            and not comms.comm_mask())


def heatbath_sweep(u, geo: LatticeGeometry, beta: float,
                   seed: Optional[int] = None):
    """One Cabibbo-Marinari heatbath sweep (all links, both parities).
    On a single GPU the native HIP kernel runs (8 launches/sweep); the
    torch path is the multi-rank / CPU / deterministic-host-RNG route."""
    if _use_native(u, geo):
        return _native_sweep(u, geo, beta, seed if seed is not None else 1,
                             0)
    gen = torch.Generator()
    if seed is not None:
        gen.manual_seed(seed)
    return _sweep(u, geo, beta, gen, "heatbath")


def overrelax_sweep(u, geo: LatticeGeometry, beta: float,
                    seed: Optional[int] = None):
    """One SU(2)-subgroup overrelaxation sweep (action-preserving)."""
    if _use_native(u, geo):
        return _native_sweep(u, geo, beta, 0, 1)
    gen = torch.Generator()
    if seed is not None:
        gen.manual_seed(seed)
    return _sweep(u, geo, beta, gen, "overrelax")
