"""Gauge fixing by checkerboard overrelaxation (ref: lib/gauge_fix_ovr.cu
+ kernels/gauge_fix_ovr.cuh computeGaugeFixingOVRQuda — re-derived:
maximize F[g] = sum_{x, mu in dirs} Re tr[g(x) U_mu(x) g(x+mu)^dag]
(dirs = 0..3 Landau, 0..2 Coulomb) with per-site polar-optimal updates and
omega-overrelaxation via the unitary eigendecomposition)."""

from __future__ import annotations

from typing import Tuple

import torch

from ..fields.geometry import LatticeGeometry
from .ops import _from_lex, _to_lex, _shift


def _su3_maximize(w: torch.Tensor, hits: int = 3) -> torch.Tensor:
    """g in SU(3) (locally) maximizing Re tr(g w) via Cabibbo-Marinari
    SU(2)-subgroup hits (each hit takes the conjugate quaternion of the
    subgroup projection, the per-subgroup argmax; same machinery as the
    heatbath, ref kernels/gauge_fix_ovr.cuh)."""
    from .heatbath import _SUBGROUPS, _su2_embed_mul, _su2_extract
    g = torch.zeros_like(w)
    g[...] = torch.eye(3, dtype=w.dtype, device=w.device)
    W = w.clone()
    for _ in range(hits):
        for (i, j) in _SUBGROUPS:
            a0, a1, a2, a3 = _su2_extract(W, i, j)
            k = torch.sqrt(a0 ** 2 + a1 ** 2 + a2 ** 2 + a3 ** 2).clamp_min(1e-30)
            r0, r1, r2, r3 = a0 / k, -a1 / k, -a2 / k, -a3 / k
            W = _su2_embed_mul(W, r0, r1, r2, r3, i, j)
            g = _su2_embed_mul(g, r0, r1, r2, r3, i, j)
    return g


def gauge_fix_quality(u: torch.Tensor, geo: LatticeGeometry,
                      dirs: int = 4) -> Tuple[float, float]:
    """(functional, theta): functional = mean Re tr U_mu / 3 over the fixed
    dirs; theta = mean |Delta(x)|^2 with Delta = sum_mu (A_mu(x) -
    A_mu(x-mu)), A = TA[U] (ref gauge_fix_ovr.cuh quality reduction)."""
    U = _to_lex(u, geo)
    func = 0.0
    V = geo.volume
    Delta = torch.zeros((V, 3, 3), dtype=u.dtype, device=u.device)
    for mu in range(dirs):
        func += torch.diagonal(U[mu], dim1=-2, dim2=-1).sum(-1).real.mean().item() / 3.0
        A = U[mu] - U[mu].conj().mT
        tr = torch.diagonal(A, dim1=-2, dim2=-1).sum(-1) / 3.0
        A = A - tr[..., None, None] * torch.eye(3, dtype=u.dtype, device=u.device)
        Delta += A - _shift(A, geo, mu, -1)
    theta = (Delta.conj() * Delta).sum().real.item() / (3 * V)
    return func / dirs, theta


def gauge_fix_ovr(u: torch.Tensor, geo: LatticeGeometry, *,
                  gauge: str = "landau", omega: float = 1.7,
                  max_iter: int = 200, tol: float = 1e-8) -> torch.Tensor:
    """Returns the gauge-fixed links (Landau: all 4 dirs; Coulomb: spatial
    only). Checkerboard sweeps; stops when theta < tol."""
    dirs = 4 if gauge == "landau" else 3
    U = _to_lex(u, geo).clone()
    parity_lex = geo.parity.to(u.device)
    for it in range(max_iter):
        for p in (0, 1):
            w = torch.zeros((geo.volume, 3, 3), dtype=u.dtype, device=u.device)
            for mu in range(dirs):
                w = w + U[mu] + _shift(U[mu], geo, mu, -1).conj().mT
            g = _su3_maximize(w)
            eye = torch.eye(3, dtype=u.dtype, device=u.device)
            sel = (parity_lex == p)
            g = torch.where(sel[:, None, None], g, eye)
            for mu in range(4):
                idx = geo.neighbor_lex(mu, +1).to(u.device)
                U[mu] = g @ U[mu] @ g[idx].conj().mT
        _, theta = gauge_fix_quality(_from_lex(U, geo), geo, dirs)
        if theta < tol:
            break
    return _from_lex(U, geo)


def _ta_field(U: torch.Tensor) -> torch.Tensor:
    A = (U - U.conj().mT) / 2.0
    tr = torch.diagonal(A, dim1=-2, dim2=-1).sum(-1) / 3.0
    eye = torch.eye(3, dtype=U.dtype, device=U.device)
    return A - tr[..., None, None] * eye


def gauge_fix_fft(u: torch.Tensor, geo: LatticeGeometry, *,
                  gauge: str = "landau", alpha: float = 0.08,
                  max_iter: int = 500, tol: float = 1e-8) -> torch.Tensor:
    """Fourier-accelerated steepest-descent gauge fixing
    (ref: lib/gauge_fix_fft.cu computeGaugeFixingFFTQuda, Davies et al.:
    g(x) = exp(alpha/2 * Finv[ phat2_max/phat2 F[Delta] ]); torch.fft is
    hipFFT on ROCm)."""
    import math
    dirs = 4 if gauge == "landau" else 3
    U = _to_lex(u, geo).clone()
    X, Y, Z, T = geo.dims
    dev = u.device
    # phat^2 on the lattice momentum grid
    ks = [torch.arange(n, dtype=torch.float64, device=dev) for n in geo.dims]
    p2 = torch.zeros((X, Y, Z, T), dtype=torch.float64, device=dev)
    for i, (n, k) in enumerate(zip(geo.dims, ks)):
        if i >= dirs:
            pass  # Coulomb still smooths in all momentum dims of the slice
        s = (2.0 * torch.sin(math.pi * k / n)) ** 2
        shape = [1, 1, 1, 1]
        shape[i] = n
        p2 = p2 + s.reshape(shape)
    p2max = p2.max()
    invp2 = torch.where(p2 > 1e-14, p2max / p2, torch.zeros_like(p2))

    for it in range(max_iter):
        # Delta(x) = sum_mu [A_mu(x) - A_mu(x-mu)]
        Delta = torch.zeros((geo.volume, 3, 3), dtype=u.dtype, device=dev)
        for mu in range(dirs):
            A = _ta_field(U[mu])
            Delta += A - _shift(A, geo, mu, -1)
        th = (Delta.conj() * Delta).sum().real.item() / (3 * geo.volume)
        if th < tol:
            break
        # Fourier precondition (lex order is x-fastest: reshape [T,Z,Y,X])
        D4 = Delta.reshape(T, Z, Y, X, 9).permute(3, 2, 1, 0, 4)
        Dk = torch.fft.fftn(D4, dim=(0, 1, 2, 3))
        Dk = Dk * invp2[..., None].to(Dk.dtype)
        phi = torch.fft.ifftn(Dk, dim=(0, 1, 2, 3))
        phi = phi.permute(3, 2, 1, 0, 4).reshape(geo.volume, 3, 3)
        g = torch.matrix_exp((-alpha / 2.0) * phi)
        # project g back to SU(3)-ish (expm of approx-TA can drift)
        from ..fields.gauge import project_su3
        g = project_su3(g)
        for mu in range(4):
            idx = geo.neighbor_lex(mu, +1).to(dev)
            U[mu] = g @ U[mu] @ g[idx].conj().mT
    return _from_lex(U, geo)
