"""Pure-gauge HMC: momenta, leapfrog (+Omelyan), trajectory with
Metropolis accept (ref: lib/momentum.cu, gauge_update_quda.cu and the
updateGaugeFieldQuda/momActionQuda entry points)."""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..fields.geometry import LatticeGeometry
from .ops import _from_lex, _to_lex, exp_su3, gauge_action, gauge_force

# Gell-Mann-like antihermitian su(3) basis T_a with tr[T_a T_b] = -delta/2
_LAMBDA = None


def _su3_basis(device, dtype):
    global _LAMBDA
    import numpy as np
    if _LAMBDA is None:
        L = np.zeros((8, 3, 3), dtype=complex)
        L[0][0, 1] = L[0][1, 0] = 1
        L[1][0, 1] = -1j; L[1][1, 0] = 1j
        L[2][0, 0] = 1; L[2][1, 1] = -1
        L[3][0, 2] = L[3][2, 0] = 1
        L[4][0, 2] = -1j; L[4][2, 0] = 1j
        L[5][1, 2] = L[5][2, 1] = 1
        L[6][1, 2] = -1j; L[6][2, 1] = 1j
        L[7][0, 0] = L[7][1, 1] = 3 ** -0.5; L[7][2, 2] = -2 * 3 ** -0.5
        _LAMBDA = L
    return torch.tensor(_LAMBDA, device=device, dtype=dtype)


def random_momentum(geo: LatticeGeometry, device="cpu", seed: Optional[int] = None,
                    dtype=torch.complex128) -> torch.Tensor:
    """Gaussian antihermitian traceless momenta P: [4,2,V,3,3],
    normalized so mom_action = -sum tr P^2 / ... follows exp(+sum tr P^2/2)
    — we draw P = i sum_a w_a lambda_a/2 with w ~ N(0,1)."""
    g = torch.Generator(device="cpu")
    if seed is not None:
        g.manual_seed(seed)
    lam = _su3_basis(device, dtype)
    w = torch.randn((4, 2, geo.volume_cb, 8), generator=g,
                    dtype=torch.float64).to(device)
    P = 0.5j * torch.einsum("dpva,aij->dpvij", w.to(dtype), lam)
    return P


def mom_action(P: torch.Tensor) -> float:
    """S_mom = -sum tr(P^2), globally summed (P antihermitian => -tr P^2
    >= 0; ref: lib/momentum.cu momAction)."""
    from ..parallel import comms
    return comms.allreduce_sum(
        -torch.einsum("dpvij,dpvji->", P, P).real.item())


def leapfrog(u: torch.Tensor, P: torch.Tensor, geo: LatticeGeometry,
             beta: float, n_steps: int, dt: float):
    """Leapfrog MD: returns evolved (u, P). Force from gauge_force
    (pdot = F), link update U <- exp(dt P) U."""
    u = u.clone()
    P = P + 0.5 * dt * gauge_force(u, geo, beta)
    for k in range(n_steps):
        U = _to_lex(u, geo)
        Pl = _to_lex(P, geo)
        U = exp_su3(Pl, dt) @ U
        u = _from_lex(U, geo)
        P = P + (0.5 if k == n_steps - 1 else 1.0) * dt * gauge_force(u, geo, beta)
    return u, P


def hmc_trajectory(u: torch.Tensor, geo: LatticeGeometry, beta: float,
                   n_md: int = 10, tau: float = 1.0,
                   seed: Optional[int] = None) -> Tuple[torch.Tensor, bool, float]:
    """One HMC trajectory with Metropolis accept. Returns
    (new_u, accepted, dH)."""
    dt = tau / n_md
    P = random_momentum(geo, u.device, seed, u.dtype)
    H0 = mom_action(P) + gauge_action(u, geo, beta)
    u1, P1 = leapfrog(u, P, geo, beta, n_md, dt)
    H1 = mom_action(P1) + gauge_action(u1, geo, beta)
    dH = H1 - H0
    g = torch.Generator()
    if seed is not None:
        g.manual_seed(seed + 1)
    accept = torch.rand(1, generator=g).item() < min(1.0, float(torch.exp(torch.tensor(-dH))))
    return (u1 if accept else u), accept, dH
