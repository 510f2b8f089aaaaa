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


def _evolve_u(u: torch.Tensor, P: torch.Tensor, geo: LatticeGeometry,
              dt: float) -> torch.Tensor:
    U = _to_lex(u, geo)
    Pl = _to_lex(P, geo)
    return _from_lex(exp_su3(Pl, dt) @ U, geo)


_OMELYAN_LAMBDA = 0.1931833275037836  # minimal-norm 2nd-order coefficient


def omelyan(u: torch.Tensor, P: torch.Tensor, geo: LatticeGeometry,
            force_fn, n_steps: int, dt: float,
            lam: float = _OMELYAN_LAMBDA):
    """Second-order minimal-norm (Omelyan/2MN) integrator: per step
    P(lam dt) U(dt/2) P((1-2lam)dt) U(dt/2) P(lam dt). Same O(dt^2)
    per-trajectory energy error order as leapfrog with a ~10x smaller
    coefficient at 1.5x the force evaluations (ref: the integrator the
    reference's HMC consumers pair with its force kernels)."""
    u = u.clone()
    for k in range(n_steps):
        P = P + (lam * dt) * force_fn(u)
        u = _evolve_u(u, P, geo, 0.5 * dt)
        P = P + ((1.0 - 2.0 * lam) * dt) * force_fn(u)
        u = _evolve_u(u, P, geo, 0.5 * dt)
        P = P + (lam * dt) * force_fn(u)
    return u, P


def nested_leapfrog(u: torch.Tensor, P: torch.Tensor, geo: LatticeGeometry,
                    levels, n_steps: int, dt: float):
    """Sexton-Weingarten multi-timescale leapfrog. `levels` is a list of
    (force_fn, n_sub) outermost (expensive, e.g. fermion) first; each
    inner level subdivides its parent's step by n_sub; the innermost
    drift is the link update. The expensive force is evaluated n_sub
    times less often than the cheap one."""
    state = {"u": u.clone(), "P": P}

    def step(h: float, lev: int):
        if lev == len(levels):
            state["u"] = _evolve_u(state["u"], state["P"], geo, h)
            return
        force, nsub = levels[lev]
        hs = h / nsub
        state["P"] = state["P"] + (0.5 * hs) * force(state["u"])
        for i in range(nsub):
            step(hs, lev + 1)
            if i != nsub - 1:
                state["P"] = state["P"] + hs * force(state["u"])
        state["P"] = state["P"] + (0.5 * hs) * force(state["u"])

    for _ in range(n_steps):
        step(dt, 0)
    return state["u"], state["P"]
