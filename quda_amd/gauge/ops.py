"""Gauge observables, smearing, flow and forces
(ref: lib/gauge_plaq.cu, gauge_ape.cu, gauge_stout.cu, gauge_wilson_flow.cu,
gauge_force.cu, gauge_qcharge.cu, gauge_polyakov_loop.cu + the kernels in
include/kernels/gauge_*.cuh — re-derived from the standard definitions)."""

from __future__ import annotations

import torch

from ..fields.geometry import LatticeGeometry
from ..ops.reference import field_strength


def _to_lex(u: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    """[4,2,V,3,3] cb -> [4,Vlex,3,3]."""
    lo = geo.lex_of_cb.to(u.device)
    U = torch.empty((4, geo.volume, 3, 3), dtype=u.dtype, device=u.device)
    U[:, lo[0]] = u[:, 0]
    U[:, lo[1]] = u[:, 1]
    return U


def _from_lex(U: torch.Tensor, geo: LatticeGeometry) -> torch.Tensor:
    lo = geo.lex_of_cb.to(U.device)
    return torch.stack([U[:, lo[0]], U[:, lo[1]]], dim=1)


def _shift(f: torch.Tensor, geo: LatticeGeometry, mu: int, disp: int):
    """f: [Vlex,...] -> f(x + disp*mu) (neighbor-rank slab on partitioned
    dims — staples/forces/smearing are multi-rank correct)."""
    from ..parallel.halo import shift_lex
    return shift_lex(f, geo, mu, disp)


def plaquette(u: torch.Tensor, geo: LatticeGeometry):
    """(total, spatial, temporal) mean plaquette Re tr P / 3
    (ref: lib/gauge_plaq.cu)."""
    from ..parallel import comms
    n_ranks = comms.comm_size() if comms.comm_mask() else 1
    U = _to_lex(u, geo)
    tot_s = tot_t = 0.0
    n_s = n_t = 0
    for mu in range(4):
        for nu in range(mu + 1, 4):
            Unu_xmu = _shift(U[nu], geo, mu, +1)
            Umu_xnu = _shift(U[mu], geo, nu, +1)
            P = U[mu] @ Unu_xmu @ Umu_xnu.conj().mT @ U[nu].conj().mT
            val = comms.allreduce_sum(
                torch.diagonal(P, dim1=-2, dim2=-1).sum(-1).real.sum().item()
            ) / (3.0 * geo.volume * n_ranks)
            if nu == 3:
                tot_t += val
                n_t += 1
            else:
                tot_s += val
                n_s += 1
    sp = tot_s / n_s
    tm = tot_t / n_t
    return ((sp * n_s + tm * n_t) / (n_s + n_t), sp, tm)


def staple_sum(U: torch.Tensor, geo: LatticeGeometry, mu: int) -> torch.Tensor:
    """Sum of the 6 staples around U_mu (lex layout input [4,Vlex,3,3]).

    staple(nu,+) = U_nu(x) U_mu(x+nu) U_nu(x+mu)^d
    staple(nu,-) = U_nu(x-nu)^d U_mu(x-nu) U_nu(x+mu-nu)
    normalized so Re tr[U_mu(x) staple^d] = sum of the Re-traces of the 6
    plaquettes containing U_mu(x).
    """
    S = torch.zeros_like(U[mu])
    for nu in range(4):
        if nu == mu:
            continue
        Unu_xmu = _shift(U[nu], geo, mu, +1)
        Umu_xnu = _shift(U[mu], geo, nu, +1)
        S = S + U[nu] @ Umu_xnu @ Unu_xmu.conj().mT
        Unu_mnu = _shift(U[nu], geo, nu, -1)
        Umu_mnu = _shift(U[mu], geo, nu, -1)
        Unu_xmu_mnu = _shift(Unu_mnu, geo, mu, +1)
        S = S + Unu_mnu.conj().mT @ Umu_mnu @ Unu_xmu_mnu
    return S


def gauge_action(u: torch.Tensor, geo: LatticeGeometry, beta: float) -> float:
    """Wilson gauge action S = beta * sum_P (1 - Re tr P / 3) (global)."""
    from ..parallel import comms
    n_ranks = comms.comm_size() if comms.comm_mask() else 1
    p_tot, _, _ = plaquette(u, geo)
    n_plaq = 6 * geo.volume * n_ranks
    return beta * n_plaq * (1.0 - p_tot)


def project_ta(M: torch.Tensor) -> torch.Tensor:
    """Traceless antihermitian part: (M - M^d)/2 - tr(M - M^d)/6."""
    A = (M - M.conj().mT) / 2.0
    tr = torch.diagonal(A, dim1=-2, dim2=-1).sum(-1) / 3.0
    eye = torch.eye(3, dtype=M.dtype, device=M.device)
    return A - tr[..., None, None] * eye


def gauge_force(u: torch.Tensor, geo: LatticeGeometry, beta: float) -> torch.Tensor:
    """Wilson-action MD force (ref: lib/gauge_force.cu), normalized for
    OUR Hamiltonian convention H = -sum tr P^2 + S(U), Udot = P U:
    energy conservation then requires Pdot = F = -(beta/6) TA[U staple^d]
    (TA = project_ta; derivation: tr(P TA[M]) = Re tr(P M), and
    dS/dt = -(beta/3) sum Re tr(P U staple^d))."""
    U = _to_lex(u, geo)
    F = torch.empty_like(U)
    for mu in range(4):
        S = staple_sum(U, geo, mu)
        F[mu] = -(beta / 6.0) * project_ta(U[mu] @ S.conj().mT)
    return _from_lex(F, geo)


def exp_su3(A: torch.Tensor, scale: float = 1.0) -> torch.Tensor:
    """exp(scale*A) for [...,3,3] antihermitian A (torch.matrix_exp)."""
    return torch.matrix_exp(scale * A)


def ape_smear(u: torch.Tensor, geo: LatticeGeometry, alpha: float,
              n_iter: int = 1, spatial_only: bool = False) -> torch.Tensor:
    """APE smearing: U' = Proj_SU3[(1-alpha) U + (alpha/6) staple]
    (ref: lib/gauge_ape.cu)."""
    from ..fields.gauge import project_su3
    out = u
    dims = range(3) if spatial_only else range(4)
    for _ in range(n_iter):
        U = _to_lex(out, geo)
        Unew = U.clone()
        for mu in dims:
            S = staple_sum(U, geo, mu)
            n_st = 6 if not spatial_only else 4
            M = (1 - alpha) * U[mu] + (alpha / n_st) * S
            Unew[mu] = project_su3(M)
        out = _from_lex(Unew, geo)
    return out


def stout_smear(u: torch.Tensor, geo: LatticeGeometry, rho: float,
                n_iter: int = 1) -> torch.Tensor:
    """Stout smearing U' = exp(rho * TA[U staple^d])^d ... standard:
    Q = TA[Omega], Omega = U_mu staple^d; U' = exp(-rho Q)... sign per
    Morningstar-Peardon: U' = exp(i rho Q_herm) U; here with antihermitian
    Q: U' = exp(rho * TA[staple U^d... ]) — we use
    U'_mu = exp(rho * TA[S U_mu^d]) U_mu (ref: lib/gauge_stout.cu)."""
    from ..parallel import comms
    import os
    if (u.device.type == "cuda" and u.dtype == torch.complex128
            and not comms.comm_mask()
            and os.environ.get("QUDA_AMD_NATIVE_SMEAR", "1") != "0"):
        # native HIP kernel (csrc/heatbath.hip k_stout): staple + TA +
        # scale-and-square exp fused per site, 4 launches per iteration
        from ..ops.dispatch import hip_ext
        ext = hip_ext()
        cur = u.contiguous().clone()
        nxt = torch.empty_like(cur)
        for _ in range(n_iter):
            for mu in range(4):
                ext.stout_smear_dir(nxt, cur, list(geo.dims),
                                    geo.parity_offset, geo.volume_cb, mu,
                                    float(rho))
            cur, nxt = nxt, cur
        return cur
    out = u
    for _ in range(n_iter):
        U = _to_lex(out, geo)
        Unew = U.clone()
        for mu in range(4):
            S = staple_sum(U, geo, mu)
            Q = project_ta(S @ U[mu].conj().mT)
            Unew[mu] = exp_su3(Q, rho) @ U[mu]
        out = _from_lex(Unew, geo)
    return out


def wilson_flow(u: torch.Tensor, geo: LatticeGeometry, eps: float,
                n_steps: int = 1) -> torch.Tensor:
    """Wilson (gradient) flow, RK3 Luscher scheme
    (ref: lib/gauge_wilson_flow.cu):
      W0 = U;  Z_i = eps * grad S(W_i)
      W1 = exp(1/4 Z0) W0
      W2 = exp(8/9 Z1 - 17/36 Z0) W1
      U' = exp(3/4 Z2 - 8/9 Z1 + 17/36 Z0) W2
    with grad = TA[S W^d] (flow toward smaller action)."""
    from ..parallel import comms
    import os
    if (u.device.type == "cuda" and u.dtype == torch.complex128
            and not comms.comm_mask()
            and os.environ.get("QUDA_AMD_NATIVE_SMEAR", "1") != "0"):
        # native kernels (csrc/heatbath.hip k_zmat/k_expmul): staple+TA
        # fused, RK3 stage combos as cheap tensor axpys
        from ..ops.dispatch import hip_ext
        ext = hip_ext()
        dims, po, Vcb = list(geo.dims), geo.parity_offset, geo.volume_cb

        def zmat_n(W):
            Z = torch.empty_like(W)
            for mu in range(4):
                ext.flow_zmat_dir(Z, W, dims, po, Vcb, mu, float(eps))
            return Z

        def expmul_n(Zc, W):
            out_ = torch.empty_like(W)
            Zc = Zc.contiguous()
            for mu in range(4):
                ext.flow_expmul_dir(out_, W, Zc, dims, po, Vcb, mu)
            return out_

        cur = u.contiguous().clone()
        for _ in range(n_steps):
            Z0 = zmat_n(cur)
            W1 = expmul_n(0.25 * Z0, cur)
            Z1 = zmat_n(W1)
            W2 = expmul_n((8.0 / 9.0) * Z1 - (17.0 / 36.0) * Z0, W1)
            Z2 = zmat_n(W2)
            cur = expmul_n(
                0.75 * Z2 - (8.0 / 9.0) * Z1 + (17.0 / 36.0) * Z0, W2)
        return cur

    def zmat(U):
        Z = torch.empty_like(U)
        for mu in range(4):
            S = staple_sum(U, geo, mu)
            Z[mu] = eps * project_ta(S @ U[mu].conj().mT)
        return Z

    out = u
    for _ in range(n_steps):
        W0 = _to_lex(out, geo)
        Z0 = zmat(W0)
        W1 = torch.matrix_exp(0.25 * Z0) @ W0
        Z1 = zmat(W1)
        W2 = torch.matrix_exp((8.0 / 9.0) * Z1 - (17.0 / 36.0) * Z0) @ W1
        Z2 = zmat(W2)
        W3 = torch.matrix_exp(0.75 * Z2 - (8.0 / 9.0) * Z1 + (17.0 / 36.0) * Z0) @ W2
        out = _from_lex(W3, geo)
    return out


def polyakov_loop(u: torch.Tensor, geo: LatticeGeometry) -> complex:
    """Mean Polyakov loop (1/3) <tr prod_t U_t(x,t)>
    (ref: lib/gauge_polyakov_loop.cu)."""
    U = _to_lex(u, geo)
    X, Y, Z, T = geo.dims
    Ut = U[3].reshape(T, Z * Y * X, 3, 3)  # t slowest in lex
    P = Ut[0]
    for t in range(1, T):
        P = P @ Ut[t]
    tr = torch.diagonal(P, dim1=-2, dim2=-1).sum(-1) / 3.0
    m = tr.mean()
    return complex(m.real.item(), m.imag.item())


def topological_charge(u: torch.Tensor, geo: LatticeGeometry) -> float:
    """Field-theoretic Q = (1/32 pi^2) sum_x eps_{munurhosig}
    tr[F_munu F_rhosig] via the clover-leaf F (ref: lib/gauge_qcharge.cu)."""
    import math
    F = field_strength(u, geo)  # hermitian convention {(mu,nu): [2,V,3,3]}
    # Q = (1/4pi^2) sum tr[F01 F23 - F02 F13 + F03 F12] with hermitian F
    def trprod(a, b):
        return torch.einsum("pvij,pvji->", F[a], F[b]).real.item()

    from ..parallel import comms
    q = trprod((0, 1), (2, 3)) - trprod((0, 2), (1, 3)) + trprod((0, 3), (1, 2))
    return comms.allreduce_sum(q) / (4.0 * math.pi ** 2)


def wilson_loop(u: torch.Tensor, geo: LatticeGeometry, R: int, T: int,
                mu: int = 0, nu: int = 3) -> complex:
    """Mean R x T rectangular Wilson loop in the (mu,nu) plane
    (ref: lib/gauge_loop_trace.cu batched loop traces)."""
    U = _to_lex(u, geo)

    # path product: R steps of +mu, T of +nu, R of -mu, T of -nu
    V = geo.volume
    P = torch.eye(3, dtype=u.dtype, device=u.device).expand(V, 3, 3).clone()
    shift_total = [0, 0, 0, 0]

    def shifted(f):
        out = f
        for d in range(4):
            s = shift_total[d]
            while s > 0:
                out = _shift(out, geo, d, +1)
                s -= 1
            while s < 0:
                out = _shift(out, geo, d, -1)
                s += 1
        return out

    for _ in range(R):
        P = P @ shifted(U[mu])
        shift_total[mu] += 1
    for _ in range(T):
        P = P @ shifted(U[nu])
        shift_total[nu] += 1
    for _ in range(R):
        shift_total[mu] -= 1
        P = P @ shifted(U[mu]).conj().mT
    for _ in range(T):
        shift_total[nu] -= 1
        P = P @ shifted(U[nu]).conj().mT
    tr = torch.diagonal(P, dim1=-2, dim2=-1).sum(-1).mean() / 3.0
    return complex(tr.real.item(), tr.imag.item())


def energy_density(u: torch.Tensor, geo: LatticeGeometry) -> tuple:
    """(E_plaq, E_clover): gradient-flow energy densities
    (ref: lib/gauge_wilson_flow.cu energy measurement + gauge_qcharge.cuh
    E from the clover F_munu). E_plaq = 2 sum_{mu<nu} Re tr(1 - P);
    E_clover = sum_{mu<nu} -tr(F F) with F the TA clover field strength.
    Both are global means per site (allreduced)."""
    from ..parallel import comms
    from ..ops.reference import field_strength
    tot, _, _ = plaquette(u, geo)
    # plaquette() returns mean Re tr P / 3 over 6 planes
    e_plaq = 2.0 * 6 * 3 * (1.0 - tot)
    e_clov = 0.0
    Fd = field_strength(u, geo)
    for F in Fd.values():
        # tr[F F^dag] = sum |F|^2 is convention-proof (hermitian or
        # antihermitian F_munu normalization)
        e_clov += torch.einsum("pvij,pvij->", F, F.conj()).real.item()
    n = comms.allreduce_sum(float(geo.volume))
    e_clov = comms.allreduce_sum(e_clov) / n
    return e_plaq, e_clov


def wilson_flow_measure(u: torch.Tensor, geo: LatticeGeometry, eps: float,
                        n_steps: int):
    """Flow + measure: returns (u_flowed, history) with history entries
    (t, E_plaq, E_clover, t^2 E_clover) per step (ref: the wflow path of
    performGaugeSmearQuda + gaugeObservables)."""
    hist = []
    out = u
    t = 0.0
    for k in range(n_steps):
        out = wilson_flow(out, geo, eps, 1)
        t += eps
        ep, ec = energy_density(out, geo)
        hist.append((t, ep, ec, t * t * ec))
    return out, hist


def flow_scale_t0(hist, target: float = 0.3):
    """t0 from t^2 E(t) = target by linear interpolation (Luscher scale
    setting); None if the flow history never crosses the target."""
    for (t1, _, _, s1), (t2, _, _, s2) in zip(hist, hist[1:]):
        if s1 < target <= s2:
            return t1 + (target - s1) * (t2 - t1) / (s2 - s1)
    return None


def flow_scale_w0(hist, target: float = 0.3):
    """w0^2 from t d/dt[t^2 E] = target (BMW w0 scale), finite-difference
    derivative on the measured history."""
    for i in range(1, len(hist) - 1):
        t = hist[i][0]
        d = t * (hist[i + 1][3] - hist[i - 1][3]) / (hist[i + 1][0]
                                                     - hist[i - 1][0])
        d_next = hist[i + 1][0] * (hist[min(i + 2, len(hist) - 1)][3]
                                   - hist[i][3]) / (
            hist[min(i + 2, len(hist) - 1)][0] - hist[i][0])
        if d < target <= d_next:
            return t + (target - d) * (hist[i + 1][0] - t) / (d_next - d)
    return None


def path_product(u: torch.Tensor, geo: LatticeGeometry,
                 path) -> torch.Tensor:
    """[V,3,3] product of links along `path` starting at every site
    (ref: lib/gauge_loop_trace.cu + gauge_path_helper.cuh — generic
    signed-direction paths). path entries: +-(mu+1), e.g. the (0,3)
    plaquette is (1, 4, -1, -4). Comm-aware via shift_lex."""
    from ..parallel.halo import shift_lex
    U = _to_lex(u, geo)
    V = geo.volume
    P = torch.eye(3, dtype=u.dtype, device=u.device).expand(V, 3, 3).clone()
    # walk the path: keep the running product in the frame of the START
    # site by shifting link fields back by the accumulated displacement
    disp = [0, 0, 0, 0]

    def fetch(field):
        out = field
        for d in range(4):
            s = disp[d]
            step = 1 if s > 0 else -1
            for _ in range(abs(s)):
                out = shift_lex(out, geo, d, step)
        return out

    for step in path:
        mu = abs(step) - 1
        if step > 0:
            P = P @ fetch(U[mu])
            disp[mu] += 1
        else:
            disp[mu] -= 1
            P = P @ fetch(U[mu]).conj().mT
    return P


def loop_trace(u: torch.Tensor, geo: LatticeGeometry, paths,
               coeffs=None) -> complex:
    """sum_i c_i <tr P_i>/3 over closed paths, globally averaged (the
    gaugeLoopTraceQuda role). Default coefficients 1."""
    from ..parallel import comms
    coeffs = coeffs or [1.0] * len(paths)
    tot = 0.0 + 0.0j
    n = comms.allreduce_sum(float(geo.volume))
    for c, p in zip(coeffs, paths):
        P = path_product(u, geo, p)
        tr = torch.diagonal(P, dim1=-2, dim2=-1).sum(-1) / 3.0
        s = tr.sum()
        tot += c * complex(comms.allreduce_sum(s.real.item()),
                           comms.allreduce_sum(s.imag.item())) / n
    return tot


def improved_gauge_action(u: torch.Tensor, geo: LatticeGeometry,
                          beta: float, *, c1: float = -1.0 / 12.0) -> float:
    """Luscher-Weisz/Symanzik-improved action
    S = beta sum_x [ c0 sum_pl (1 - Re tr P/3)
                   + c1 sum_rect (1 - Re tr R/3) ],  c0 = 1 - 8 c1
    (c1 = -1/12 tree-level Symanzik, -0.331 Iwasaki; ref: the
    computeGaugeForceQuda path-coefficient interface)."""
    from ..parallel import comms
    c0 = 1.0 - 8.0 * c1
    n = comms.allreduce_sum(float(geo.volume))
    s_pl = 0.0
    s_rt = 0.0
    for mu in range(4):
        for nu in range(4):
            if mu == nu:
                continue
            if mu < nu:
                P = path_product(u, geo, (mu + 1, nu + 1, -(mu + 1),
                                          -(nu + 1)))
                tr = torch.diagonal(P, dim1=-2, dim2=-1).sum(-1).real / 3.0
                s_pl += (1.0 - comms.allreduce_sum(tr.sum().item()) / n)
            # 2x1 rectangles: both orientations (mu-long and nu-long)
            R = path_product(u, geo, (mu + 1, mu + 1, nu + 1, -(mu + 1),
                                      -(mu + 1), -(nu + 1)))
            tr = torch.diagonal(R, dim1=-2, dim2=-1).sum(-1).real / 3.0
            s_rt += (1.0 - comms.allreduce_sum(tr.sum().item()) / n)
    return beta * (c0 * s_pl + c1 * s_rt) * n


def improved_gauge_force(u: torch.Tensor, geo: LatticeGeometry,
                         beta: float, *, c1: float = -1.0 / 12.0
                         ) -> torch.Tensor:
    """Force of the improved action by reverse-mode differentiation of
    the loop traces (single-rank; the plaquette-only force keeps the
    analytic multi-rank path). Same validated convention as
    autograd_fermion_force: F = (1/2) TA[U g^dag] with g the torch
    Wirtinger grad of S — c1=0 reproduces gauge_force exactly and the
    tests check finite differences + dH conservation."""
    from ..parallel import comms
    assert comms.comm_size() == 1, "autograd gauge force is single-rank"
    u_req = u.detach().clone().requires_grad_(True)
    s = _action_autograd(u_req, geo, beta, c1)
    s.backward()
    g = u_req.grad
    F = torch.empty_like(u)
    for mu in range(4):
        for p in (0, 1):
            F[mu, p] = 0.5 * project_ta(u[mu, p] @ g[mu, p].conj().mT)
    return F


def _action_autograd(z, geo, beta, c1, c0=None):
    """improved_gauge_action as a differentiable torch scalar (no
    .item() calls). c0 defaults to the Luscher-Weisz normalization
    1 - 8 c1; pass it explicitly for over-improved smearing weights."""
    if c0 is None:
        c0 = 1.0 - 8.0 * c1
    from ..parallel.halo import shift_lex
    s_pl = 0.0
    s_rt = 0.0
    for mu in range(4):
        for nu in range(4):
            if mu == nu:
                continue
            if mu < nu:
                P = path_product(z, geo, (mu + 1, nu + 1, -(mu + 1),
                                          -(nu + 1)))
                s_pl = s_pl + (geo.volume
                               - torch.diagonal(P, dim1=-2, dim2=-1)
                               .sum(-1).real.sum() / 3.0)
            R = path_product(z, geo, (mu + 1, mu + 1, nu + 1, -(mu + 1),
                                      -(mu + 1), -(nu + 1)))
            s_rt = s_rt + (geo.volume
                           - torch.diagonal(R, dim1=-2, dim2=-1)
                           .sum(-1).real.sum() / 3.0)
    return beta * (c0 * s_pl + c1 * s_rt)


def det_trace(u: torch.Tensor, geo: LatticeGeometry) -> tuple:
    """Global mean link determinant and trace (ref:
    lib/pgauge_det_trace.cu)."""
    from ..parallel import comms
    U = _to_lex(u, geo)
    det = torch.linalg.det(U.reshape(-1, 3, 3))
    tr = torch.diagonal(U, dim1=-2, dim2=-1).sum(-1).reshape(-1) / 3.0
    n = comms.allreduce_sum(float(det.numel()))
    d = complex(comms.allreduce_sum(det.real.mean().item() * det.numel()) / n,
                comms.allreduce_sum(det.imag.mean().item() * det.numel()) / n)
    t = complex(comms.allreduce_sum(tr.real.sum().item()) / n,
                comms.allreduce_sum(tr.imag.sum().item()) / n)
    return d, t


def hyp_smear(u: torch.Tensor, geo: LatticeGeometry,
              alpha: tuple = (0.75, 0.6, 0.3), n_iter: int = 1
              ) -> torch.Tensor:
    """HYP (hypercubic) smearing, Hasenfratz-Knechtli 3-level restricted
    staples (ref: lib/gauge_hyp.cu / kernels/gauge_hyp.cuh — re-derived:
    level-1 decorated links Vbar_{mu;nu rho} use staples only in the
    remaining direction, level-2 Vtil_{mu;nu} from Vbar staples, level 3
    assembles the fat link; each level applies the COVARIANT polar SU(3)
    projection)."""
    from ..fields.gauge import project_su3_polar as project_su3
    from ..parallel.halo import shift_lex
    a1, a2, a3 = alpha

    def staple_dir(Ua, Ub, mu, nu):
        """staple of link field Ua (direction mu) through links Ub
        (direction nu): Ub(x) Ua(x+nu) Ub(x+mu)^dag + backward."""
        fwd = Ub @ shift_lex(Ua, geo, nu, +1) @ \
            shift_lex(Ub, geo, mu, +1).conj().mT
        Ub_m = shift_lex(Ub, geo, nu, -1)
        bwd = Ub_m.conj().mT @ shift_lex(Ua, geo, nu, -1) @ \
            shift_lex(Ub_m, geo, mu, +1)
        return fwd + bwd

    out = u
    for _ in range(n_iter):
        U = _to_lex(out, geo)
        # level 1: Vbar[mu][(nu,rho)] decorated in the single remaining dir
        Vbar = {}
        for mu in range(4):
            for nu in range(4):
                if nu == mu:
                    continue
                for rho in range(nu + 1, 4):
                    if rho == mu:
                        continue
                    eta = next(d for d in range(4)
                               if d not in (mu, nu, rho))
                    S = staple_dir(U[mu], U[eta], mu, eta)
                    Vbar[(mu, nu, rho)] = project_su3(
                        (1 - a3) * U[mu] + (a3 / 2.0) * S)
        # level 2: Vtil[mu][nu] from Vbar staples in the two remaining dirs
        Vtil = {}
        for mu in range(4):
            for nu in range(4):
                if nu == mu:
                    continue
                S = torch.zeros_like(U[mu])
                for rho in range(4):
                    if rho in (mu, nu):
                        continue
                    key_a = (mu,) + tuple(sorted((nu, rho)))
                    key_b = (rho,) + tuple(sorted((nu, mu)))
                    S = S + staple_dir(Vbar[key_a], Vbar[key_b], mu, rho)
                Vtil[(mu, nu)] = project_su3(
                    (1 - a2) * U[mu] + (a2 / 4.0) * S)
        # level 3: final fat link from Vtil staples in all three dirs
        Unew = torch.empty_like(U)
        for mu in range(4):
            S = torch.zeros_like(U[mu])
            for nu in range(4):
                if nu == mu:
                    continue
                S = S + staple_dir(Vtil[(mu, nu)], Vtil[(nu, mu)], mu, nu)
            Unew[mu] = project_su3((1 - a1) * U[mu] + (a1 / 6.0) * S)
        out = _from_lex(Unew, geo)
    return out


def over_improved_stout_smear(u: torch.Tensor, geo: LatticeGeometry,
                              rho: float, n_iter: int = 1,
                              epsilon: float = -0.25) -> torch.Tensor:
    """Over-improved stout smearing (Moran-Leinweber; ref: the
    over-improved branch of lib/gauge_stout.cu): the smearing direction
    is the traceless-antihermitian derivative of the action with
    plaquette weight (5 - 2 eps)/3 and rectangle weight -(1 - eps)/12;
    eps = 1 reduces EXACTLY to plain stout (rectangles drop out), the
    default eps = -0.25 is tuned to preserve instantons. Computed by
    reverse-mode differentiation of the loop traces (single-rank)."""
    from ..parallel import comms
    assert comms.comm_size() == 1, "autograd smear direction: single-rank"
    c0 = (5.0 - 2.0 * epsilon) / 3.0
    c1 = -(1.0 - epsilon) / 12.0
    out = u
    for _ in range(n_iter):
        u_req = out.detach().clone().requires_grad_(True)
        s = _action_autograd(u_req, geo, 1.0, c1, c0=c0)
        s.backward()
        g = u_req.grad
        Q = torch.empty_like(out)
        for mu in range(4):
            for p in (0, 1):
                # Q_stout = TA[S U^d] = +3 TA[U g^d] (from the validated
                # force relation F = 1/2 TA[U g^d] = -(1/6) TA[U S^d])
                Q[mu, p] = 3.0 * project_ta(out[mu, p]
                                            @ g[mu, p].conj().mT)
        U = _to_lex(out, geo)
        Ql = _to_lex(Q, geo)
        out = _from_lex(exp_su3(Ql, rho) @ U, geo)
    return out


def topological_charge_density(u: torch.Tensor, geo: LatticeGeometry
                               ) -> torch.Tensor:
    """Per-site topological charge density q(x) ([2, V_cb] real;
    the qChargeDensity output of gauge_qcharge.cuh): sums to
    topological_charge on this rank's sub-lattice."""
    import math
    F = field_strength(u, geo)

    def trprod(a, b):
        return torch.einsum("pvij,pvji->pv", F[a], F[b]).real

    q = (trprod((0, 1), (2, 3)) - trprod((0, 2), (1, 3))
         + trprod((0, 3), (1, 2)))
    return q / (4.0 * math.pi ** 2)
