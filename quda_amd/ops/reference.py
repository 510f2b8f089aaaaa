"""CPU/complex-tensor oracle ops (analogue of the reference's
tests/host_reference/, e.g. wilson_dslash_reference.cpp — reimplemented
from the operator definitions, not the reference code).

All functions work on the oracle layout:
  spinor  psi : [V_cb, 4, 3] complex (single parity) or [2, V_cb, 4, 3]
  gauge   u   : [4, 2, V_cb, 3, 3] complex
Works on CPU or GPU tensors (pure torch ops) at complex64/complex128.

Conventions (QUDA-compatible):
  Dslash:  D psi(x) = sum_mu [ U_mu(x) P(-mu) psi(x+mu)
                             + U_mu(x-mu)^dag P(+mu) psi(x-mu) ]
  with P(+-mu) = (1 -+ gamma_mu)/2 for dagger=False  (sign flips for dagger).
  Wilson operator (kappa normalization): M = 1 - kappa * D   (full lattice)
  Even-odd preconditioned: M_pc = 1 - kappa^2 * D_eo D_oe    (on even sites)
"""

from __future__ import annotations

import numpy as np
import torch

from ..fields.gamma import GAMMA, GAMMA5, projector, sigma_munu
from ..fields.geometry import LatticeGeometry


def _gamma_tensors(device, dtype):
    key = (device, dtype)
    cache = _gamma_tensors.__dict__.setdefault("cache", {})
    if key not in cache:
        P = np.stack([np.stack([projector(mu, -1), projector(mu, +1)])
                      for mu in range(4)])  # [mu, 0:minus/1:plus, 4, 4]
        cache[key] = torch.tensor(P, dtype=dtype, device=device)
    return cache[key]


def dslash_wilson_parity(u: torch.Tensor, psi: torch.Tensor,
                         geo: LatticeGeometry, parity: int,
                         dagger: bool = False, halo=None) -> torch.Tensor:
    """Apply the parity-hopping Wilson stencil: out_parity <- D psi_other.

    psi is the SOURCE (at parity 1-parity), result is at `parity`.
    `halo` (multi-rank / forced-partition): dict with 'mask' (bit per
    partitioned dim), 'psi' {(mu,dir): [Fcb,4,3]} ghost spinors in ghost
    order (dir=1 from +mu), 'u_bwd' {mu: [Fcb,3,3]} U_mu(x-mu) ghosts for
    the x_mu=0 face of this parity.
    """
    dev, dt = psi.device, psi.dtype
    P = _gamma_tensors(dev, dt)
    other = 1 - parity
    out = torch.zeros_like(psi)
    sgn = 1 if not dagger else 0  # index into P: dagger swaps minus<->plus
    mask = halo["mask"] if halo else 0
    for mu in range(4):
        part = (mask >> mu) & 1
        # forward: U_mu(x) P(-mu) psi(x+mu)
        fwd_idx = geo.neighbor_cb(parity, mu, +1).to(dev)
        psi_f = psi[fwd_idx]
        if part:
            fidx = geo.face_index_cb(parity, mu, geo.dims[mu] - 1).to(dev)
            psi_f[fidx] = halo["psi"][(mu, 1)].to(dt)
        proj = torch.einsum("st,vtc->vsc", P[mu, 1 - sgn], psi_f)
        out += torch.einsum("vij,vsj->vsi", u[mu, parity], proj)
        # backward: U_mu(x-mu)^dag P(+mu) psi(x-mu)
        bwd_idx = geo.neighbor_cb(parity, mu, -1).to(dev)
        psi_b = psi[bwd_idx]
        u_b = u[mu, other][bwd_idx]
        if part:
            fidx0 = geo.face_index_cb(parity, mu, 0).to(dev)
            psi_b[fidx0] = halo["psi"][(mu, 0)].to(dt)
            u_b = u_b.clone()
            u_b[fidx0] = halo["u_bwd"][mu].to(dt)
        proj = torch.einsum("st,vtc->vsc", P[mu, sgn], psi_b)
        out += torch.einsum("vji,vsj->vsi", u_b.conj(), proj)
    return out


def dslash_wilson_full(u, psi_full, geo, dagger: bool = False):
    """[2, V, 4, 3] -> [2, V, 4, 3]: out_p = D psi_{1-p} for both parities."""
    out = torch.empty_like(psi_full)
    for p in (0, 1):
        out[p] = dslash_wilson_parity(u, psi_full[1 - p], geo, p, dagger)
    return out


def mat_wilson(u, psi_full, geo, kappa: float, dagger: bool = False):
    """M psi = psi - kappa * D psi (full lattice, kappa normalization;
    ref semantics: dirac_wilson.cpp DiracWilson::M)."""
    return psi_full - kappa * dslash_wilson_full(u, psi_full, geo, dagger)


def apply_twist(psi: torch.Tensor, br: float, bi: float) -> torch.Tensor:
    """T(b) psi = br psi + i bi g5 psi (g5 = diag(1,1,-1,-1))."""
    g5 = torch.tensor([1.0, 1.0, -1.0, -1.0], dtype=psi.real.dtype,
                      device=psi.device)
    coef = br + 1j * bi * g5
    return psi * coef.view(*([1] * (psi.dim() - 2)), 4, 1)


def apply_gamma5(psi: torch.Tensor) -> torch.Tensor:
    g5 = torch.tensor(np.diag(GAMMA5).real, dtype=psi.real.dtype,
                      device=psi.device)
    return psi * g5.view(*([1] * (psi.dim() - 2)), 4, 1)


# ---------------------------------------------------------------------------
# Clover term (ref: lib/clover_quda.cu CloverCompute + dslash_wilson_clover)
# ---------------------------------------------------------------------------

def field_strength(u: torch.Tensor, geo: LatticeGeometry):
    """Clover-leaf F_munu (antihermitian traceless part).

    Returns dict {(mu,nu): [2, V_cb, 3, 3] complex} for mu<nu.
    F_munu = (1/8) sum_leaves (leaf - leaf^dag) (traceless antihermitian)
    (ref: kernels/field_strength_tensor.cuh — rebuilt from the standard
    4-leaf clover definition).
    """
    dev = u.device
    # full-lattice link array indexed lexicographically for easy shifting
    V = geo.volume
    U = torch.empty((4, V, 3, 3), dtype=u.dtype, device=dev)
    lo = geo.lex_of_cb.to(dev)
    for mu in range(4):
        U[mu, lo[0]] = u[mu, 0]
        U[mu, lo[1]] = u[mu, 1]

    def shift(f, mu, disp):
        """f: [V,3,3] field; returns f(x + disp*mu_hat) — neighbor-rank
        slabs on partitioned dims (the clover term is then consistent at
        rank boundaries)."""
        from ..parallel.halo import shift_lex
        return shift_lex(f, geo, mu, disp)

    out = {}
    for mu in range(4):
        for nu in range(mu + 1, 4):
            Umu, Unu = U[mu], U[nu]
            Umu_xnu = shift(Umu, nu, +1)   # U_mu(x+nu)
            Unu_xmu = shift(Unu, mu, +1)   # U_nu(x+mu)
            # leaf 1: U_mu(x) U_nu(x+mu) U_mu(x+nu)^d U_nu(x)^d
            P1 = Umu @ Unu_xmu @ Umu_xnu.conj().mT @ Unu.conj().mT
            # leaf 2: U_nu(x) U_mu(x-mu+nu)^d U_nu(x-mu)^d U_mu(x-mu)
            Umu_mx = shift(Umu, mu, -1)
            Unu_mx = shift(Unu, mu, -1)
            Umu_mxnu = shift(shift(Umu, mu, -1), nu, +1)
            P2 = Unu @ Umu_mxnu.conj().mT @ Unu_mx.conj().mT @ Umu_mx
            # leaf 3: U_mu(x-mu)^d U_nu(x-mu-nu)^d U_mu(x-mu-nu) U_nu(x-nu)
            Unu_mxmnu = shift(shift(Unu, mu, -1), nu, -1)
            Umu_mxmnu = shift(shift(Umu, mu, -1), nu, -1)
            Unu_mnu = shift(Unu, nu, -1)
            P3 = Umu_mx.conj().mT @ Unu_mxmnu.conj().mT @ Umu_mxmnu @ Unu_mnu
            # leaf 4: U_nu(x-nu)^d U_mu(x-nu) U_nu(x+mu-nu) U_mu(x)^d
            Umu_mnu = shift(Umu, nu, -1)
            Unu_xmu_mnu = shift(shift(Unu, nu, -1), mu, +1)
            P4 = Unu_mnu.conj().mT @ Umu_mnu @ Unu_xmu_mnu @ Umu.conj().mT
            Fsum = P1 + P2 + P3 + P4
            # hermitian field strength: F = (Q - Q^dag) / (8i)
            F = (Fsum - Fsum.conj().mT) / 8.0j
            # remove trace
            tr = torch.diagonal(F, dim1=-2, dim2=-1).sum(-1) / 3.0
            F = F - tr[..., None, None] * torch.eye(3, dtype=u.dtype, device=dev)
            # back to cb layout
            Fcb = torch.stack([F[lo[0]], F[lo[1]]])
            out[(mu, nu)] = Fcb
    return out


def clover_matrix(u: torch.Tensor, geo: LatticeGeometry, kappa: float,
                  csw: float) -> torch.Tensor:
    """Full clover site matrix A = 1 + (kappa*csw/2) * sigma_munu F_munu.

    Returns [2, V_cb, 12, 12] complex (spin x color flattened, s*3+c),
    hermitian. Block-diagonal in chirality in the DeGrand-Rossi basis.
    """
    F = field_strength(u, geo)
    dev, dt = u.device, u.dtype
    V = geo.volume_cb
    A = torch.zeros((2, V, 4, 3, 4, 3), dtype=dt, device=dev)
    eye = torch.eye(12, dtype=dt, device=dev).reshape(4, 3, 4, 3)
    A += eye
    coeff = kappa * csw / 2.0
    for (mu, nu), Fmn in F.items():
        sig = torch.tensor(sigma_munu(mu, nu), dtype=dt, device=dev)
        # sigma acts on spin, i*F on color; sum over mu<nu twice (munu + numu)
        # sigma_{nu mu} F_{nu mu} = sigma_{mu nu} F_{mu nu} so factor 2... but
        # standard convention sums mu<nu with both orderings equal:
        A += 2 * coeff * torch.einsum("st,pvij->pvsitj", sig, Fmn).reshape(2, V, 4, 3, 4, 3)
    return A.reshape(2, V, 12, 12)


def apply_clover(A: torch.Tensor, psi: torch.Tensor, parity=None) -> torch.Tensor:
    """A psi, site-local 12x12. psi [(.,)V,4,3]; A [2,V,12,12] or [V,12,12]."""
    if psi.dim() == 4:  # full field [2,V,4,3]
        out = torch.einsum("pvij,pvj->pvi", A, psi.reshape(2, -1, 12))
        return out.reshape(psi.shape)
    Ap = A if A.dim() == 3 else A[parity]
    out = torch.einsum("vij,vj->vi", Ap, psi.reshape(-1, 12))
    return out.reshape(psi.shape)


# ---------------------------------------------------------------------------
# Gauge observables
# ---------------------------------------------------------------------------

def plaquette(u: torch.Tensor, geo: LatticeGeometry):
    """Mean plaquette Re tr P / 3: returns (total, spatial, temporal)
    (ref: lib/gauge_plaq.cu semantics)."""
    dev = u.device
    V = geo.volume
    U = torch.empty((4, V, 3, 3), dtype=u.dtype, device=dev)
    lo = geo.lex_of_cb.to(dev)
    for mu in range(4):
        U[mu, lo[0]] = u[mu, 0]
        U[mu, lo[1]] = u[mu, 1]
    sp, tp = [], []
    for mu in range(4):
        for nu in range(mu + 1, 4):
            Unu_xmu = U[nu][geo.neighbor_lex(mu, +1).to(dev)]
            Umu_xnu = U[mu][geo.neighbor_lex(nu, +1).to(dev)]
            P = U[mu] @ Unu_xmu @ Umu_xnu.conj().mT @ U[nu].conj().mT
            val = torch.diagonal(P, dim1=-2, dim2=-1).sum(-1).real.mean().item() / 3.0
            (tp if nu == 3 else sp).append(val)
    s = sum(sp) / len(sp)
    t = sum(tp) / len(tp)
    return (s + t) / 2, s, t


# ---------------------------------------------------------------------------
# Staggered (Kogut-Susskind) oracle (ref: tests/host_reference/
# staggered_dslash_reference.cpp — re-derived from the operator definition)
# ---------------------------------------------------------------------------

def staggered_phases(geo: LatticeGeometry, parity: int) -> torch.Tensor:
    """[V_cb, 4] +-1 eta_mu(x) = (-1)^(x_0+..+x_{mu-1}) for parity sites."""
    c = geo.coords_of_cb(parity).to(torch.int64)
    pref = torch.zeros_like(c)
    pref[:, 1] = c[:, 0]
    pref[:, 2] = c[:, 0] + c[:, 1]
    pref[:, 3] = c[:, 0] + c[:, 1] + c[:, 2]
    return torch.where(pref % 2 == 0, 1.0, -1.0).to(torch.float64)


def dslash_staggered_parity(u: torch.Tensor, psi: torch.Tensor,
                            geo: LatticeGeometry, parity: int,
                            halo=None) -> torch.Tensor:
    """out(parity) = sum_mu eta_mu(x)[U_mu(x) psi(x+mu)
                                      - U_mu(x-mu)^dag psi(x-mu)].

    psi: [V_cb, 3] complex at parity 1-parity. halo as in
    dslash_wilson_parity but with [Fcb, 3] spinor ghosts."""
    dev, dt = psi.device, psi.dtype
    other = 1 - parity
    out = torch.zeros_like(psi)
    eta = staggered_phases(geo, parity).to(dev)
    mask = halo["mask"] if halo else 0
    for mu in range(4):
        part = (mask >> mu) & 1
        fwd_idx = geo.neighbor_cb(parity, mu, +1).to(dev)
        psi_f = psi[fwd_idx]
        if part:
            fidx = geo.face_index_cb(parity, mu, geo.dims[mu] - 1).to(dev)
            psi_f[fidx] = halo["psi"][(mu, 1)].to(dt)
        e = eta[:, mu].to(dt).unsqueeze(-1)
        out += e * torch.einsum("vij,vj->vi", u[mu, parity], psi_f)
        bwd_idx = geo.neighbor_cb(parity, mu, -1).to(dev)
        psi_b = psi[bwd_idx]
        u_b = u[mu, other][bwd_idx]
        if part:
            fidx0 = geo.face_index_cb(parity, mu, 0).to(dev)
            psi_b[fidx0] = halo["psi"][(mu, 0)].to(dt)
            u_b = u_b.clone()
            u_b[fidx0] = halo["u_bwd"][mu].to(dt)
        out -= e * torch.einsum("vji,vj->vi", u_b.conj(), psi_b)
    return out


def dslash_staggered_full(u, psi_full, geo):
    out = torch.empty_like(psi_full)
    for p in (0, 1):
        out[p] = dslash_staggered_parity(u, psi_full[1 - p], geo, p)
    return out


def mat_staggered(u, psi_full, geo, mass: float):
    """M psi = 2m psi + D psi (mass normalization; D antihermitian)."""
    return 2.0 * mass * psi_full + dslash_staggered_full(u, psi_full, geo)


# ---------------------------------------------------------------------------
# Domain-wall / Moebius 5th-dimension oracle (ref: tests/host_reference/
# domain_wall_dslash_reference.cpp — re-derived; P+ = spins 0,1 and
# P- = spins 2,3 in DeGrand-Rossi, so s-hops are spin-diagonal)
# ---------------------------------------------------------------------------

def dslash5(psi5: torch.Tensor, Ls: int, alpha: float, beta: float,
            mf: float, dagger: bool = False) -> torch.Tensor:
    """[Ls*V, 4, 3] -> alpha psi + beta (Ds psi); Ds hops upper spins from
    s-1 and lower from s+1 (swapped for dagger), with -mf boundary wraps."""
    V = psi5.shape[0] // Ls
    v = psi5.reshape(Ls, V, 4, 3)
    if not dagger:
        up_from = torch.roll(v, shifts=1, dims=0).clone()   # s-1 -> s
        up_from[0] = -mf * v[Ls - 1]
        dn_from = torch.roll(v, shifts=-1, dims=0).clone()  # s+1 -> s
        dn_from[Ls - 1] = -mf * v[0]
    else:
        up_from = torch.roll(v, shifts=-1, dims=0).clone()
        up_from[Ls - 1] = -mf * v[0]
        dn_from = torch.roll(v, shifts=1, dims=0).clone()
        dn_from[0] = -mf * v[Ls - 1]
    out = alpha * v.clone()
    out[:, :, 0:2, :] += beta * up_from[:, :, 0:2, :]
    out[:, :, 2:4, :] += beta * dn_from[:, :, 2:4, :]
    return out.reshape(Ls * V, 4, 3)


def _m5_matrix(Ls, alpha, beta, mf, upper: bool, dagger: bool):
    """Dense [Ls, Ls] chirality-block matrix of alpha + beta Ds."""
    import numpy as np
    A = alpha * np.eye(Ls)
    for s in range(Ls):
        src = s - 1 if (upper != dagger) else s + 1
        w = 1.0
        if src < 0:
            src += Ls
            w = -mf
        if src >= Ls:
            src -= Ls
            w = -mf
        A[s, src] += beta * w
    return A


def m5inv(psi5: torch.Tensor, Ls: int, alpha: float, beta: float, mf: float,
          dagger: bool = False) -> torch.Tensor:
    """(alpha + beta Ds)^{-1} psi via dense per-chirality [Ls,Ls] solves."""
    import numpy as np
    V = psi5.shape[0] // Ls
    v = psi5.reshape(Ls, V, 4, 3)
    out = torch.empty_like(v)
    for upper, sl in ((True, slice(0, 2)), (False, slice(2, 4))):
        A = _m5_matrix(Ls, alpha, beta, mf, upper, dagger)
        Ainv = torch.tensor(np.linalg.inv(A), dtype=psi5.dtype,
                            device=psi5.device)
        out[:, :, sl, :] = torch.einsum("st,tvxc->svxc", Ainv, v[:, :, sl, :])
    return out.reshape(Ls * V, 4, 3)


def dslash_staggered_naik_parity(n: torch.Tensor, psi: torch.Tensor,
                                 geo: LatticeGeometry, parity: int,
                                 halo=None) -> torch.Tensor:
    """Naik 3-hop term: sum_mu eta_mu(x)[N_mu(x) psi(x+3mu)
    - N_mu(x-3mu)^dag psi(x-3mu)]. n: long links [4,2,V,3,3].

    halo (partitioned dims): {"mask": int, "psi3": depth-3 ghosts from
    exchange_psi_oracle(..., depth=3), "n_bwd": {(mu, l): [Fcb,3,3]}
    long links N_mu(x-3mu) for the x_mu = l face (from
    GaugeField.bwd_ghost of the shift-3 stencil field)}."""
    dev, dt = psi.device, psi.dtype
    other = 1 - parity
    out = torch.zeros_like(psi)
    eta = staggered_phases(geo, parity).to(dev)
    mask = halo["mask"] if halo else 0

    def nbr3(p, mu, disp):
        c = geo.coords_of_cb(p).to(torch.int64).clone()
        c[:, mu] = (c[:, mu] + disp) % geo.dims[mu]
        X, Y, Z, _ = geo.dims
        lex = ((c[:, 3] * Z + c[:, 2]) * Y + c[:, 1]) * X + c[:, 0]
        return geo.cb_of_lex[lex].to(dev)

    for mu in range(4):
        part = (mask >> mu) & 1
        Xm = geo.dims[mu]
        e = eta[:, mu].to(dt).unsqueeze(-1)
        fwd = nbr3(parity, mu, +3)
        psi_f = psi[fwd]
        if part:
            for l in range(3):  # sites x_mu = Xm-3+l read fwd ghost layer l
                fidx = geo.face_index_cb(parity, mu, Xm - 3 + l).to(dev)
                psi_f[fidx] = halo["psi3"][(mu, 1)][l].to(dt)
        out += e * torch.einsum("vij,vj->vi", n[mu, parity], psi_f)
        bwd = nbr3(parity, mu, -3)
        psi_b = psi[bwd]
        n_b = n[mu, other][bwd]
        if part:
            n_b = n_b.clone()
            for l in range(3):  # sites x_mu = 2-l read bwd ghost layer l
                fidx = geo.face_index_cb(parity, mu, 2 - l).to(dev)
                psi_b[fidx] = halo["psi3"][(mu, 0)][l].to(dt)
                n_b[fidx] = halo["n_bwd"][(mu, 2 - l)].to(dt)
        out -= e * torch.einsum("vji,vj->vi", n_b.conj(), psi_b)
    return out


def _zm5_matrix(Ls, diag, hop, mf, upper: bool, dagger: bool):
    """Dense [Ls,Ls] complex chirality-block matrix of the zMobius
    s-operator diag[s] + hop[s]*Ds (per-slice complex coefficients;
    dagger = conjugate transpose of the non-dagger matrix)."""
    import numpy as np
    A = np.diag(np.asarray(diag, dtype=complex))
    for s in range(Ls):
        src = s - 1 if upper else s + 1
        w = 1.0
        if src < 0:
            src += Ls
            w = -mf
        if src >= Ls:
            src -= Ls
            w = -mf
        A[s, src] += complex(hop[s]) * w
    return A.conj().T if dagger else A


def zdslash5(psi5: torch.Tensor, Ls: int, diag, hop, mf: float,
             dagger: bool = False) -> torch.Tensor:
    """zMobius s-operator apply via dense chirality-block matrices."""
    V = psi5.shape[0] // Ls
    v = psi5.reshape(Ls, V, 4, 3)
    out = torch.empty_like(v)
    for upper, sl in ((True, slice(0, 2)), (False, slice(2, 4))):
        A = torch.tensor(_zm5_matrix(Ls, diag, hop, mf, upper, dagger),
                         dtype=psi5.dtype, device=psi5.device)
        out[:, :, sl, :] = torch.einsum("st,tvxc->svxc", A, v[:, :, sl, :])
    return out.reshape(Ls * V, 4, 3)


def zm5inv(psi5: torch.Tensor, Ls: int, diag, hop, mf: float,
           dagger: bool = False) -> torch.Tensor:
    """Inverse of the zMobius s-operator via dense solves."""
    import numpy as np
    V = psi5.shape[0] // Ls
    v = psi5.reshape(Ls, V, 4, 3)
    out = torch.empty_like(v)
    for upper, sl in ((True, slice(0, 2)), (False, slice(2, 4))):
        A = _zm5_matrix(Ls, diag, hop, mf, upper, dagger)
        Ainv = torch.tensor(np.linalg.inv(A), dtype=psi5.dtype,
                            device=psi5.device)
        out[:, :, sl, :] = torch.einsum("st,tvxc->svxc", Ainv, v[:, :, sl, :])
    return out.reshape(Ls * V, 4, 3)


def _m5_eofa_matrix(Ls, alpha, beta, mf, sh, pm, u, w, upper: bool,
                    dagger: bool):
    """Dense chirality block of the EOFA M5: base Moebius + rank-1
    sh |u><w| on the pm chirality (pm=+1 -> upper block)."""
    import numpy as np
    A = _m5_matrix(Ls, alpha, beta, mf, upper, dagger)
    on = (pm > 0) == upper
    if on:
        uu = np.asarray(u, dtype=float)
        ww = np.asarray(w, dtype=float)
        R1 = sh * np.outer(uu, ww)
        A = A + (R1.conj().T if dagger else R1)
    return A


def m5_eofa(psi5, Ls, alpha, beta, mf, sh, pm, u, w, dagger=False):
    V = psi5.shape[0] // Ls
    v = psi5.reshape(Ls, V, 4, 3)
    out = torch.empty_like(v)
    for upper, sl in ((True, slice(0, 2)), (False, slice(2, 4))):
        A = torch.tensor(_m5_eofa_matrix(Ls, alpha, beta, mf, sh, pm, u, w,
                                         upper, dagger),
                         dtype=psi5.dtype, device=psi5.device)
        out[:, :, sl, :] = torch.einsum("st,tvxc->svxc", A, v[:, :, sl, :])
    return out.reshape(Ls * V, 4, 3)


def m5inv_eofa(psi5, Ls, alpha, beta, mf, sh, pm, u, w, dagger=False):
    import numpy as np
    V = psi5.shape[0] // Ls
    v = psi5.reshape(Ls, V, 4, 3)
    out = torch.empty_like(v)
    for upper, sl in ((True, slice(0, 2)), (False, slice(2, 4))):
        A = _m5_eofa_matrix(Ls, alpha, beta, mf, sh, pm, u, w, upper, dagger)
        Ainv = torch.tensor(np.linalg.inv(A), dtype=psi5.dtype,
                            device=psi5.device)
        out[:, :, sl, :] = torch.einsum("st,tvxc->svxc", Ainv, v[:, :, sl, :])
    return out.reshape(Ls * V, 4, 3)
