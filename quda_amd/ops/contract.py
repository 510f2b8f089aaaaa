"""Contractions, stochastic sources and LAPH-style projections
(ref: lib/contract.cu + kernels/contraction.cuh contractQuda/contractFTQuda,
lib/evec_project.cu, kernels/spinor_noise.cuh / spinor_dilute.cuh —
re-derived; einsum/batched-GEMM shaped, torch dispatches to rocBLAS)."""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from ..fields.gamma import GAMMA, GAMMA5
from ..fields.geometry import LatticeGeometry, checkerboard_join
from ..fields.spinor import SpinorField


def contract_open_spin(x: SpinorField, y: SpinorField) -> torch.Tensor:
    """Open-spin color-contracted bilinear per site:
    C[x, s, s'] = sum_c conj(x(s,c)) y(s',c)  ->  [V, 4, 4] (lex order)
    (ref: contractQuda QUDA_CONTRACT_TYPE_OPEN)."""
    geo = x.geo
    xc = checkerboard_join(x.to_complex(), geo)
    yc = checkerboard_join(y.to_complex(), geo)
    return torch.einsum("vsc,vtc->vst", xc.conj(), yc)


def contract_dr(x: SpinorField, y: SpinorField) -> torch.Tensor:
    """Gamma-insertion contraction: tr_spin[ G_i^dag X^dag G_i Y ] summed
    over color, for the 16 DeGrand-Rossi gamma-basis elements
    -> [V, 16] (ref: QUDA_CONTRACT_TYPE_DR)."""
    C = contract_open_spin(x, y)  # [V,4,4]
    dev, dt = C.device, C.dtype
    gammas = _gamma_basis(dev, dt)
    # tr[ G^d C^T_spin? ] — standard: sum_{s,s'} G[s',s] C[s,s']
    return torch.einsum("gts,vst->vg", gammas, C)


def _gamma_basis(device, dtype):
    """16-element gamma basis: 1, g1..g4, g5, g5 g_mu, sigma_munu-like
    products (ordered: I, g1, g2, g3, g4, g5, g5g1..g5g4, g1g2,...)."""
    mats = [np.eye(4)]
    mats += [GAMMA[mu] for mu in range(4)]
    mats += [GAMMA5]
    mats += [GAMMA5 @ GAMMA[mu] for mu in range(4)]
    for mu in range(4):
        for nu in range(mu + 1, 4):
            mats.append(GAMMA[mu] @ GAMMA[nu])
    return torch.tensor(np.stack(mats), device=device, dtype=dtype)


def contract_ft(x: SpinorField, y: SpinorField, momenta: List[tuple],
                reduct_dim: int = 3) -> torch.Tensor:
    """Momentum-projected timeslice contraction (ref: contractFTQuda):
    C[p, t, g] = sum_{vec x} e^{-i p.x} tr[...] -> [n_mom, T, 16]."""
    geo = x.geo
    c = contract_dr(x, y)  # [V, 16] lex
    coords = geo.coords.to(torch.float64)
    dims = geo.dims
    out = torch.empty((len(momenta), dims[reduct_dim], 16),
                      dtype=c.dtype, device=c.device)
    tcoord = geo.coords[:, reduct_dim].to(torch.int64)
    for ip, p in enumerate(momenta):
        phase_arg = sum(2 * np.pi * p[i] * coords[:, i] / dims[i]
                        for i in range(4) if i != reduct_dim)
        ph = torch.exp(-1j * phase_arg).to(c.dtype)
        w = c * ph.unsqueeze(-1)
        slab = torch.zeros((dims[reduct_dim], 16), dtype=c.dtype,
                           device=c.device)
        slab.index_add_(0, tcoord, w)
        out[ip] = slab
    from ..parallel.comms import allreduce_tensor
    return allreduce_tensor(out)


def evec_project(evecs: List[SpinorField], psi: SpinorField,
                 reduct_dim: int = 3) -> torch.Tensor:
    """LapH sink projection <evec_i | psi> per timeslice and spin
    -> [n_ev, T, 4] (ref: lib/evec_project.cu; evecs are nspin=1 laplace
    eigenvectors, psi a 4-spin propagator field)."""
    geo = psi.geo
    T = geo.dims[reduct_dim]
    tcoord = geo.coords[:, reduct_dim].to(torch.int64)
    pc = checkerboard_join(psi.to_complex(), geo)  # [V,4,3]
    out = torch.empty((len(evecs), T, 4), dtype=pc.dtype, device=pc.device)
    for i, v in enumerate(evecs):
        vc = checkerboard_join(v.to_complex(), geo)  # [V,3]
        w = torch.einsum("vc,vsc->vs", vc.conj(), pc)
        slab = torch.zeros((T, 4), dtype=pc.dtype, device=pc.device)
        slab.index_add_(0, tcoord.to(pc.device), w)
        out[i] = slab
    from ..parallel.comms import allreduce_tensor
    return allreduce_tensor(out)


# -- stochastic sources (ref: spinor_noise.cuh / spinor_dilute.cuh) ---------

def z4_noise(f: SpinorField, seed: int) -> SpinorField:
    """Z4 noise {±1, ±i}/1 per component."""
    g = torch.Generator().manual_seed(seed)
    k = torch.randint(0, 4, (f.n_parity, f.volume_cb, *f.site_shape),
                      generator=g)
    vals = torch.tensor([1 + 0j, -1 + 0j, 1j, -1j], dtype=torch.complex128)
    f.from_complex(vals[k].to(f.device))
    return f


def gaussian_noise(f: SpinorField, seed: int) -> SpinorField:
    return f.gaussian_(seed=seed)


def dilute(src: SpinorField, scheme: str = "spin") -> List[SpinorField]:
    """Split a source into orthogonal dilution components
    (ref: spinor_dilute.cuh; schemes: spin, color, even-odd,
    time-slice)."""
    geo = src.geo
    c = src.to_complex()
    out = []
    if scheme == "spin":
        for s in range(src.nspin):
            f = src.clone_empty()
            cc = torch.zeros_like(c)
            if src.nspin > 1:
                cc[:, :, s, :] = c[:, :, s, :]
            else:
                cc = c.clone()
            f.from_complex(cc)
            out.append(f)
    elif scheme == "color":
        for col in range(3):
            f = src.clone_empty()
            cc = torch.zeros_like(c)
            cc[..., col] = c[..., col]
            f.from_complex(cc)
            out.append(f)
    elif scheme == "even-odd":
        assert src.n_parity == 2
        for p in (0, 1):
            f = src.clone_empty()
            cc = torch.zeros_like(c)
            cc[p] = c[p]
            f.from_complex(cc)
            out.append(f)
    elif scheme == "time":
        tcoord = geo.coords[:, 3].to(torch.int64)
        from ..fields.geometry import checkerboard_split
        tc = torch.stack([tcoord[geo.lex_of_cb[0]],
                          tcoord[geo.lex_of_cb[1]]])  # [2, Vcb]
        if src.ls > 1:
            tc = tc.repeat(1, src.ls)
        for t in range(geo.dims[3]):
            f = src.clone_empty()
            cc = torch.zeros_like(c)
            m = (tc == t)
            cc[m] = c[m]
            f.from_complex(cc)
            out.append(f)
    else:
        raise ValueError(scheme)
    return out


def sequential_source(prop_col: SpinorField, t_sink: int,
                      gamma: str = "g5") -> SpinorField:
    """Sequential source for 3-point functions (ref: the sequential-
    propagator workflow the reference's covdev/contract kernels serve):
    restrict one propagator column to the sink timeslice and hit it with
    the sink gamma (g5 for the pion)."""
    import torch
    geo = prop_col.geo
    c = prop_col.to_complex().clone()
    tcoord = geo.coords[:, 3].to(torch.int64)
    tc = torch.stack([tcoord[geo.lex_of_cb[0]], tcoord[geo.lex_of_cb[1]]])
    mask = (tc == t_sink).unsqueeze(-1).unsqueeze(-1)
    c = torch.where(mask, c, torch.zeros_like(c))
    if gamma == "g5":
        c[..., 2:4, :] = -c[..., 2:4, :]
    else:
        raise ValueError(gamma)
    out = prop_col.clone_empty()
    out.from_complex(c)
    return out
