"""Kahler-Dirac block preconditioning for staggered fermions
(ref: lib/dirac_staggered_kd.cpp, lib/dirac_improved_staggered_kd.cpp,
lib/staggered_kd_build_xinv.cu — re-designed: the 2^4-hypercube block of
the staggered operator is a [48 x 48] matrix per block (16 corners x 3
colors); we build X for all blocks as one batched tensor, invert with
torch.linalg.inv (batched LU -> rocBLAS/hipSOLVER on device, the role of
the reference's BatchInvertMatrix), and apply X^-1 as a block-diagonal
batched matmul. No hand kernel needed: the apply IS a batched GEMM.)

The KD-preconditioned operator is X^-1 M (left preconditioning); at
small mass X captures the intra-block hopping exactly, dramatically
improving the conditioning of the staggered normal equations.
"""

from __future__ import annotations

import torch

from ..fields.gauge import GaugeField
from ..fields.geometry import LatticeGeometry
from ..fields.spinor import SpinorField
from ..ops import blas
from .staggered import DiracImprovedStaggered, DiracStaggered


def _block_maps(geo: LatticeGeometry, device):
    """(perm, inv_perm): lex site index <-> (block, corner) flattening.
    perm[blk*16 + corner] = lex index; corner = c0%2 + 2*(c1%2) + ..."""
    c = geo.coords.to(torch.int64)
    X, Y, Z, T = geo.dims
    bx, by, bz, bt = X // 2, Y // 2, Z // 2, T // 2
    blk = ((c[:, 3] // 2 * bz + c[:, 2] // 2) * by + c[:, 1] // 2) * bx \
        + c[:, 0] // 2
    corner = (c[:, 0] % 2) + 2 * (c[:, 1] % 2) + 4 * (c[:, 2] % 2) \
        + 8 * (c[:, 3] % 2)
    key = blk * 16 + corner
    perm = torch.argsort(key)
    return perm.to(device), key.to(device)


class KDBlockInverse:
    """Batched X^-1 of the 2^4-block staggered operator (mass included):
    X = 2m + (intra-block hops of D). `u` is the (fat) link field
    [4,2,Vcb,3,3] complex."""

    def __init__(self, u: torch.Tensor, geo: LatticeGeometry, mass: float):
        for d in geo.dims:
            assert d % 2 == 0
        self.geo = geo
        self.mass = float(mass)
        dev = u.device
        dt = u.dtype
        V = geo.volume
        nblk = V // 16
        self.perm, _ = _block_maps(geo, dev)

        # lex-ordered links [4, V, 3, 3]
        lo = geo.lex_of_cb
        u_lex = torch.empty((4, V, 3, 3), dtype=dt, device=dev)
        for mu in range(4):
            u_lex[mu, lo[0].to(dev)] = u[mu, 0]
            u_lex[mu, lo[1].to(dev)] = u[mu, 1]
        c = geo.coords.to(torch.int64).to(dev)
        pref = torch.zeros_like(c)
        pref[:, 1] = c[:, 0]
        pref[:, 2] = c[:, 0] + c[:, 1]
        pref[:, 3] = c[:, 0] + c[:, 1] + c[:, 2]
        eta = torch.where(pref % 2 == 0, 1.0, -1.0).to(dt)  # [V,4]

        Xb = torch.zeros((nblk, 48, 48), dtype=dt, device=dev)
        idx = torch.eye(48, dtype=dt, device=dev)
        Xb += (2.0 * self.mass) * idx

        # site -> (blk, corner) of every lex site
        _, key = _block_maps(geo, dev)
        blk_of = key // 16
        cor_of = key % 16
        for mu in range(4):
            low = (c[:, mu] % 2 == 0)  # sites whose +mu hop stays in-block
            src = torch.nonzero(low, as_tuple=True)[0]
            # +mu neighbor lex index
            cc = c[src].clone()
            cc[:, mu] += 1
            X_, Y_, Z_, _ = geo.dims
            nl = ((cc[:, 3] * Z_ + cc[:, 2]) * Y_ + cc[:, 1]) * X_ + cc[:, 0]
            b = blk_of[src]
            s_cor = cor_of[src]
            d_cor = cor_of[nl]
            e = eta[src, mu]
            U = u_lex[mu, src]  # [n,3,3]
            # forward: out(dst) += eta(src_site? no: eta of the OUT site)
            # D psi(x) = eta_mu(x)[U_mu(x) psi(x+mu) - U_mu(x-mu)^dag psi(x-mu)]
            # row = out site, col = in site.
            # row dst (x+mu): backward hop term  -eta_mu(x+mu) U_mu(x)^dag
            # row src (x):    forward  hop term  +eta_mu(x)   U_mu(x)
            e_dst = eta[nl, mu]
            rows = (s_cor * 3).unsqueeze(-1) + torch.arange(3, device=dev)
            cols = (d_cor * 3).unsqueeze(-1) + torch.arange(3, device=dev)
            # scatter 3x3 tiles: X[b, src_rows, dst_cols] += e * U
            bi = b.view(-1, 1, 1).expand(-1, 3, 3)
            ri = rows.unsqueeze(-1).expand(-1, 3, 3)
            ci = cols.unsqueeze(-2).expand(-1, 3, 3)
            Xb.index_put_((bi, ri, ci), e.view(-1, 1, 1) * U,
                          accumulate=True)
            # dagger tile: X[b, dst_rows, src_cols] += -e_dst * U^dag
            rdi = cols.unsqueeze(-1).expand(-1, 3, 3)
            cdi = rows.unsqueeze(-2).expand(-1, 3, 3)
            Xb.index_put_((bi, rdi, cdi),
                          (-e_dst).view(-1, 1, 1) * U.conj().mT,
                          accumulate=True)
        self.Xinv = torch.linalg.inv(Xb)  # [nblk,48,48] batched LU
        self.X = Xb

    def _to_blocks(self, psi_cb: torch.Tensor) -> torch.Tensor:
        """[2,Vcb,3] cb complex -> [nblk, 48]"""
        geo = self.geo
        V = geo.volume
        lo = geo.lex_of_cb
        lex = torch.empty((V, 3), dtype=psi_cb.dtype, device=psi_cb.device)
        lex[lo[0].to(psi_cb.device)] = psi_cb[0]
        lex[lo[1].to(psi_cb.device)] = psi_cb[1]
        return lex[self.perm].reshape(-1, 48)

    def _from_blocks(self, b: torch.Tensor) -> torch.Tensor:
        geo = self.geo
        V = geo.volume
        lex = torch.empty((V, 3), dtype=b.dtype, device=b.device)
        lex[self.perm] = b.reshape(-1, 3)
        lo = geo.lex_of_cb
        out = torch.stack([lex[lo[0].to(b.device)], lex[lo[1].to(b.device)]])
        return out

    def apply(self, out: SpinorField, inp: SpinorField,
              dagger: bool = False) -> SpinorField:
        v = self._to_blocks(inp.to_complex())
        M = self.Xinv.conj().mT if dagger else self.Xinv
        r = torch.einsum("bij,bj->bi", M, v)
        out.from_complex(self._from_blocks(r))
        return out

    def apply_X(self, out: SpinorField, inp: SpinorField,
                dagger: bool = False) -> SpinorField:
        v = self._to_blocks(inp.to_complex())
        M = self.X.conj().mT if dagger else self.X
        r = torch.einsum("bij,bj->bi", M, v)
        out.from_complex(self._from_blocks(r))
        return out


class _KDPrecondMixin:
    """M_kd = X^-1 M_plain; prepare maps b -> X^-1 b, the solution is
    unchanged (left preconditioning)."""

    def M(self, out: SpinorField, inp: SpinorField, dagger: bool = False):
        t = self._tmp_kd(inp)
        if not dagger:
            self._plain_M(t, inp, dagger=False)
            self.kd.apply(out, t, dagger=False)
        else:
            # (X^-1 M)^dag = M^dag X^-dag
            self.kd.apply(t, inp, dagger=True)
            self._plain_M(out, t, dagger=True)
        return out

    def _tmp_kd(self, like: SpinorField) -> SpinorField:
        cache = self.__dict__.setdefault("_kd_tmps", {})
        key = (like.precision, str(like.device))
        t = cache.get(key)
        if t is None:
            t = SpinorField(self.geo, like.precision, like.device,
                            like.n_parity, nspin=1)
            cache[key] = t
        return t

    def prepare(self, b: SpinorField) -> SpinorField:
        bp = SpinorField(self.geo, b.precision, b.device, b.n_parity,
                         nspin=1)
        return self.kd.apply(bp, b)

    def reconstruct(self, x_full: SpinorField, x: SpinorField,
                    b: SpinorField) -> SpinorField:
        blas.copy(x_full, x)
        return x_full


class DiracStaggeredKD(_KDPrecondMixin, DiracStaggered):
    """KD-preconditioned staggered operator (ref: DiracStaggeredKD
    dirac.h:1533). Solve M_kd x = X^-1 b with a nonsymmetric solver."""

    def __init__(self, gauge: GaugeField, mass: float):
        DiracStaggered.__init__(self, gauge, mass)
        self.kd = KDBlockInverse(gauge.to_complex(), gauge.geo, mass)

    def _plain_M(self, out, inp, dagger=False):
        return DiracStaggered.M(self, out, inp, dagger)


class DiracImprovedStaggeredKD(_KDPrecondMixin, DiracImprovedStaggered):
    """KD-preconditioned improved staggered: X built from the FAT links
    only (the Naik term is long-range and excluded from the block, ref
    dirac_improved_staggered_kd.cpp)."""

    def __init__(self, fat: GaugeField, lng: GaugeField, mass: float):
        DiracImprovedStaggered.__init__(self, fat, lng, mass)
        self.kd = KDBlockInverse(fat.to_complex(), fat.geo, mass)

    def _plain_M(self, out, inp, dagger=False):
        return DiracImprovedStaggered.M(self, out, inp, dagger)
