"""Dirac gamma matrices (DeGrand-Rossi basis) + spin-projector structure.

The reference's site-level spinor math lives in include/color_spinor.h:
project(dim,sign)/reconstruct and include/gamma.cuh (basis tables). Here the
basis is defined ONCE as explicit 4x4 matrices; the CPU oracle consumes them
via einsum, and csrc/generate_proj.py derives the half-spinor projection /
reconstruction coefficient tables from them to emit the HIP kernel header —
no hand-transcribed spin algebra anywhere.

Basis: DeGrand-Rossi (the reference's native QUDA_DEGRAND_ROSSI_GAMMA_BASIS),
in which all gamma_mu are purely off-diagonal with entries in {+-1, +-i} and
gamma5 = gamma1*gamma2*gamma3*gamma4 is diagonal.
"""

from __future__ import annotations

import numpy as np

I = 1j

# gamma_mu, mu = 0..3 == (x, y, z, t)
GAMMA = [
    np.array([[0, 0, 0, I],
              [0, 0, I, 0],
              [0, -I, 0, 0],
              [-I, 0, 0, 0]], dtype=np.complex128),
    np.array([[0, 0, 0, -1],
              [0, 0, 1, 0],
              [0, 1, 0, 0],
              [-1, 0, 0, 0]], dtype=np.complex128),
    np.array([[0, 0, I, 0],
              [0, 0, 0, -I],
              [-I, 0, 0, 0],
              [0, I, 0, 0]], dtype=np.complex128),
    np.array([[0, 0, 1, 0],
              [0, 0, 0, 1],
              [1, 0, 0, 0],
              [0, 1, 0, 0]], dtype=np.complex128),
]

GAMMA5 = (GAMMA[0] @ GAMMA[1] @ GAMMA[2] @ GAMMA[3]).round().astype(np.complex128)

ID4 = np.eye(4, dtype=np.complex128)


def projector(mu: int, sign: int) -> np.ndarray:
    """P = (1 + sign*gamma_mu)/2, sign in {+1,-1}. Rank 2."""
    return (ID4 + sign * GAMMA[mu]) / 2


def sigma_munu(mu: int, nu: int) -> np.ndarray:
    """sigma_{mu nu} = (i/2)[gamma_mu, gamma_nu] (used by the clover term)."""
    return 0.5j * (GAMMA[mu] @ GAMMA[nu] - GAMMA[nu] @ GAMMA[mu])


def _as_unit(c: complex) -> complex:
    """Round a coefficient to the nearest element of {0, +-1, +-i, +-1/2...}."""
    return complex(np.round(c.real, 12) + 1j * np.round(c.imag, 12))


def half_projector_structure(mu: int, sign: int):
    """Decompose P = (1 + sign*gamma_mu)/2 into half-spinor form.

    Returns (proj, recon) where:
      proj  : [2][4] complex — h_s = sum_t proj[s][t] * psi_t  (s = 0,1);
              exactly two entries per row are nonzero, values in {+-1, +-i}
              (the overall 1/2 is absorbed: P psi = 1/2 * recon(h)).
      recon : [4][2] complex — (2 P psi)_r = sum_s recon[r][s] * h_s, with
              recon[0] = (1,0), recon[1] = (0,1) and rows 2,3 a single
              unit-phase entry each.
    The Wilson stencil applies U * h (2 columns instead of 4), halving both
    flops and the halo payload — same structure the reference exploits
    (include/kernels/dslash_wilson.cuh:84 applyWilson).
    """
    P = projector(mu, sign)
    # upper 2x4 block of 2P defines the half spinor
    proj = (2 * P[0:2, :]).copy()
    # rows 2,3 of 2P are linear combinations of rows 0,1 of 2P:
    # solve recon[r] s.t. (2P)[r] = recon[r][0]*proj[0] + recon[r][1]*proj[1]
    A = proj.T  # [4,2]
    recon = np.zeros((4, 2), dtype=np.complex128)
    recon[0, 0] = 1.0
    recon[1, 1] = 1.0
    for r in (2, 3):
        coef, res, _, _ = np.linalg.lstsq(A, (2 * P[r, :]), rcond=None)
        assert np.allclose(A @ coef, 2 * P[r, :], atol=1e-12), (mu, sign, r)
        recon[r] = [_as_unit(c) for c in coef]
    proj = np.vectorize(_as_unit)(proj)
    return proj, recon


def check_algebra():
    """Sanity checks used by tests: Clifford algebra + hermiticity."""
    for mu in range(4):
        assert np.allclose(GAMMA[mu].conj().T, GAMMA[mu]), f"gamma{mu} not hermitian"
        for nu in range(4):
            anti = GAMMA[mu] @ GAMMA[nu] + GAMMA[nu] @ GAMMA[mu]
            expect = 2 * np.eye(4) if mu == nu else np.zeros((4, 4))
            assert np.allclose(anti, expect), f"Clifford fails for {mu},{nu}"
    assert np.allclose(GAMMA5 @ GAMMA5, ID4)
    assert np.allclose(GAMMA5.conj().T, GAMMA5)
    # gamma5 diagonal in this basis
    assert np.allclose(GAMMA5, np.diag(np.diag(GAMMA5)))
    # projector structure coefficients are unit phases
    for mu in range(4):
        for sign in (+1, -1):
            proj, recon = half_projector_structure(mu, sign)
            for row in proj:
                nz = [c for c in row if c != 0]
                assert len(nz) == 2 and all(abs(abs(c) - 1) < 1e-12 for c in nz)
            for r in (2, 3):
                nz = [c for c in recon[r] if c != 0]
                assert len(nz) == 1 and abs(abs(nz[0]) - 1) < 1e-12
    return True
