"""Multi-BLAS / multi-reduce: tiled vector-set operations
(ref: lib/multi_blas_quda.cu caxpy(a[NxM],x,y) + lib/multi_reduce_quda.cu
cDotProduct matrices — MI355X-first: the vector set is exposed as ONE
complex matrix and the whole tile runs as a single rocBLAS complex GEMM
through torch.matmul, which is exactly the GEMM-shaped work MFMA wants)."""

from __future__ import annotations

from typing import List, Sequence

import numpy as np
import torch

from ..fields.spinor import SpinorField
from ..parallel.comms import allreduce_tensor


def _cview(x: SpinorField) -> torch.Tensor:
    """Flat complex view of the native data (no copy; half dequantizes)."""
    if x.precision == "half":
        return x.to_complex().reshape(-1)
    d = x.data
    c = torch.view_as_complex(d.reshape(*d.shape[:-1], d.shape[-1] // 2, 2))
    return c.reshape(-1)


def block_cdot(xs: Sequence[SpinorField], ys: Sequence[SpinorField]
               ) -> torch.Tensor:
    """G[i,j] = <x_i, y_j> as one complex GEMM (ref cDotProduct NxM)."""
    X = torch.stack([_cview(x) for x in xs])
    Y = torch.stack([_cview(y) for y in ys])
    if X.dtype == torch.complex64:
        X, Y = X.to(torch.complex128), Y.to(torch.complex128)
    G = X.conj() @ Y.mT if Y.dim() > 1 else None
    G = X.conj() @ Y.transpose(0, 1) if G is None else G
    return allreduce_tensor(G)


def block_caxpy(A, xs: Sequence[SpinorField], ys: Sequence[SpinorField]):
    """y_i += sum_j A[i,j] x_j as one complex GEMM (ref caxpy NxM)."""
    if not torch.is_tensor(A):
        A = torch.tensor(np.asarray(A, dtype=complex))
    X = torch.stack([_cview(x) for x in xs])
    A = A.to(device=X.device, dtype=X.dtype if X.is_complex() else torch.complex128)
    if X.dtype == torch.complex64:
        upd = (A.to(torch.complex64) @ X)
    else:
        upd = A @ X
    for i, y in enumerate(ys):
        if y.precision == "half":
            c = y.to_complex().reshape(-1) + upd[i]
            y.from_complex(c.reshape(y.n_parity, y.volume_cb, *y.site_shape))
        else:
            v = _cview(y)
            v += upd[i].to(v.dtype)
    return ys
