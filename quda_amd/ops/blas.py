"""BLAS / reduction layer on SpinorFields (ref: lib/blas_quda.cu,
lib/reduce_quda.cu — ~60 fused vector ops; here the core set the solvers
need, with the same fusion boundaries so the HIP kernels drop in 1:1).

Dispatch rule: double/single fields -> direct torch ops on the chunked
storage (torch on ROCm is already a bandwidth-bound eager path; the fused
HIP kernels in csrc/blas.hip replace the multi-kernel sequences on GPU).
half fields -> dequantize/requantize via complex (CPU oracle only; on GPU
the HIP kernels consume (data, norm) natively).

All reductions return python floats/complex and, in multi-rank runs, are
globally summed over the process grid (ref: reducer.h allreduce sites).
"""

from __future__ import annotations

import torch

from ..fields.spinor import SpinorField
from ..parallel.comms import allreduce_sum


def _is_half(x: SpinorField) -> bool:
    return x.precision == "half"


def _pairs(t: torch.Tensor) -> torch.Tensor:
    """View chunked real storage as [..., n, 2] (re, im) pairs."""
    return t.reshape(*t.shape[:-1], t.shape[-1] // 2, 2)


def _requant(x: SpinorField, vals: torch.Tensor):
    """Write complex values [P,V,12] back into a (possibly half) field."""
    x.from_complex(vals.reshape(x.n_parity, x.volume_cb, 4, 3))


def _cvals(x: SpinorField) -> torch.Tensor:
    return x.to_complex(torch.complex128).reshape(x.n_parity, x.volume_cb, 12)


# -- reductions -------------------------------------------------------------

def norm2(x: SpinorField) -> float:
    if _is_half(x):
        v = _cvals(x)
        r = (v.real ** 2 + v.imag ** 2).sum().item()
    else:
        d = x.data.to(torch.float64) if x.data.dtype != torch.float64 else x.data
        r = (d * d).sum().item()
    return allreduce_sum(r)


def re_dot(x: SpinorField, y: SpinorField) -> float:
    """Re <x, y>."""
    if _is_half(x) or _is_half(y):
        vx, vy = _cvals(x), _cvals(y)
        r = (vx.conj() * vy).real.sum().item()
    else:
        r = (x.data.to(torch.float64) * y.data.to(torch.float64)).sum().item()
    return allreduce_sum(r)


def c_dot(x: SpinorField, y: SpinorField) -> complex:
    """<x, y> = sum conj(x) y."""
    if _is_half(x) or _is_half(y):
        vx, vy = _cvals(x), _cvals(y)
        s = (vx.conj() * vy).sum().item()
        return complex(allreduce_sum(s.real), allreduce_sum(s.imag))
    px, py = _pairs(x.data.to(torch.float64)), _pairs(y.data.to(torch.float64))
    re = (px * py).sum().item()
    im = (px[..., 0] * py[..., 1] - px[..., 1] * py[..., 0]).sum().item()
    return complex(allreduce_sum(re), allreduce_sum(im))


def axpy_norm2(a: float, x: SpinorField, y: SpinorField) -> float:
    """y += a*x; returns ||y||^2 (fused in csrc/blas.hip; ref axpyNorm2)."""
    axpy(a, x, y)
    return norm2(y)


def xmy_norm2(x: SpinorField, y: SpinorField) -> float:
    """y = x - y; returns ||y||^2."""
    if _is_half(x) or _is_half(y):
        v = _cvals(x) - _cvals(y)
        _requant(y, v)
        return allreduce_sum((v.real ** 2 + v.imag ** 2).sum().item())
    y.data.copy_(x.data - y.data)
    return norm2(y)


# -- elementwise ------------------------------------------------------------

def copy(dst: SpinorField, src: SpinorField) -> SpinorField:
    dst.copy_(src)
    return dst


def zero(x: SpinorField) -> SpinorField:
    return x.zero_()


def axpy(a: float, x: SpinorField, y: SpinorField) -> SpinorField:
    """y = a*x + y."""
    if _is_half(x) or _is_half(y):
        _requant(y, _cvals(y) + a * _cvals(x))
    else:
        y.data.add_(x.data.to(y.data.dtype), alpha=float(a))
    return y


def xpay(x: SpinorField, a: float, y: SpinorField) -> SpinorField:
    """y = x + a*y."""
    if _is_half(x) or _is_half(y):
        _requant(y, _cvals(x) + a * _cvals(y))
    else:
        y.data.mul_(float(a)).add_(x.data.to(y.data.dtype))
    return y


def axpby(a: float, x: SpinorField, b: float, y: SpinorField) -> SpinorField:
    """y = a*x + b*y."""
    if _is_half(x) or _is_half(y):
        _requant(y, a * _cvals(x) + b * _cvals(y))
    else:
        y.data.mul_(float(b)).add_(x.data.to(y.data.dtype), alpha=float(a))
    return y


def caxpy(a: complex, x: SpinorField, y: SpinorField) -> SpinorField:
    """y += a*x (complex a)."""
    if _is_half(x) or _is_half(y):
        _requant(y, _cvals(y) + a * _cvals(x))
        return y
    px = _pairs(x.data.to(y.data.dtype))
    py = _pairs(y.data)
    py[..., 0] += a.real * px[..., 0] - a.imag * px[..., 1]
    py[..., 1] += a.real * px[..., 1] + a.imag * px[..., 0]
    return y


def scal(a: float, x: SpinorField) -> SpinorField:
    if _is_half(x):
        x.norm.mul_(abs(float(a)))
        if a < 0:
            x.data.neg_()
    else:
        x.data.mul_(float(a))
    return x
