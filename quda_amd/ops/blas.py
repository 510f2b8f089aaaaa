"""BLAS / reduction layer on SpinorFields (ref: lib/blas_quda.cu,
lib/reduce_quda.cu — the fused op set the solvers need; HIP kernels in
csrc/blas.hip are the GPU path for every precision, plain torch the CPU
path; half on CPU goes through dequantized complex).

All reductions return python floats/complex, globally summed over the
process grid (ref: include/reducer.h allreduce call sites).
"""

from __future__ import annotations

import torch

from ..fields.spinor import SpinorField
from ..parallel.comms import allreduce_sum
from .dispatch import hip_ext, norm_or_empty, on_gpu


def _is_half(x: SpinorField) -> bool:
    return x.precision in ("half", "quarter")


def _pairs(t: torch.Tensor) -> torch.Tensor:
    return t.reshape(*t.shape[:-1], t.shape[-1] // 2, 2)


def _requant(x: SpinorField, vals: torch.Tensor):
    x.from_complex(vals.reshape(x.n_parity, x.volume_cb, *x.site_shape))


def _cvals(x: SpinorField) -> torch.Tensor:
    return x.to_complex(torch.complex128).reshape(x.n_parity, x.volume_cb,
                                                  x.ncomp // 2)


_DETERMINISTIC = None


def deterministic_reductions() -> bool:
    """Fixed-order GPU reductions (ref: QUDA_DETERMINISTIC_REDUCE):
    per-block partials + ordered tree sum instead of f64 atomics. Env
    QUDA_AMD_DETERMINISTIC_REDUCE=1 or set_deterministic(True)."""
    global _DETERMINISTIC
    if _DETERMINISTIC is None:
        import os
        _DETERMINISTIC = os.environ.get("QUDA_AMD_DETERMINISTIC_REDUCE",
                                        "0") == "1"
    return _DETERMINISTIC


def set_deterministic(v: bool) -> None:
    global _DETERMINISTIC
    _DETERMINISTIC = bool(v)


def _gpu_blas(op: int, a, b, x: SpinorField, y: SpinorField):
    ext = hip_ext()
    return ext.blas_op(op, float(a), float(b), x.data, norm_or_empty(x),
                       y.data, norm_or_empty(y), x.volume_cb,
                       x.n_parity * x.volume_cb, ncomp=x.ncomp,
                       deterministic=deterministic_reductions())


# -- reductions -------------------------------------------------------------

def norm2(x: SpinorField) -> float:
    if on_gpu(x):
        ext = hip_ext()
        r = _gpu_blas(ext.BLAS_NORM2, 0, 0, x, x)[0].item()
    elif _is_half(x):
        v = _cvals(x)
        r = (v.real ** 2 + v.imag ** 2).sum().item()
    else:
        d = x.data.to(torch.float64)
        r = (d * d).sum().item()
    return allreduce_sum(r)


def re_dot(x: SpinorField, y: SpinorField) -> float:
    if on_gpu(x, y):
        ext = hip_ext()
        r = _gpu_blas(ext.BLAS_REDOT, 0, 0, x, y)[0].item()
    elif _is_half(x) or _is_half(y):
        r = (_cvals(x).conj() * _cvals(y)).real.sum().item()
    else:
        r = (x.data.to(torch.float64) * y.data.to(torch.float64)).sum().item()
    return allreduce_sum(r)


def c_dot(x: SpinorField, y: SpinorField) -> complex:
    if on_gpu(x, y):
        ext = hip_ext()
        t = _gpu_blas(ext.BLAS_CDOT, 0, 0, x, y)
        re, im = t[0].item(), t[1].item()
        return complex(allreduce_sum(re), allreduce_sum(im))
    if _is_half(x) or _is_half(y):
        s = (_cvals(x).conj() * _cvals(y)).sum().item()
        return complex(allreduce_sum(s.real), allreduce_sum(s.imag))
    px, py = _pairs(x.data.to(torch.float64)), _pairs(y.data.to(torch.float64))
    re = (px * py).sum().item()
    im = (px[..., 0] * py[..., 1] - px[..., 1] * py[..., 0]).sum().item()
    return complex(allreduce_sum(re), allreduce_sum(im))


def axpy_norm2(a: float, x: SpinorField, y: SpinorField) -> float:
    """y += a*x; returns ||y||^2 (fused; ref axpyNorm2)."""
    if on_gpu(x, y):
        ext = hip_ext()
        r = _gpu_blas(ext.BLAS_AXPY_NORM2, a, 0, x, y)[0].item()
        return allreduce_sum(r)
    axpy(a, x, y)
    return norm2(y)


def triple_cg_update(a: float, p: SpinorField, Ap: SpinorField,
                     x: SpinorField, r: SpinorField) -> float:
    """Fused CG inner update (ref reduce_core.cuh tripleCGUpdate):
    x += a*p; r -= a*Ap; returns ||r||^2 — one kernel instead of two,
    one residual pass instead of two."""
    if on_gpu(p, Ap, x, r):
        ext = hip_ext()
        res = ext.blas_op(ext.BLAS_TRIPLE_CG, float(a), 0.0, p.data,
                          norm_or_empty(p), Ap.data, norm_or_empty(Ap),
                          p.volume_cb, p.n_parity * p.volume_cb,
                          ncomp=p.ncomp,
                          deterministic=deterministic_reductions(),
                          z=x.data, z_n=norm_or_empty(x),
                          w=r.data, w_n=norm_or_empty(r))
        return allreduce_sum(res[0].item())
    axpy(a, p, x)
    return axpy_norm2(-a, Ap, r)


def xmy_norm2(x: SpinorField, y: SpinorField) -> float:
    """y = x - y; returns ||y||^2."""
    if on_gpu(x, y):
        ext = hip_ext()
        r = _gpu_blas(ext.BLAS_XMY_NORM2, 0, 0, x, y)[0].item()
        return allreduce_sum(r)
    if _is_half(x) or _is_half(y):
        v = _cvals(x) - _cvals(y)
        _requant(y, v)
        return allreduce_sum((v.real ** 2 + v.imag ** 2).sum().item())
    y.data.copy_(x.data - y.data)
    return norm2(y)


# -- elementwise ------------------------------------------------------------

def copy(dst: SpinorField, src: SpinorField) -> SpinorField:
    if dst is src:
        return dst
    if on_gpu(dst, src) and dst.precision != src.precision:
        ext = hip_ext()
        ext.convert(dst.data, norm_or_empty(dst), src.data, norm_or_empty(src),
                    dst.volume_cb, dst.n_parity * dst.volume_cb,
                    ncomp=dst.ncomp)
        return dst
    dst.copy_(src)
    return dst


def zero(x: SpinorField) -> SpinorField:
    return x.zero_()


def axpy(a: float, x: SpinorField, y: SpinorField) -> SpinorField:
    """y = a*x + y."""
    if on_gpu(x, y):
        ext = hip_ext()
        _gpu_blas(ext.BLAS_AXPY, a, 0, x, y)
    elif _is_half(x) or _is_half(y):
        _requant(y, _cvals(y) + a * _cvals(x))
    else:
        y.data.add_(x.data.to(y.data.dtype), alpha=float(a))
    return y


def xpay(x: SpinorField, a: float, y: SpinorField) -> SpinorField:
    """y = x + a*y."""
    if on_gpu(x, y):
        ext = hip_ext()
        _gpu_blas(ext.BLAS_XPAY, a, 0, x, y)
    elif _is_half(x) or _is_half(y):
        _requant(y, _cvals(x) + a * _cvals(y))
    else:
        y.data.mul_(float(a)).add_(x.data.to(y.data.dtype))
    return y


def axpby(a: float, x: SpinorField, b: float, y: SpinorField) -> SpinorField:
    """y = a*x + b*y."""
    if on_gpu(x, y):
        ext = hip_ext()
        _gpu_blas(ext.BLAS_AXPBY, a, b, x, y)
    elif _is_half(x) or _is_half(y):
        _requant(y, a * _cvals(x) + b * _cvals(y))
    else:
        y.data.mul_(float(b)).add_(x.data.to(y.data.dtype), alpha=float(a))
    return y


def caxpy(a: complex, x: SpinorField, y: SpinorField) -> SpinorField:
    """y += a*x (complex a)."""
    if on_gpu(x, y):
        ext = hip_ext()
        _gpu_blas(ext.BLAS_CAXPY, a.real, a.imag, x, y)
        return y
    if _is_half(x) or _is_half(y):
        _requant(y, _cvals(y) + a * _cvals(x))
        return y
    px = _pairs(x.data.to(y.data.dtype))
    py = _pairs(y.data)
    re = a.real * px[..., 0] - a.imag * px[..., 1]
    im = a.real * px[..., 1] + a.imag * px[..., 0]
    py[..., 0] += re
    py[..., 1] += im
    return y


def caxpby(a: complex, x: SpinorField, b: complex, y: SpinorField) -> SpinorField:
    """y = a*x + b*y (complex a, b; ref blas_core.cuh caxpby)."""
    a, b = complex(a), complex(b)
    if on_gpu(x, y):
        ext = hip_ext()
        ext.blas_op(ext.BLAS_CAXPBY, a.real, a.imag, x.data, norm_or_empty(x),
                    y.data, norm_or_empty(y), x.volume_cb,
                    x.n_parity * x.volume_cb, b.real, b.imag, ncomp=x.ncomp)
        return y
    _requant_any(y, a * _cvals(x) + b * _cvals(y))
    return y


def _requant_any(y: SpinorField, vals):
    """Write complex values back into y at its precision."""
    y.from_complex(vals.reshape(y.n_parity, y.volume_cb, *y.site_shape))


def scal(a: float, x: SpinorField) -> SpinorField:
    if on_gpu(x):
        ext = hip_ext()
        _gpu_blas(ext.BLAS_SCAL, a, 0, x, x)
    elif _is_half(x):
        x.norm.mul_(abs(float(a)))
        if a < 0:
            x.data.neg_()
    else:
        x.data.mul_(float(a))
    return x
