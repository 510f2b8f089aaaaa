"""Op dispatch: HIP extension on GPU tensors, torch oracle on CPU.

The HIP extension (quda_amd_hip, built from csrc/ by setup.py) is the ONLY
compute path on a GPU box: if a field lives on a CUDA/HIP device and the
extension is missing we raise — no silent eager fallback (per-project rule:
GPU tests must exercise the native kernels).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from ..fields.gauge import GaugeField
from ..fields.geometry import LatticeGeometry
from ..fields.spinor import SpinorField
from . import reference as ref

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import quda_amd_hip  # built in-tree: csrc/ -> quda_amd_hip.so
        _EXT = quda_amd_hip
    except ImportError as e:
        _EXT_ERR = str(e)
    return _EXT


def hip_ext(required: bool):
    ext = _load_ext()
    if ext is None and required:
        raise RuntimeError(
            f"quda_amd_hip extension not available ({_EXT_ERR}); "
            "build it with `python setup.py build_ext --inplace` — the GPU "
            "path never falls back to eager torch.")
    return ext


def _on_gpu(*fields) -> bool:
    return any(f.device.type == "cuda" for f in fields if f is not None)


# ---------------------------------------------------------------------------
# Wilson dslash
# ---------------------------------------------------------------------------

def dslash_wilson(out: SpinorField, inp: SpinorField, gauge: GaugeField,
                  parity: int, dagger: bool = False,
                  xpay: Optional[tuple] = None):
    """out(parity) = D in(1-parity)  [+ a * x(parity) if xpay=(a, x)].

    out/inp are single-parity fields (n_parity==1). The xpay fusion is the
    reference's DslashXpay (dirac_quda.h:268).
    """
    geo = out.geo
    if _on_gpu(out.data, inp.data):
        ext = hip_ext(required=True)
        _gpu_dslash_wilson(ext, out, inp, gauge, parity, dagger, xpay)
        return out
    # oracle path
    u = gauge.to_complex()
    psi = inp.to_complex()[0]
    res = ref.dslash_wilson_parity(u, psi, geo, parity, dagger)
    if xpay is not None:
        a, x = xpay
        res = a * x.to_complex()[0] + res
    out.from_complex(res.unsqueeze(0))
    return out


def _gpu_dslash_wilson(ext, out, inp, gauge, parity, dagger, xpay):
    a = 0.0
    xdata = out.data  # unused when a == 0
    xnorm = _norm_or_empty(out)
    if xpay is not None:
        a, x = xpay
        xdata, xnorm = x.data, _norm_or_empty(x)
    ext.dslash_wilson(out.data, _norm_or_empty(out), inp.data,
                      _norm_or_empty(inp), gauge.data,
                      list(out.geo.dims), parity, bool(dagger),
                      float(a), xdata, xnorm)


def _norm_or_empty(f: SpinorField):
    if f.norm is not None:
        return f.norm
    return torch.empty(0, dtype=torch.float32, device=f.device)


# ---------------------------------------------------------------------------
# Clover apply
# ---------------------------------------------------------------------------

def apply_clover(out: SpinorField, inp: SpinorField, clover, parity: int,
                 inverse: bool = False):
    """out = A(parity) in  (site-local 12x12; clover is a CloverField)."""
    if _on_gpu(out.data, inp.data):
        ext = hip_ext(required=True)
        ext.clover_apply(out.data, _norm_or_empty(out), inp.data,
                         _norm_or_empty(inp),
                         clover.inv_data if inverse else clover.data,
                         parity)
        return out
    A = clover.to_complex(inverse=inverse)[parity]
    psi = inp.to_complex()[0]
    out.from_complex(ref.apply_clover(A, psi).unsqueeze(0))
    return out
