"""Op dispatch: HIP extension on GPU tensors, torch oracle on CPU.

The HIP extension (quda_amd_hip.so, built in-tree from csrc/ by
build_hip.py) is the ONLY compute path on a GPU box: if a field lives on a
CUDA/HIP device and the extension is missing we raise — no silent eager
fallback (GPU tests must exercise the native kernels).

Fused dslash modes (csrc/dslash_wilson.h):
  PLAIN      : out = [x +] a * (D in)
  CLOV_POST  : out = [x +] a * (A (D in))     A = clover or its inverse
  CLOV_X     : out = A x + a * (D in)
"""

from __future__ import annotations

from typing import Optional

import torch

from ..fields.gauge import GaugeField
from ..fields.spinor import SpinorField
from . import reference as ref

PLAIN, CLOV_POST, CLOV_X = 0, 1, 2
TWIST_POST, TWIST_X, CLOVTW_X = 3, 4, 5

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import quda_amd_hip
        _EXT = quda_amd_hip
    except ImportError as e:
        _EXT_ERR = str(e)
    return _EXT


_TRACE = None
_TRACE_LOG = []


def trace_enabled() -> bool:
    """Per-launch trace (ref: QUDA_ENABLE_TRACE -> trace_N.tsv): records
    (op, key) tuples in dispatch order; QUDA_AMD_TRACE=1 enables,
    trace_log()/trace_dump() read/persist."""
    global _TRACE
    if _TRACE is None:
        import os
        _TRACE = os.environ.get("QUDA_AMD_TRACE", "0") == "1"
    return _TRACE


def set_trace(v: bool) -> None:
    global _TRACE
    _TRACE = bool(v)


def trace_log():
    return _TRACE_LOG


def trace_record(op: str, key: str) -> None:
    if trace_enabled():
        _TRACE_LOG.append((op, key))


def trace_dump(path: str) -> None:
    with open(path, "w") as f:
        f.write("# op\tkey\n")
        for op, key in _TRACE_LOG:
            f.write(f"{op}\t{key}\n")


_POLICY = None


def dslash_policy() -> str:
    """'overlap' (interior/exterior split around async RCCL transfers,
    default) or 'fused' (blocking exchange, single ghost-aware kernel).
    Env QUDA_AMD_DSLASH_POLICY overrides; QUDA_AMD_BLOCKING_COMMS=1 is
    the safe-mode switch (forces 'fused' AND disables the policy
    autotuner's overlap candidates — the multi-GPU first-run fallback)."""
    global _POLICY
    if _POLICY is None:
        import os
        if os.environ.get("QUDA_AMD_BLOCKING_COMMS", "0") == "1":
            _POLICY = "fused"
        else:
            _POLICY = os.environ.get("QUDA_AMD_DSLASH_POLICY", "overlap")
    return _POLICY


def set_dslash_policy(p: str) -> None:
    global _POLICY
    assert p in ("overlap", "fused", "ipc")
    _POLICY = p


def hip_ext(required: bool = True):
    ext = _load_ext()
    if ext is None and required:
        raise RuntimeError(
            f"quda_amd_hip extension not available ({_EXT_ERR}); "
            "build it with `python build_hip.py` — the GPU path never "
            "falls back to eager torch.")
    return ext


def on_gpu(*fields) -> bool:
    return any(f is not None and f.device.type == "cuda" for f in fields)


def _empty(dev):
    return torch.empty(0, dtype=torch.float32, device=dev)


def norm_or_empty(f: Optional[SpinorField]):
    if f is None or f.norm is None:
        return _empty("cpu" if f is None else f.device)
    return f.norm


from ..fields.gauge import RECON_COMPS

_AUTOTUNE = None

# hook: zero-arg callable executed inside the dslash comms window (ref:
# dslash::aux_worker, lib/dslash_quda.cu:73 / inv_multi_cg_quda.cpp:115)
aux_worker = None


def _autotune_on() -> bool:
    """Dispatch-level autotuning (QUDA_AMD_AUTOTUNE=0 disables; on by
    default like the reference's tuneLaunch)."""
    global _AUTOTUNE
    if _AUTOTUNE is None:
        import os
        _AUTOTUNE = os.environ.get("QUDA_AMD_AUTOTUNE", "1") != "0"
    return _AUTOTUNE


def dslash_wilson(out: SpinorField, inp: SpinorField, gauge: GaugeField,
                  parity: int, dagger: bool = False, mode: int = PLAIN,
                  a: float = 1.0, x: Optional[SpinorField] = None,
                  clover=None, clover_inverse: bool = False,
                  twist=(0.0, 0.0)):
    """Apply the fused Wilson(-clover/-twisted) stencil; out at `parity`,
    in at the opposite parity; x (same parity as out) enables the xpay
    term; twist=(b_re,b_im) feeds the TWIST_*/CLOVTW_* epilogues
    (T(b) v = b_re v + i b_im g5 v)."""
    from ..parallel import comms
    geo = out.geo
    xpay = x is not None
    mask = comms.comm_mask()
    trace_record("dslash_wilson",
                 f"{geo.dims}/p{parity}/m{mode}/dag{int(dagger)}/{inp.precision}")
    if on_gpu(out, inp):
        ext = hip_ext()
        cl_t = torch.empty(0, dtype=out.data.dtype, device=out.device)
        if mode in (CLOV_POST, CLOV_X, CLOVTW_X):
            cl_t = clover.inv_data if clover_inverse else clover.data
        xf = x if x is not None else out

        def launch(kt, ghosts=[], nrms=[], face_cb=[]):
            ext.dslash_wilson(
                out.data, norm_or_empty(out), inp.data, norm_or_empty(inp),
                gauge.data, cl_t, xf.data, norm_or_empty(xf),
                list(geo.dims), geo.parity_offset, geo.volume_cb, parity,
                bool(dagger), mode, xpay, float(a),
                RECON_COMPS[gauge.reconstruct], ghosts, nrms, face_cb,
                mask if kt else 0, kt, float(twist[0]), float(twist[1]))

        if not mask:
            # autotuned workgroup size (ref: lib/tune.cpp tuneLaunch —
            # first call per (kernel, dims, precision, mode) key measures
            # the candidates, later calls hit the cache; the dslash
            # overwrites `out`, so re-running it during tuning is safe)
            if _autotune_on():
                from ..utils.tune import tune_dslash
                key = (f"dslash_wilson/{'x'.join(map(str, geo.dims))}/"
                       f"{inp.precision}/r{RECON_COMPS[gauge.reconstruct]}"
                       f"/m{mode}/x{int(xpay)}")
                tune_dslash(lambda: launch(0), key)
            launch(0)
            return out
        # comm-overlap policy (role of ref lib/dslash_policy.hpp): pack ->
        # start RCCL transfers (their kernels run on NCCL's streams) ->
        # INTERIOR on the compute stream overlaps them -> stream-waits on
        # the transfers -> EXTERIOR adds ghost hops. "fused" = blocking.
        from ..parallel.halo import get_spinor_halo
        h = get_spinor_halo(geo, inp.precision, inp.device, mask)

        if dslash_policy() == "ipc":
            # remote-write transport (comm_target.cpp:41-134 role): pack
            # kernels scatter each face straight into the peer rank's
            # recv buffer over xGMI; interior overlaps the peers' packs
            from ..parallel.ipc_halo import get_ipc_halo
            hi = get_ipc_halo(geo, inp.precision, inp.device, mask)
            hi.pack_remote(ext, inp, 1 - parity, bool(dagger))
            ghosts, nrms, face_cb = hi.ghost_args()
            launch(2, ghosts, nrms, face_cb)   # interior
            hi.complete()
            launch(3, ghosts, nrms, face_cb)   # exterior
            return out

        def run_halo():
            h.pack(ext, inp, 1 - parity, bool(dagger))
            ghosts, nrms, face_cb = h.ghost_args()
            if dslash_policy() == "fused":
                h.exchange()
                launch(1, ghosts, nrms, face_cb)
            else:
                reqs = h.exchange_start()
                launch(2, ghosts, nrms, face_cb)   # interior
                if aux_worker is not None:
                    # independent work queued into the comms window (ref:
                    # dslash::aux_worker lib/dslash_quda.cu:73 — the
                    # multishift p-updates ride here)
                    aux_worker()
                for r in reqs:
                    r.wait()
                launch(3, ghosts, nrms, face_cb)   # exterior

        import os
        if _autotune_on() and os.environ.get("QUDA_AMD_BLOCKING_COMMS",
                                             "0") != "1":
            # policy-level autotune (ref DslashPolicyTune): every rank
            # reaches this collectively, so candidate runs stay in
            # lockstep; rank-0's winner is broadcast by the tuner
            from ..utils.tune import tune_dslash_policy
            key = (f"dslash_policy/{'x'.join(map(str, geo.dims))}/"
                   f"{inp.precision}/mask{mask}")
            tune_dslash_policy(run_halo, key)
        run_halo()
        return out
    # ---- oracle path ----
    u = gauge.to_complex()
    psi = inp.to_complex()[0]
    halo = None
    if mask:
        from ..parallel.halo import active_dims, exchange_psi_oracle
        halo = {
            "mask": mask,
            "psi": exchange_psi_oracle(psi, geo, 1 - parity, mask),
            "u_bwd": {mu: gauge.bwd_ghost(mu, parity)
                      for mu in active_dims(mask)},
        }
    res = ref.dslash_wilson_parity(u, psi, geo, parity, dagger, halo=halo)
    if mode == CLOV_POST:
        A = clover.to_complex(inverse=clover_inverse)[parity]
        res = ref.apply_clover(A, res)
    if mode == TWIST_POST:
        res = ref.apply_twist(res, twist[0], twist[1])
    if mode == CLOV_X:
        A = clover.to_complex(inverse=clover_inverse)[parity]
        res = ref.apply_clover(A, x.to_complex()[0]) + a * res
    elif mode == TWIST_X:
        res = ref.apply_twist(x.to_complex()[0], twist[0], twist[1]) + a * res
    elif mode == CLOVTW_X:
        A = clover.to_complex(inverse=clover_inverse)[parity]
        xv = x.to_complex()[0]
        res = (ref.apply_clover(A, xv)
               + ref.apply_twist(xv, 0.0, twist[1]) + a * res)
    elif xpay:
        res = x.to_complex()[0] + a * res
    else:
        res = a * res
    out.from_complex(res.unsqueeze(0))
    return out


_MRHS_GROUP = None


def mrhs_group_size() -> int:
    """Preferred kernel NRHS (QUDA_AMD_MRHS_GROUP: 4, 2, or 1=disable).

    Default 1 (per-RHS launches) — measured on MI355X: the 256 MB
    Infinity Cache keeps the ~100 MB gauge+clover streams resident
    ACROSS per-RHS launches at 32^3x64, so kernel-level RHS batching
    saves no HBM traffic there, while the direction-major NRHS loop caps
    per-wave load ILP at NRHS x 4 chunks (vs 32 for the 8-direction
    gather-first single-RHS kernel) and measures ~2x slower
    (profiles/r02_dslash_sweep.md). The batched-HALO merging (one
    message per face for the whole batch) is independent of this knob
    and always on. Set 2/4 for volumes whose gauge stream exceeds L3."""
    global _MRHS_GROUP
    if _MRHS_GROUP is None:
        import os
        v = int(os.environ.get("QUDA_AMD_MRHS_GROUP", "1"))
        _MRHS_GROUP = v if v in (1, 2, 4) else 1
    return _MRHS_GROUP


def _mrhs_groups(n: int):
    """Split n RHS into kernel-supported group sizes; the remainder runs
    the per-RHS kernel."""
    g = mrhs_group_size()
    groups, r0 = [], 0
    if g >= 4:
        while n - r0 >= 4:
            groups.append((r0, 4))
            r0 += 4
    if g >= 2:
        while n - r0 >= 2:
            groups.append((r0, 2))
            r0 += 2
    return groups, r0


def dslash_wilson_batch(outs, inps, gauge: GaugeField, parity: int,
                        dagger: bool = False, a: float = 1.0, xs=None,
                        mode: int = PLAIN, clover=None,
                        clover_inverse: bool = False):
    """Multi-RHS Wilson(-clover) dslash: kernel-level NRHS batching (the
    k_dslash_wilson_mrhs kernel processes 2 or 4 sides per gauge/clover
    load — ref dslash_wilson.cuh:38-40 MAX_MULTI_RHS arrays) plus MERGED
    halos (one message per face for the whole batch; ref
    create_comms_batch, dslash_wilson.hpp:48)."""
    from ..parallel import comms
    geo = outs[0].geo
    n = len(inps)
    mask = comms.comm_mask()
    xpay = xs is not None

    def per_rhs(i, kt=None, ghosts=None, nrms=None, face_cb=None, h=None):
        dslash_wilson(outs[i], inps[i], gauge, parity, dagger=dagger,
                      a=a, x=xs[i] if xs else None, mode=mode,
                      clover=clover, clover_inverse=clover_inverse)

    if n == 1 or not on_gpu(outs[0], inps[0]):
        if not mask or n == 1:
            for i in range(n):
                per_rhs(i)
            return outs
        return _dslash_batch_oracle(outs, inps, gauge, parity, dagger, a,
                                    xs, mode, clover, clover_inverse)

    ext = hip_ext()
    groups, rem0 = _mrhs_groups(n)
    cl_t = torch.empty(0, dtype=outs[0].data.dtype, device=outs[0].device)
    if mode == CLOV_POST:
        cl_t = clover.inv_data if clover_inverse else clover.data

    def launch_group(r0, g, kt, ghosts, nrms, face_cb):
        sl = slice(r0, r0 + g)
        xg = xs[sl] if xs else []
        ext.dslash_wilson_mrhs(
            [o.data for o in outs[sl]], [norm_or_empty(o) for o in outs[sl]],
            [i_.data for i_ in inps[sl]],
            [norm_or_empty(i_) for i_ in inps[sl]],
            gauge.data, cl_t,
            [x_.data for x_ in xg], [norm_or_empty(x_) for x_ in xg],
            list(geo.dims), geo.parity_offset, geo.volume_cb, parity,
            bool(dagger), mode, xpay, float(a),
            RECON_COMPS[gauge.reconstruct], ghosts, nrms, face_cb,
            mask if kt else 0, kt)

    if not mask:
        for r0, g in groups:
            launch_group(r0, g, 0, [], [], [])
        for i in range(rem0, n):
            per_rhs(i)
        return outs

    from ..parallel.halo import get_batch_halo
    h = get_batch_halo(geo, inps[0].precision, inps[0].device, mask, n)
    for i in range(n):
        h.pack_one(ext, i, inps[i], 1 - parity, bool(dagger))
    reqs = h.exchange_start()

    def launch_one(i, kt):
        ghosts, nrms, face_cb = h.ghost_args(i)
        xf = xs[i] if xs else outs[i]
        ext.dslash_wilson(
            outs[i].data, norm_or_empty(outs[i]), inps[i].data,
            norm_or_empty(inps[i]), gauge.data, cl_t, xf.data,
            norm_or_empty(xf), list(geo.dims), geo.parity_offset,
            geo.volume_cb, parity, bool(dagger), mode, xpay, float(a),
            RECON_COMPS[gauge.reconstruct], ghosts, nrms, face_cb,
            mask, kt, 0.0, 0.0)

    # interiors overlap the batched transfer: mrhs kernels offset into
    # the batch ghost slabs from the group's base slice
    for r0, g in groups:
        ghosts, nrms, face_cb = h.ghost_args(r0)
        launch_group(r0, g, 2, ghosts, nrms, face_cb)
    for i in range(rem0, n):
        launch_one(i, 2)
    for r in reqs:
        r.wait()
    for i in range(n):  # exteriors are boundary-only: per-RHS kernels
        launch_one(i, 3)
    return outs


def _dslash_batch_oracle(outs, inps, gauge, parity, dagger, a, xs,
                         mode=PLAIN, clover=None, clover_inverse=False):
    # ---- CPU oracle path: ONE merged exchange for the batch ----
    from ..parallel import comms
    from ..parallel.halo import (active_dims, exchange_tensors)
    geo = outs[0].geo
    n = len(inps)
    mask = comms.comm_mask()
    xpay = xs is not None
    psis = [inp.to_complex()[0] for inp in inps]
    sends, recvs = {}, {}
    pin = 1 - parity
    for mu in active_dims(mask):
        hi = geo.dims[mu] - 1
        i0 = geo.face_index_cb(pin, mu, 0)
        i1 = geo.face_index_cb(pin, mu, hi)
        sends[(mu, 0)] = torch.stack([p[i0] for p in psis]).contiguous()
        sends[(mu, 1)] = torch.stack([p[i1] for p in psis]).contiguous()
        recvs[(mu, 0)] = torch.empty_like(sends[(mu, 1)])
        recvs[(mu, 1)] = torch.empty_like(sends[(mu, 0)])
    exchange_tensors(sends, recvs)
    u = gauge.to_complex()
    for i in range(n):
        halo = {
            "mask": mask,
            "psi": {k: v[i] for k, v in recvs.items()},
            "u_bwd": {mu: gauge.bwd_ghost(mu, parity)
                      for mu in active_dims(mask)},
        }
        res = ref.dslash_wilson_parity(u, psis[i], geo, parity, dagger,
                                       halo=halo)
        if mode == CLOV_POST:
            A = clover.to_complex(inverse=clover_inverse)[parity]
            res = ref.apply_clover(A, res)
        if xpay:
            res = xs[i].to_complex()[0] + a * res
        else:
            res = a * res
        outs[i].from_complex(res.unsqueeze(0))
    return outs


def apply_twist_field(out: SpinorField, inp: SpinorField, br: float,
                      bi: float, tau3: bool = False, acc: bool = False):
    """out = [out +] br*in + i*bi*g5*in. tau3=True: flavor-doublet mode
    (ls=2 field): the g5 coefficient flips sign for flavor 1 (g5 tau3)."""
    if on_gpu(out, inp):
        ext = hip_ext()
        ext.twist_apply(out.data, norm_or_empty(out), inp.data,
                        norm_or_empty(inp), float(br), float(bi),
                        out.volume_cb, out.n_parity * out.volume_cb,
                        tau3_vcb=(out.geo.volume_cb if tau3 else 0),
                        acc=acc)
        return out
    c = inp.to_complex()
    if tau3:
        assert inp.ls == 2
        V = inp.geo.volume_cb
        r = torch.empty_like(c)
        r[:, :V] = ref.apply_twist(c[:, :V], br, bi)
        r[:, V:] = ref.apply_twist(c[:, V:], br, -bi)
    else:
        r = ref.apply_twist(c, br, bi)
    if acc:
        r = r + out.to_complex()
    out.from_complex(r)
    return out


def apply_clover(out: SpinorField, inp: SpinorField, clover, parity: int,
                 inverse: bool = False, v_stride: int = 0,
                 s_offset: int = 0):
    """out = A(parity) in (standalone; used by prepare/reconstruct).
    v_stride/s_offset address one 4-d slice of a 5-d (doublet) field, as
    in dslash_wilson_slice; the CPU path for slices is handled by the
    callers (they stage the full complex tensor)."""
    if on_gpu(out, inp):
        ext = hip_ext()
        ext.clover_apply(out.data, norm_or_empty(out), inp.data,
                         norm_or_empty(inp),
                         clover.inv_data if inverse else clover.data,
                         parity, out.geo.volume_cb, v_stride, s_offset)
        return out
    assert v_stride == 0, "CPU slice path handled by callers"
    A = clover.to_complex(inverse=inverse)[parity]
    psi = inp.to_complex()[0]
    out.from_complex(ref.apply_clover(A, psi).unsqueeze(0))
    return out


def dslash_staggered(out: SpinorField, inp: SpinorField, gauge: GaugeField,
                     parity: int, a: float = 0.0,
                     b: float = 1.0, x: Optional[SpinorField] = None,
                     long_gauge: Optional[GaugeField] = None):
    """Staggered stencil: out = [a*x +] b*(D in); out at `parity`, in at
    the opposite parity (nspin=1 fields). D^dag = -D: pass b=-b for the
    dagger. `long_gauge` (shift=3 stencil field) adds the Naik 3-hop term
    (improved staggered). Multi-rank Naik uses depth-3 ghosts and always
    runs the fused policy (no interior/exterior overlap for 3-hop halos
    yet); the long-link gauge needs no ghost at apply time — the stencil
    layout pre-shifts and exchanges bwd links at load time."""
    from ..parallel import comms
    geo = out.geo
    xpay = x is not None
    mask = comms.comm_mask()
    depth = 3 if (long_gauge is not None and mask) else 1
    if depth == 3:
        assert min(geo.dims) >= 4, "Naik halos need local extents >= 4"
    if on_gpu(out, inp):
        ext = hip_ext()
        xf = x if x is not None else out
        lng = (long_gauge.data if long_gauge is not None
               else torch.empty(0, dtype=out.data.dtype, device=out.device))

        def launch(kt, ghosts=[], nrms=[], face_cb=[]):
            ext.dslash_staggered(
                out.data, norm_or_empty(out), inp.data, norm_or_empty(inp),
                gauge.data, lng, xf.data, norm_or_empty(xf), list(geo.dims),
                geo.parity_offset, geo.volume_cb, parity, xpay, float(a),
                float(b), RECON_COMPS[gauge.reconstruct], ghosts, nrms,
                face_cb, mask if kt else 0, kt, depth)

        if not mask:
            launch(0)
            return out
        from ..parallel.halo import get_spinor_halo
        h = get_spinor_halo(geo, inp.precision, inp.device, mask, ncomp=6,
                            depth=depth)
        h.pack(ext, inp, 1 - parity, False)
        ghosts, nrms, face_cb = h.ghost_args()
        if depth == 3 or dslash_policy() == "fused":
            h.exchange()
            launch(1, ghosts, nrms, face_cb)
        else:
            reqs = h.exchange_start()
            launch(2, ghosts, nrms, face_cb)
            for r in reqs:
                r.wait()
            launch(3, ghosts, nrms, face_cb)
        return out
    # ---- oracle path ----
    u = gauge.to_complex()
    psi = inp.to_complex()[0]
    halo = None
    if mask:
        from ..parallel.halo import active_dims, exchange_psi_oracle
        halo = {
            "mask": mask,
            "psi": exchange_psi_oracle(psi, geo, 1 - parity, mask),
            "u_bwd": {mu: gauge.bwd_ghost(mu, parity)
                      for mu in active_dims(mask)},
        }
    res = ref.dslash_staggered_parity(u, psi, geo, parity, halo=halo)
    if long_gauge is not None:
        halo3 = None
        if mask:
            from ..parallel.halo import active_dims, exchange_psi_oracle
            halo3 = {
                "mask": mask,
                "psi3": exchange_psi_oracle(psi, geo, 1 - parity, mask,
                                            depth=3),
                "n_bwd": {(mu, c): long_gauge.bwd_ghost(mu, parity, c)
                          for mu in active_dims(mask) for c in range(3)},
            }
        res = res + ref.dslash_staggered_naik_parity(
            long_gauge.to_complex(), psi, geo, parity, halo=halo3)
    res = b * res
    if xpay:
        res = a * x.to_complex()[0] + res
    out.from_complex(res.unsqueeze(0))
    return out


def dwf5_op(out: SpinorField, inp: SpinorField, alpha: float, beta: float,
            mf: float, kind: int, dagger: bool = False, a: float = 1.0,
            x: Optional[SpinorField] = None):
    """5th-dimension ops on single-parity 5-d fields (csrc/dslash_dwf.h):
    kind=0: out = [a*x +] alpha*in + beta*(Ds in)
    kind=1: out = [x +] a * (alpha + beta*Ds)^{-1} in"""
    Ls = inp.ls
    xpay = x is not None
    if on_gpu(out, inp):
        ext = hip_ext()
        xf = x if x is not None else out
        ext.dwf5(out.data, norm_or_empty(out), inp.data, norm_or_empty(inp),
                 xf.data, norm_or_empty(xf), inp.geo.volume_cb, Ls,
                 xpay, bool(dagger), float(a), float(alpha), float(beta),
                 float(mf), kind)
        return out
    psi = inp.to_complex()[0]
    if kind == 0:
        res = ref.dslash5(psi, Ls, alpha, beta, mf, dagger)
        if xpay:
            res = a * x.to_complex()[0] + res
    else:
        res = a * ref.m5inv(psi, Ls, alpha, beta, mf, dagger)
        if xpay:
            res = x.to_complex()[0] + res
    out.from_complex(res.unsqueeze(0))
    return out


def _ztables(Ls, diag, hop, mf, dagger):
    """Host-side zMobius table assembly (csrc ZCoef): hop source/weight per
    chirality, and the sequence-ordered bidiagonal (1/diag, off, corner)
    for the inverse. diag[s] + hop[s]*Ds rows; dagger = conjugate transpose
    (coefficients then attach to the SOURCE slice)."""
    diag = [complex(d) for d in diag]
    hop = [complex(h) for h in hop]
    if not dagger:
        dg = diag
        su = [(s - 1) % Ls for s in range(Ls)]
        wu = [hop[s] * (-mf if s == 0 else 1.0) for s in range(Ls)]
        sl = [(s + 1) % Ls for s in range(Ls)]
        wl = [hop[s] * (-mf if s == Ls - 1 else 1.0) for s in range(Ls)]
    else:
        dg = [d.conjugate() for d in diag]
        su = [(s + 1) % Ls for s in range(Ls)]
        wu = [hop[(s + 1) % Ls].conjugate() * (-mf if s == Ls - 1 else 1.0)
              for s in range(Ls)]
        sl = [(s - 1) % Ls for s in range(Ls)]
        wl = [hop[(s - 1) % Ls].conjugate() * (-mf if s == 0 else 1.0)
              for s in range(Ls)]

    def seq(src, w):
        # order the rows so each row's off-diagonal hits the previous
        # step's variable; the remaining corner gets Sherman-Morrison
        inv = {src[r]: r for r in range(Ls)}
        o = [0]
        for _ in range(1, Ls):
            o.append(inv[o[-1]])
        di = [1.0 / dg[r] for r in o]
        e = [0j] + [w[o[i]] for i in range(1, Ls)]
        cw = w[o[0]]
        return o, di, e, cw

    ou, diu, eu, cwu = seq(su, wu)
    ol, dil, el, cwl = seq(sl, wl)

    def flat(v):
        out = []
        for z in v:
            out.extend((z.real, z.imag))
        return out

    return dict(au=flat(dg), al=flat(dg), wu=flat(wu), wl=flat(wl),
                su=su, sl=sl, ord_u=ou, ord_l=ol,
                diu=flat(diu), eu=flat(eu), dil=flat(dil), el=flat(el),
                cwu_re=cwu.real, cwu_im=cwu.imag,
                cwl_re=cwl.real, cwl_im=cwl.imag)


_ztable_cache = {}


def zdwf5_op(out: SpinorField, inp: SpinorField, diag, hop, mf: float,
             kind: int, dagger: bool = False, a=1.0,
             x: Optional[SpinorField] = None):
    """zMobius per-slice-complex 5th-dimension ops (csrc k_zdslash5 /
    k_zm5inv; ref: the zMobius branch of lib/dslash5_domain_wall.cu):
    kind=0: out = [a*x +] (diag + hop*Ds) in
    kind=1: out = [x +] a * (diag + hop*Ds)^{-1} in"""
    Ls = inp.ls
    a = complex(a)
    xpay = x is not None
    if on_gpu(out, inp):
        key = (tuple(complex(d) for d in diag), tuple(complex(h) for h in hop),
               float(mf), bool(dagger), Ls)
        t = _ztable_cache.get(key)
        if t is None:
            t = _ztables(Ls, diag, hop, float(mf), bool(dagger))
            if len(_ztable_cache) > 64:
                _ztable_cache.clear()
            _ztable_cache[key] = t
        ext = hip_ext()
        xf = x if x is not None else out
        ext.zdwf5(out.data, norm_or_empty(out), inp.data, norm_or_empty(inp),
                  xf.data, norm_or_empty(xf), inp.geo.volume_cb, Ls,
                  xpay, a.real, a.imag, kind,
                  t["au"], t["al"], t["wu"], t["wl"], t["su"], t["sl"],
                  t["ord_u"], t["ord_l"], t["diu"], t["eu"], t["dil"],
                  t["el"], t["cwu_re"], t["cwu_im"], t["cwl_re"],
                  t["cwl_im"])
        return out
    psi = inp.to_complex()[0]
    if kind == 0:
        res = ref.zdslash5(psi, Ls, diag, hop, mf, dagger)
        if xpay:
            res = a * x.to_complex()[0] + res
    else:
        res = a * ref.zm5inv(psi, Ls, diag, hop, mf, dagger)
        if xpay:
            res = x.to_complex()[0] + res
    out.from_complex(res.unsqueeze(0))
    return out


def eofa5_op(out: SpinorField, inp: SpinorField, alpha: float, beta: float,
             mf: float, kind: int, sh: float, pm: int, u, w,
             dagger: bool = False, a: float = 1.0,
             x: Optional[SpinorField] = None):
    """EOFA rank-1-extended 5th-dimension ops (csrc k_m5_eofa /
    k_m5inv_eofa; ref lib/dslash5_mobius_eofa.cu):
    kind=0: out = [a*x +] (alpha + beta Ds + sh P_pm |u><w|) in
    kind=1: out = [x +] a * (alpha + beta Ds + sh P_pm |u><w|)^{-1} in
    The dagger swaps u <-> w and daggers the base (host-side); for kind=1
    the kernel receives B^-1 u and sh/denom (Sherman-Morrison folded)."""
    import numpy as np
    Ls = inp.ls
    xpay = x is not None
    if on_gpu(out, inp):
        ext = hip_ext()
        xf = x if x is not None else out
        uu = [float(v) for v in u]
        ww = [float(v) for v in w]
        if dagger:
            uu, ww = ww, uu
        if kind == 1:
            # host Sherman-Morrison setup on the base bidiagonal+corner
            # (the rank-1 lives on chirality pm; the base block there is
            # the upper block for pm=+1)
            A = ref._m5_matrix(Ls, alpha, beta, mf, pm > 0, dagger)
            bu = np.linalg.solve(A, np.asarray(uu, dtype=float))
            den = 1.0 + sh * float(np.dot(ww, bu))
            uu = [float(v) for v in bu]
            sh_k = sh / den
        else:
            sh_k = sh
        ext.eofa5(out.data, norm_or_empty(out), inp.data, norm_or_empty(inp),
                  xf.data, norm_or_empty(xf), inp.geo.volume_cb, Ls, xpay,
                  bool(dagger), float(a), float(alpha), float(beta),
                  float(mf), uu, ww, float(sh_k), int(pm), kind)
        return out
    psi = inp.to_complex()[0]
    if kind == 0:
        res = ref.m5_eofa(psi, Ls, alpha, beta, mf, sh, pm, u, w, dagger)
        if xpay:
            res = a * x.to_complex()[0] + res
    else:
        res = a * ref.m5inv_eofa(psi, Ls, alpha, beta, mf, sh, pm, u, w,
                                 dagger)
        if xpay:
            res = x.to_complex()[0] + res
    out.from_complex(res.unsqueeze(0))
    return out


def dwf_halo_exchange(inp: SpinorField, parity_in: int, dagger: bool):
    """Pack + exchange all s-slice faces of a 5-d input (blocking; returns
    the halo object to hand to dslash_wilson_slice, or None when no dim is
    partitioned)."""
    from ..parallel import comms
    mask = comms.comm_mask()
    if not mask:
        return None
    if on_gpu(inp):
        from ..parallel.halo import get_dwf_halo
        h = get_dwf_halo(inp.geo, inp.precision, inp.device, mask, inp.ls)
        h.pack_exchange(hip_ext(), inp, parity_in, bool(dagger))
        return ("native", mask, h)
    from ..parallel.halo import exchange_psi5_oracle
    ghosts = exchange_psi5_oracle(inp.to_complex()[0], inp.geo, parity_in,
                                  mask, inp.ls)
    return ("oracle", mask, ghosts)


def dslash_wilson_slices(out: SpinorField, inp: SpinorField,
                         gauge: GaugeField, parity: int, dagger: bool = False,
                         a: float = 1.0, x: Optional[SpinorField] = None,
                         halo=None):
    """4-d Wilson hop on ALL Ls slices of 5-d fields, s-batched through
    the multi-RHS kernel: groups of slices share each gauge load (the
    domain-wall traffic fusion the reference gets from
    dslash_domain_wall_4d_fused_m5 — on MI355X the gauge stream, not the
    M5, is the fusible term; the s-batch amortizes it NRHS-fold)."""
    from ..parallel import comms
    geo = out.geo
    mask = comms.comm_mask()
    Ls = out.ls
    Vcb = geo.volume_cb
    gpu = on_gpu(out, inp)
    groups, rem0 = _mrhs_groups(Ls) if gpu else ([], 0)
    if gpu and groups:
        ext = hip_ext()
        xpay = x is not None
        xf = x if x is not None else out
        cl_t = torch.empty(0, dtype=out.data.dtype, device=out.device)
        for s0, gsz in groups:
            ghosts, nrms, face_cb, kt = [], [], [], 0
            if mask:
                kind, hmask, h = halo
                assert kind == "native" and hmask == mask
                ghosts, nrms, face_cb = h.ghost_args(s0)
                kt = 1
            ext.dslash_wilson_mrhs(
                [out.data] * gsz, [norm_or_empty(out)] * gsz,
                [inp.data] * gsz, [norm_or_empty(inp)] * gsz,
                gauge.data, cl_t, [xf.data] * gsz if xpay else [],
                [norm_or_empty(xf)] * gsz if xpay else [],
                list(geo.dims), geo.parity_offset, Vcb, parity,
                bool(dagger), PLAIN, xpay, float(a),
                RECON_COMPS[gauge.reconstruct], ghosts, nrms, face_cb,
                mask, kt, inp.volume_cb,
                [(s0 + r) * Vcb for r in range(gsz)])
    for s in range(rem0, Ls):
        dslash_wilson_slice(out, inp, gauge, parity, s, dagger, a=a, x=x,
                            halo=halo)
    return out


def dslash_wilson_slice(out: SpinorField, inp: SpinorField, gauge: GaugeField,
                        parity: int, s: int, dagger: bool = False,
                        a: float = 1.0, x: Optional[SpinorField] = None,
                        halo=None):
    """4-d Wilson hop on the s-th slice of 5-d fields (out/in/x all 5-d
    single-parity): out[s] = [x[s] +] a * (D in[s]). `halo` from
    dwf_halo_exchange(inp, 1-parity, dagger) when dims are partitioned."""
    from ..parallel import comms
    geo = out.geo
    mask = comms.comm_mask()
    if mask:
        assert halo is not None, "call dwf_halo_exchange first"
    xpay = x is not None
    Vcb = geo.volume_cb
    if on_gpu(out, inp):
        ext = hip_ext()
        xf = x if x is not None else out
        cl_t = torch.empty(0, dtype=out.data.dtype, device=out.device)
        ghosts, nrms, face_cb, kt = [], [], [], 0
        if mask:
            kind, hmask, h = halo
            assert kind == "native" and hmask == mask
            ghosts, nrms, face_cb = h.ghost_args(s)
            kt = 1  # fused ghost-aware kernel (exchange already done)
        ext.dslash_wilson(
            out.data, norm_or_empty(out), inp.data, norm_or_empty(inp),
            gauge.data, cl_t, xf.data, norm_or_empty(xf),
            list(geo.dims), geo.parity_offset, Vcb, parity, bool(dagger),
            PLAIN, xpay, float(a), RECON_COMPS[gauge.reconstruct], ghosts,
            nrms, face_cb, mask, kt, 0.0, 0.0, inp.volume_cb, s * Vcb)
        return out
    u = gauge.to_complex()
    sl = slice(s * Vcb, (s + 1) * Vcb)
    psi = inp.to_complex()[0][sl]
    oh = None
    if mask:
        kind, hmask, ghosts5 = halo
        assert kind == "oracle" and hmask == mask
        from ..parallel.halo import active_dims
        oh = {
            "mask": mask,
            "psi": {k: v[s] for k, v in ghosts5.items()},
            "u_bwd": {mu: gauge.bwd_ghost(mu, parity)
                      for mu in active_dims(mask)},
        }
    res = ref.dslash_wilson_parity(u, psi, geo, parity, dagger, halo=oh)
    if xpay:
        res = x.to_complex()[0][sl] + a * res
    else:
        res = a * res
    oc = out.to_complex()
    oc[0][sl] = res
    out.from_complex(oc)
    return out
