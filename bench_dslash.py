#!/usr/bin/env python3
"""Dslash microbenchmark: per-precision Wilson(-clover) stencil GFLOPS and
effective bandwidth on one GPU (the reference prints the same metric from
tests/dslash_test_utils.h:1048).

Usage: python bench_dslash.py [--lattice 32,32,32,64] [--reps 100]
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from quda_amd import GaugeField, LatticeGeometry, SpinorField  # noqa: E402
from quda_amd.fields.clover import CloverField, pack_clover  # noqa: E402
from quda_amd.ops import reference as ref  # noqa: E402
from quda_amd.ops.dispatch import CLOV_POST, PLAIN, dslash_wilson  # noqa: E402

SIZEOF = {"double": 8, "single": 4, "half": 2, "quarter": 1}


def run(prec, recon_name, mode, geo, u, A, reps):
    dev = "cuda"
    g = GaugeField(geo, prec, dev, reconstruct=recon_name).from_complex(u)
    s_in = SpinorField(geo, prec, dev, n_parity=1).gaussian_(seed=5)
    s_out = SpinorField(geo, prec, dev, n_parity=1)
    cl = None
    if mode == CLOV_POST:
        cl = CloverField(geo, prec, dev)
        cl.data.copy_(cl._to_native(pack_clover(A)))
    # warmup
    for _ in range(5):
        dslash_wilson(s_out, s_in, g, 0, mode=mode, clover=cl)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        dslash_wilson(s_out, s_in, g, 0, mode=mode, clover=cl)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps
    V = geo.volume_cb
    flops = V * (1320 + (504 if mode == CLOV_POST else 0))
    # bytes: unique in-spinor (V), out (V), gauge 8 links, clover, norms
    el = SIZEOF[prec]
    L = 18 if recon_name == "none" else 12
    nb = 4 if prec == "half" else 0
    bytes_ = V * (24 * el + nb) * 2 + V * 8 * L * el
    if mode == CLOV_POST:
        bytes_ += V * 72 * el
    return dict(us=dt * 1e6, gflops=flops / dt / 1e9, gbs=bytes_ / dt / 1e9)


def sweep(geo, u, reps):
    """(block x waves) launch-config sweep of the hot Wilson kernels."""
    from quda_amd.ops.dispatch import hip_ext as _ext
    A = ref.clover_matrix(u, geo, 0.135, 1.0)
    results = {}
    for waves in (0, 3):
        _ext().set_dslash_waves(waves)
        blocks = [64] if waves == 3 else [64, 128, 256]
        for blk in blocks:
            _ext().set_dslash_block(blk)
            recons = ([("half", "twelve"), ("half", "eight"),
                       ("single", "twelve"), ("single", "eight"),
                       ("quarter", "eight")] if (waves == 0 and blk == 64)
                      else [("half", "twelve"), ("single", "twelve")])
            for prec, recon in recons:
                for mode, mname in [(PLAIN, "wilson"),
                                    (CLOV_POST, "wilson_clover")]:
                    r = run(prec, recon, mode, geo, u, A, reps)
                    rc = {"twelve": 12, "eight": 8, "none": 18}[recon]
                    key = f"{mname}/{prec}/r{rc}/b{blk}w{waves}"
                    results[key] = round(r["gflops"])
                    print(f"{key:40s} {r['us']:8.1f} us  {r['gflops']:8.0f} "
                          f"GFLOPS", flush=True)
    _ext().set_dslash_waves(0)
    _ext().set_dslash_block(64)
    print(json.dumps(results))
    return results


def lds_family(geo, u, reps):
    """LDS-tiled kernel (k_dslash_wilson_lds) vs the gather-first default."""
    from quda_amd.ops.dispatch import hip_ext as _ext
    A = ref.clover_matrix(u, geo, 0.135, 1.0)
    results = {}
    for lds in (0, 1):
        _ext().set_dslash_lds(lds)
        for prec, recon in [("half", "twelve"), ("half", "none"),
                            ("quarter", "twelve")]:
            for mode, mname in [(PLAIN, "wilson"), (CLOV_POST, "wilson_clover")]:
                r = run(prec, recon, mode, geo, u, A, reps)
                rc = {"twelve": 12, "eight": 8, "none": 18}[recon]
                key = f"{mname}/{prec}/r{rc}/lds{lds}"
                results[key] = round(r["gflops"])
                print(f"{key:40s} {r['us']:8.1f} us  {r['gflops']:8.0f} "
                      f"GFLOPS", flush=True)
    _ext().set_dslash_lds(0)
    print(json.dumps(results))
    return results


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--lattice", default="32,32,32,64")
    ap.add_argument("--reps", type=int, default=100)
    ap.add_argument("--families", default="wilson",
                    help="comma list: wilson,staggered,mobius,dwf5,all,sweep")
    args = ap.parse_args()
    fams = set(args.families.split(","))
    if "all" in fams:
        fams = {"wilson", "staggered", "mobius"}
    dims = tuple(int(x) for x in args.lattice.split(","))
    geo = LatticeGeometry(dims)
    gen = torch.Generator().manual_seed(9)
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen, dtype=torch.float32)
    from quda_amd.fields.gauge import project_su3
    u = project_su3(torch.view_as_complex(m.to(torch.float64)).cuda())
    results = {}
    if "sweep" in fams:
        results.update(sweep(geo, u, args.reps))
        fams.discard("sweep")
    if "lds" in fams:
        results.update(lds_family(geo, u, args.reps))
        fams.discard("lds")
        if not fams:
            return
    if "wilson" in fams:
        A = ref.clover_matrix(u, geo, 0.135, 1.0)
        for prec, recon in [("double", "none"), ("single", "none"),
                            ("single", "twelve"), ("half", "twelve")]:
            for mode, mname in [(PLAIN, "wilson"), (CLOV_POST, "wilson_clover")]:
                r = run(prec, recon, mode, geo, u, A, args.reps)
                key = f"{mname}/{prec}/r{18 if recon == 'none' else 12}"
                results[key] = r["gflops"]
                print(f"{key:32s} {r['us']:8.1f} us  {r['gflops']:8.0f} GFLOPS  "
                      f"{r['gbs']:7.0f} GB/s(model)", flush=True)
    if "staggered" in fams:
        results.update(bench_staggered(geo, u, args.reps))
    if "mobius" in fams:
        results.update(bench_mobius(geo, u, max(args.reps // 5, 3)))
    if "dwf5" in fams:
        results.update(bench_dwf5(geo, max(args.reps, 20)))
    if "mrhs" in fams:
        results.update(bench_mrhs(geo, u, max(args.reps // 2, 10)))
    print(json.dumps({k: round(v) for k, v in results.items()}))


def bench_staggered(geo, u, reps):
    """Naive + improved staggered rates (ref staggered flop model 570/site
    naive, 1146 improved)."""
    from quda_amd.gauge.hisq import asqtad_coefficients, fat_links, naik_links
    from quda_amd.ops.dispatch import dslash_staggered
    dev = "cuda"
    out = {}
    for prec in ("single", "half"):
        g = GaugeField(geo, prec, dev).from_complex(u)
        s_in = SpinorField(geo, prec, dev, n_parity=1, nspin=1).gaussian_(seed=7)
        s_out = SpinorField(geo, prec, dev, n_parity=1, nspin=1)
        for _ in range(5):
            dslash_staggered(s_out, s_in, g, 0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            dslash_staggered(s_out, s_in, g, 0)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / reps
        gf = geo.volume_cb * 570 / dt / 1e9
        key = f"staggered/{prec}/r18"
        out[key] = gf
        print(f"{key:32s} {dt*1e6:8.1f} us  {gf:8.0f} GFLOPS", flush=True)
    # improved (fat+long)
    fat = fat_links(u, geo, asqtad_coefficients())
    lng = naik_links(u, geo)
    for prec in ("single",):
        gf_ = GaugeField(geo, prec, dev).from_complex(fat)
        gl = GaugeField(geo, prec, dev, shift=3).from_complex(lng)
        s_in = SpinorField(geo, prec, dev, n_parity=1, nspin=1).gaussian_(seed=8)
        s_out = SpinorField(geo, prec, dev, n_parity=1, nspin=1)
        for _ in range(5):
            dslash_staggered(s_out, s_in, gf_, 0, long_gauge=gl)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            dslash_staggered(s_out, s_in, gf_, 0, long_gauge=gl)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / reps
        gf2 = geo.volume_cb * 1146 / dt / 1e9
        key = f"hisq/{prec}/r18"
        out[key] = gf2
        print(f"{key:32s} {dt*1e6:8.1f} us  {gf2:8.0f} GFLOPS", flush=True)
    return out


def bench_dwf5(geo, reps, Ls=12):
    """5th-dimension kernel rates: M5 apply/inverse, zMobius, EOFA
    (bandwidth-bound s-structure ops; useful to confirm they stay
    negligible next to the 4-d hops)."""
    from quda_amd.ops.dispatch import dwf5_op, eofa5_op, zdwf5_op
    dev = "cuda"
    out = {}
    V5 = geo.volume_cb * Ls
    for prec in ("double", "single"):
        a = SpinorField(geo, prec, dev, n_parity=1, ls=Ls).gaussian_(seed=7)
        o = SpinorField(geo, prec, dev, n_parity=1, ls=Ls)
        zb = [1.5 + 0.05j * s for s in range(Ls)]
        zc = [0.5 - 0.03j * s for s in range(Ls)]
        import math
        kap = 0.5
        eu = [kap ** s for s in range(Ls)]
        n = math.sqrt(sum(x * x for x in eu))
        eu = [x / n for x in eu]
        cases = {
            "m5": lambda: dwf5_op(o, a, 1.8, -0.5, 0.04, kind=0),
            "m5inv": lambda: dwf5_op(o, a, 1.8, -0.5, 0.04, kind=1),
            "zm5": lambda: zdwf5_op(o, a, zb, zc, 0.04, 0),
            "zm5inv": lambda: zdwf5_op(o, a, zb, zc, 0.04, 1),
            "eofa_m5": lambda: eofa5_op(o, a, 1.8, -0.5, 0.04, 0, -0.2, 1,
                                        eu, eu),
            "eofa_m5inv": lambda: eofa5_op(o, a, 1.8, -0.5, 0.04, 1, -0.2,
                                           1, eu, eu),
        }
        bytes_per = {"double": 8, "single": 4}[prec] * 24 * 2  # r+w
        for name, fn in cases.items():
            for _ in range(3):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(reps):
                fn()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / reps
            out[f"{name}_Ls{Ls}/{prec}"] = {
                "us": dt * 1e6,
                "GBps": V5 * bytes_per / dt / 1e9,
            }
    return out


def bench_mobius(geo, u, reps, Ls=12):
    """Moebius full-operator application rate (4-d hops x Ls + s-structure;
    per-slice Wilson flop model + M5 ops)."""
    from quda_amd.models.dwf import DiracMobius
    dev = "cuda"
    out = {}
    for prec in ("single",):
        g = GaugeField(geo, prec, dev).from_complex(u)
        d = DiracMobius(g, 0.04, 1.8, Ls)
        psi = SpinorField(geo, prec, dev, ls=Ls).gaussian_(seed=9)
        res = SpinorField(geo, prec, dev, ls=Ls)
        for _ in range(3):
            d.M(res, psi)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            d.M(res, psi)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / reps
        gf = geo.volume * Ls * (1320 + 96) / dt / 1e9
        key = f"mobius_M_Ls{Ls}/{prec}"
        out[key] = gf
        print(f"{key:32s} {dt*1e6:8.1f} us  {gf:8.0f} GFLOPS", flush=True)
    return out




def bench_mrhs(geo, u, reps):
    """Kernel-level multi-RHS amortization: 8 RHS as groups of 4/2 vs
    per-RHS launches (gauge+clover loads shared across the group)."""
    import quda_amd.ops.dispatch as dsp
    from quda_amd.ops.dispatch import dslash_wilson_batch
    dev = "cuda"
    out = {}
    n = 8
    for prec in ("half", "single"):
        g = GaugeField(geo, prec, dev, reconstruct="twelve").from_complex(u)
        inps = [SpinorField(geo, prec, dev, n_parity=1).gaussian_(seed=40 + r)
                for r in range(n)]
        outs = [SpinorField(geo, prec, dev, n_parity=1) for _ in range(n)]
        for gsz in (1, 2, 4):
            dsp._MRHS_GROUP = gsz
            for _ in range(3):
                dslash_wilson_batch(outs, inps, g, 0)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(reps):
                dslash_wilson_batch(outs, inps, g, 0)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / reps
            gf = n * geo.volume_cb * 1320 / dt / 1e9
            key = f"mrhs{gsz}/{prec}"
            out[key] = round(gf)
            print(f"{key:32s} {dt*1e6:8.1f} us ({n} rhs) {gf:8.0f} GFLOPS",
                  flush=True)
        dsp._MRHS_GROUP = None
    print(json.dumps(out))
    return out


if __name__ == "__main__":
    main()
