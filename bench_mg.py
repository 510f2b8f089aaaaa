#!/usr/bin/env python3
"""Multigrid benchmark (BASELINE config 4 shape, single GPU): Wilson-clover
near-critical solve, MG-preconditioned GCR vs plain GCR — reports setup
time, iteration counts and time-to-solution (the reference's
multigrid_benchmark_test role)."""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from quda_amd import GaugeField, LatticeGeometry, SpinorField  # noqa: E402
from quda_amd.fields.clover import CloverField  # noqa: E402
from quda_amd.mg import MG, MGParam  # noqa: E402
from quda_amd.models import DiracClover  # noqa: E402
from quda_amd.ops import blas  # noqa: E402
from quda_amd.ops import reference as ref  # noqa: E402
from quda_amd.solvers import gcr_solve  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--lattice", default="16,16,16,32")
    ap.add_argument("--kappa", type=float, default=0.1245)
    ap.add_argument("--kappa-list", default=None,
                    help="comma list: sweep kappas on ONE thermalized "
                         "field (amortizes the heatbath cost)")
    ap.add_argument("--smear", type=int, default=0,
                    help="stout steps on the random field")
    ap.add_argument("--therm", type=int, default=0,
                    help="heatbath(+3 OR) thermalization iterations at "
                         "--beta from a cold start (0 = random field)")
    ap.add_argument("--beta", type=float, default=6.0)
    ap.add_argument("--block", default="4,4,4,4")
    ap.add_argument("--nvec", type=int, default=8)
    ap.add_argument("--levels", type=int, default=2)
    ap.add_argument("--nvec2", type=int, default=8)
    ap.add_argument("--tol", type=float, default=1e-8)
    ap.add_argument("--device", default=None)
    ap.add_argument("--smooth-sweep", action="store_true",
                    help="after one MG setup, sweep (nu_post, coarse_tol) "
                         "V-cycle configs reusing the same null vectors")
    args = ap.parse_args()
    dev = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    dims = tuple(int(x) for x in args.lattice.split(","))
    geo = LatticeGeometry(dims)
    gen = torch.Generator().manual_seed(77)
    from quda_amd.fields.gauge import project_su3
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float32)
    if args.therm:
        from quda_amd.gauge import heatbath_sweep, overrelax_sweep, plaquette
        if dev != "cpu":
            from quda_amd.gauge.heatbath import set_device_rng
            set_device_rng(True)
        u = GaugeField(geo, "double", dev).unit_().to_complex()
        t0 = time.perf_counter()
        for it in range(args.therm):
            u = heatbath_sweep(u, geo, args.beta, seed=1000 + 7 * it)
            for j in range(3):
                u = overrelax_sweep(u, geo, args.beta, seed=5000 + 13 * it + j)
        p, _, _ = plaquette(u, geo)
        print(f"# thermalized {args.therm} it at beta={args.beta}: "
              f"plaq={p:.4f} ({time.perf_counter()-t0:.0f}s)", flush=True)
    else:
        u = project_su3(torch.view_as_complex(m.to(torch.float64)).to(dev))
        if args.smear:
            from quda_amd.gauge import stout_smear
            u = stout_smear(u, geo, 0.12, args.smear)
    kappas = ([float(x) for x in args.kappa_list.split(",")]
              if args.kappa_list else [args.kappa])

    def sync():
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    import math
    g = GaugeField(geo, "double", dev).from_complex(u)
    for kappa in kappas:
        A = ref.clover_matrix(u, geo, kappa, 1.0)
        cl = CloverField(geo, "double", dev).from_matrices(A)
        d = DiracClover(g, cl, kappa)
        b = SpinorField(geo, "double", dev).gaussian_(seed=78)

        t0 = time.perf_counter()
        mg = MG(d, MGParam(block=tuple(int(x) for x in args.block.split(",")),
                           n_vec=args.nvec, nu_post=4, coarse_tol=5e-2,
                           levels=args.levels, n_vec2=args.nvec2,
                           block2=(2, 2, 2, 2),
                           null_tol=1e-4, null_maxiter=300))
        sync()
        t_setup = time.perf_counter() - t0

        x0 = SpinorField(geo, "double", dev)
        t0 = time.perf_counter()
        st_plain = gcr_solve(d, x0, b, tol=args.tol, maxiter=2000,
                             nkrylov=24)
        sync()
        t_plain = time.perf_counter() - t0

        if args.smooth_sweep:
            import dataclasses
            for nu_post, ctol, cmax in [(4, 5e-2, 200), (2, 5e-2, 200),
                                        (4, 1e-1, 100), (2, 1e-1, 100),
                                        (1, 1e-1, 100), (2, 2e-1, 60),
                                        (1, 2e-1, 60)]:
                mg.param = dataclasses.replace(mg.param, nu_post=nu_post,
                                               coarse_tol=ctol,
                                               coarse_maxiter=cmax)
                xs = SpinorField(geo, "double", dev)
                t0 = time.perf_counter()
                st = gcr_solve(d, xs, b, tol=args.tol, maxiter=2000,
                               nkrylov=24, precond=mg.precond)
                sync()
                print(json.dumps({
                    "sweep": {"nu_post": nu_post, "coarse_tol": ctol,
                              "coarse_maxiter": cmax},
                    "kappa": kappa, "iters": st.iters,
                    "secs": round(time.perf_counter() - t0, 2),
                    "converged": st.converged}), flush=True)
            continue

        x1 = SpinorField(geo, "double", dev)
        t0 = time.perf_counter()
        st_mg = gcr_solve(d, x1, b, tol=args.tol, maxiter=2000, nkrylov=24,
                          precond=mg.precond)
        sync()
        t_mg = time.perf_counter() - t0
        r = SpinorField(geo, "double", dev)
        d.M(r, x1)
        tr = math.sqrt(blas.xmy_norm2(b, r) / blas.norm2(b))
        print(json.dumps({
            "metric": "mg_gcr_speedup",
            "lattice": "x".join(map(str, dims)), "kappa": kappa,
            "mg_setup_s": round(t_setup, 2),
            "plain_gcr": {"iters": st_plain.iters, "secs": round(t_plain, 2),
                          "converged": st_plain.converged},
            "mg_gcr": {"iters": st_mg.iters, "secs": round(t_mg, 2),
                       "converged": st_mg.converged, "true_res": tr},
            "iter_reduction": round(st_plain.iters / max(st_mg.iters, 1), 1),
        }), flush=True)


if __name__ == "__main__":
    main()
