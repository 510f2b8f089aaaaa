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

SIZEOF = {"double": 8, "single": 4, "half": 2}


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


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--lattice", default="32,32,32,64")
    ap.add_argument("--reps", type=int, default=100)
    args = ap.parse_args()
    dims = tuple(int(x) for x in args.lattice.split(","))
    geo = LatticeGeometry(dims)
    gen = torch.Generator().manual_seed(9)
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen, dtype=torch.float32)
    from quda_amd.fields.gauge import project_su3
    u = project_su3(torch.view_as_complex(m.to(torch.float64)).cuda())
    A = ref.clover_matrix(u, geo, 0.135, 1.0)
    results = {}
    for prec, recon in [("double", "none"), ("single", "none"),
                        ("single", "twelve"), ("half", "twelve")]:
        for mode, mname in [(PLAIN, "wilson"), (CLOV_POST, "wilson_clover")]:
            r = run(prec, recon, mode, geo, u, A, args.reps)
            key = f"{mname}/{prec}/r{18 if recon == 'none' else 12}"
            results[key] = r
            print(f"{key:32s} {r['us']:8.1f} us  {r['gflops']:8.0f} GFLOPS  "
                  f"{r['gbs']:7.0f} GB/s(model)", flush=True)
    print(json.dumps({k: round(v["gflops"]) for k, v in results.items()}))


if __name__ == "__main__":
    main()
