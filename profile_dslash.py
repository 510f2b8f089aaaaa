#!/usr/bin/env python3
"""Minimal dslash profiling target: random-data fields (no SU(3)
projection, no oracle clover — numerics don't matter for counters), only
quda_amd HIP kernels in the measured region. Keeps rocprofv3 --pmc happy
(torch reduce kernels crash under counter collection)."""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from quda_amd import GaugeField, LatticeGeometry, SpinorField  # noqa: E402
from quda_amd.fields.clover import CloverField  # noqa: E402
from quda_amd.ops.dispatch import CLOV_POST, PLAIN, dslash_wilson  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--lattice", default="32,32,32,64")
    ap.add_argument("--reps", type=int, default=20)
    ap.add_argument("--prec", default="half")
    ap.add_argument("--recon", default="twelve")
    ap.add_argument("--mode", default="both", choices=["wilson", "clover", "both"])
    args = ap.parse_args()
    dims = tuple(int(x) for x in args.lattice.split(","))
    geo = LatticeGeometry(dims)
    dev = "cuda"
    g = GaugeField(geo, args.prec, dev, reconstruct=args.recon)
    g.data.uniform_(-0.5, 0.5)
    s_in = SpinorField(geo, args.prec, dev, n_parity=1)
    s_in.data.uniform_(-0.5, 0.5)
    if s_in.norm is not None:
        s_in.norm.fill_(1.0)
    s_out = SpinorField(geo, args.prec, dev, n_parity=1)
    cl = CloverField(geo, args.prec, dev)
    cl.data.uniform_(-0.5, 0.5)

    modes = []
    if args.mode in ("wilson", "both"):
        modes.append(("wilson", PLAIN, None))
    if args.mode in ("clover", "both"):
        modes.append(("clover", CLOV_POST, cl))
    for name, mode, clov in modes:
        for _ in range(3):
            dslash_wilson(s_out, s_in, g, 0, mode=mode, clover=clov)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.reps):
            dslash_wilson(s_out, s_in, g, 0, mode=mode, clover=clov)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.reps
        flops = geo.volume_cb * (1320 + (504 if mode == CLOV_POST else 0))
        print(f"{name}/{args.prec}: {dt*1e6:.1f} us  {flops/dt/1e9:.0f} GFLOPS",
              flush=True)


if __name__ == "__main__":
    main()
