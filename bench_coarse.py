#!/usr/bin/env python3
"""Coarse-dslash microbenchmark: k_coarse_dslash_mfma (f32 matrix cores,
csrc/coarse.hip) vs the torch einsum/rocBLAS path on the same synthetic
coarse tensors (the reference's dslash_coarse_mma vs dslash_coarse
comparison, kernels/dslash_coarse_mma.cuh:755).

Usage: python bench_coarse.py [--na 8192] [--nc 48] [--reps 200]
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from quda_amd.mg.coarse import CoarseOp  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cd", default="16,16,16,8",
                    help="coarse dims (Na = product)")
    ap.add_argument("--nc", type=int, default=48)
    ap.add_argument("--reps", type=int, default=100)
    args = ap.parse_args()
    cd = tuple(int(x) for x in args.cd.split(","))
    Na = cd[0] * cd[1] * cd[2] * cd[3]
    Nc = args.nc
    gen = torch.Generator().manual_seed(11)

    def rnd(*shape):
        return torch.view_as_complex(
            torch.randn((*shape, 2), generator=gen, dtype=torch.float32)
        ).to(torch.complex64).cuda()

    X = rnd(Na, Nc, Nc)
    Y = [rnd(Na, Nc, Nc) for _ in range(8)]
    results = {}
    for nr in (1, 4, 8, 16):
        C = rnd(Na, Nc, nr)
        co = CoarseOp(X, Y, cd)
        out = {}
        for name, use_hip in (("mfma", True), ("torch", False)):
            co.use_hip = use_hip
            co._m9 = None  # rebuild lazily
            f = (lambda: co.apply_block(C)) if nr > 1 else \
                (lambda: co.apply(C[:, :, 0]))
            for _ in range(5):
                f()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.reps):
                f()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.reps
            flops = 9 * Na * Nc * Nc * nr * 8
            bytes_ = 9 * Na * Nc * Nc * 8 + 2 * Na * Nc * nr * 8
            out[name] = dt
            print(f"nc{Nc}/nr{nr}/{name:6s} {dt*1e6:9.1f} us "
                  f"{flops/dt/1e12:7.2f} TFLOP/s  {bytes_/dt/1e9:7.0f} GB/s",
                  flush=True)
        results[f"nr{nr}_speedup"] = round(out["torch"] / out["mfma"], 2)
    print(json.dumps(results))


if __name__ == "__main__":
    main()
