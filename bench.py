#!/usr/bin/env python3
"""Flagship benchmark: Wilson-clover even-odd-preconditioned CG,
32^3 x 64 local lattice, double-outer / half-sloppy mixed precision —
the BASELINE.json headline config ("Wilson-clover Dslash GFLOPS/GPU +
CG time-to-solution, 32^3x64 mixed-prec").

One "step" = `--iters` mixed-precision CG iterations of the hot loop
(fixed work; the production solver path, nothing skipped). The metric is
aggregate sustained GFLOPS over the whole job computed from the reference
flop model (ref: include/dslash.h:467 — Wilson 1320 flop/site, clover 504;
blas/reduction flops counted per op), on synthetic data: random SU(3)
gauge + gaussian source, random-init = same protocol as the reference's
tests (tests/utils/host_utils.cpp constructRandomGaugeField).

Weak scaling: each rank owns a full 32^3x64 local lattice; ranks are laid
out along T. (Multi-rank path requires the halo exchange engine; N=1 is
the single-GPU path.)
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from quda_amd import GaugeField, LatticeGeometry, SpinorField  # noqa: E402
from quda_amd.fields.clover import CloverField, pack_clover  # noqa: E402
from quda_amd.models import DiracCloverPC  # noqa: E402
from quda_amd.ops import blas  # noqa: E402
from quda_amd.ops import reference as ref  # noqa: E402
from quda_amd.parallel import comms  # noqa: E402
from quda_amd.solvers import cg_solve  # noqa: E402

KAPPA = 0.135
CSW = 1.0


def setup_fields(geo, device, sloppy_prec, seed):
    """Random SU(3) gauge + clover at double and sloppy precision."""
    g = GaugeField(geo, "double", device)
    # generate directly on device (host gen at 32^3x64 is minutes of QR)
    gen = torch.Generator(device="cpu")
    gen.manual_seed(seed)
    m = torch.randn((4, 2, geo.volume_cb, 3, 3, 2), generator=gen,
                    dtype=torch.float32)
    from quda_amd.fields.gauge import project_su3
    u = project_su3(torch.view_as_complex(m.to(torch.float64)).to(device))
    g.from_complex(u)
    A = ref.clover_matrix(u, geo, KAPPA, CSW)
    cl = CloverField(geo, "double", device).from_matrices(A)
    # sloppy copies
    recon = "twelve" if sloppy_prec != "double" else "none"
    gs = GaugeField(geo, sloppy_prec, device, reconstruct=recon).from_complex(u)
    cls = CloverField(geo, sloppy_prec, device)
    cls.data.copy_(cls._to_native(pack_clover(A)))
    cls.inv_data.copy_(cls._to_native(pack_clover(cl.to_complex(inverse=True).to(device))))
    return g, cl, gs, cls


def flops_per_iter(Vcb: int) -> float:
    """Flop count of one CG iteration on the even-odd clover system.

    MdagM = M (2 fused dslash-clover launches: 2*(1320+504+48 xpay/scale))
          + Mdag (clover-inv 504 + fused 1824 + dslash 1320 + 48)
    blas  = re_dot 48, axpy 48, axpy_norm2 96, xpay 48   (2 flop/real=24/site *2)
    """
    mdagm = (2 * (1320 + 504) + 48) + (504 + 1824 + 1320 + 48)
    blas_f = 48 + 48 + 96 + 48
    return float(Vcb) * (mdagm + blas_f)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--iters", type=int, default=25,
                    help="CG iterations per step (fixed work)")
    ap.add_argument("--lattice", type=str, default="32,32,32,64")
    ap.add_argument("--sloppy", type=str, default="half",
                    choices=["double", "single", "half"])
    ap.add_argument("--seed", type=int, default=777)
    ap.add_argument("--device", default=None,
                    help="override (cpu for harness tests; default cuda:LOCAL_RANK)")
    args = ap.parse_args()

    dims = tuple(int(d) for d in args.lattice.split(","))
    rank, world = comms.init_comms()
    assert world == args.gpus or "WORLD_SIZE" not in os.environ, \
        f"world {world} != --gpus {args.gpus}"
    if args.device:
        device = args.device
    else:
        device = f"cuda:{int(os.environ.get('LOCAL_RANK', 0))}"
        torch.cuda.set_device(device)

    geo = LatticeGeometry(dims, parity_offset=comms.parity_offset_of_rank(dims))
    g, cl, gs, cls = setup_fields(geo, device, args.sloppy, args.seed + rank)
    d = DiracCloverPC(g, cl, KAPPA)
    ds = DiracCloverPC(gs, cls, KAPPA)

    b = SpinorField(geo, "double", device, n_parity=1).gaussian_(seed=args.seed + 100 + rank)
    x = SpinorField(geo, "double", device, n_parity=1)

    def step():
        x.zero_()
        # fixed-iteration mixed-precision CG hot loop (tol=0: never exits
        # early; delta tiny: reliable updates off at this iteration count,
        # matching production behaviour for <50 iters between updates)
        cg_solve(d, x, b, op_sloppy=ds, sloppy=args.sloppy, tol=0.0,
                 maxiter=args.iters, delta=1e-30)

    for _ in range(args.warmup):
        step()
    comms.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    comms.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    # max over ranks
    if comms.is_distributed():
        import torch.distributed as dist
        dev = "cuda" if dist.get_backend() == "nccl" else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    ms_per_step = elapsed / args.steps * 1000.0
    total_flops = flops_per_iter(geo.volume_cb) * args.iters * args.steps * world
    gflops = total_flops / elapsed / 1e9

    if rank == 0:
        out = {
            "metric": "wilson_clover_cg_gflops",
            "value": round(gflops, 3),
            "unit": "GFLOPS",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "double-half" if args.sloppy == "half" else f"double-{args.sloppy}",
            "data": "synthetic",
            "config": {
                "model": "wilson_clover_eo_pc_cg",
                "lattice_per_gpu": "x".join(str(v) for v in dims),
                "global_batch": args.iters,
                "seq_len": geo.volume,
                "kappa": KAPPA,
                "csw": CSW,
                "cg_iters_per_step": args.iters,
                "parallelism": f"dd_t{world}",
                "gauge_recon_sloppy": 12 if args.sloppy != "double" else 18,
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
