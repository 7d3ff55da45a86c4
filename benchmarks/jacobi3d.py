#!/usr/bin/env python3
"""jacobi3d CLI (reference: bin/jacobi3d.cu / bin/jacobi3d_strong.cu).

Weak scaling (default): per-GPU domain scaled up with GPU count.
Strong scaling (--strong): fixed total domain.
CSV row matches the reference schema in spirit:
  jacobi3d,<methods>,ranks,gpus,x,y,z,<bytes_kernel>,<bytes_rccl>,min,trimean
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

import stencil_amd as sa
from stencil_amd.models.jacobi3d import Jacobi3D
from stencil_amd.parallel.placement import PlacementStrategy
from stencil_amd.utils.statistics import Statistics


def weak_dims(n):
    from stencil_amd import prime_factors

    d = [1, 1, 1]
    for f in prime_factors(n):
        d[d.index(min(d))] *= f
    return sorted(d, reverse=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--size", type=int, default=512, help="per-GPU (weak) or total (strong) edge")
    ap.add_argument("--strong", action="store_true")
    ap.add_argument("--radius", type=int, default=1)
    ap.add_argument("--halo-multiplier", type=int, default=1,
                    help="exchange every m-th step with m*radius halos (temporal blocking)")
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--trivial", action="store_true", help="trivial placement")
    ap.add_argument("--no-overlap", action="store_true")
    ap.add_argument("--paraview", action="store_true")
    ap.add_argument("--period", type=int, default=-1, help="paraview dump every N iters")
    ap.add_argument("--prefix", default="jacobi3d_")
    ap.add_argument("--backend", default="native")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        import torch
        import torch.distributed as dist

        torch.cuda.set_device(
            int(os.environ.get("LOCAL_RANK", rank)) % max(1, torch.cuda.device_count())
        )
        dist.init_process_group(backend="cpu:gloo,cuda:nccl")
        n, gpus = world, None
    else:
        n = args.gpus
        gpus = list(range(n)) if args.backend == "native" else [0] * n

    if args.strong:
        size = (args.size,) * 3
    else:
        dims = weak_dims(n)
        size = tuple(args.size * d for d in dims)

    placement = PlacementStrategy.Trivial if args.trivial else PlacementStrategy.NodeAware
    app = Jacobi3D(size, backend=args.backend, gpus=gpus, placement=placement,
                   radius=args.radius, halo_multiplier=args.halo_multiplier)
    app.realize()

    stats = Statistics()
    for i in range(args.iters):
        if world > 1:
            import torch.distributed as dist

            dist.barrier()
        t0 = time.perf_counter()
        app.step(overlap=not args.no_overlap)
        stats.insert(time.perf_counter() - t0)
        if args.paraview and args.period > 0 and (i + 1) % args.period == 0:
            app.dd.write_paraview(f"{args.prefix}iter{i + 1}_")

    mn, tm = stats.min(), stats.trimean()
    if world > 1:
        import torch
        import torch.distributed as dist

        t = torch.tensor([mn, tm], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        mn, tm = float(t[0]), float(t[1])
        dist.destroy_process_group()

    if rank == 0:
        bk = app.dd.bytes_by_method["direct_kernel"]
        br = app.dd.bytes_by_method["rccl"]
        mode = "strong" if args.strong else "weak"
        print(
            f"jacobi3d,{mode},ranks={world},gpus={n},{size[0]},{size[1]},{size[2]},"
            f"bytes_kernel={bk},bytes_rccl={br},min={mn:.6f},trimean={tm:.6f}",
            flush=True,
        )


if __name__ == "__main__":
    main()
