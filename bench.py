#!/usr/bin/env python3
"""Flagship benchmark: jacobi3d weak scaling, 750^3 grid points per GPU,
fp32, radius 1 (the reference's headline configuration; BASELINE.md:
scripts/summit/1node_jacobi3d.sh). Reports aggregate cell-updates/s.

Single process N GPUs:   python bench.py --gpus N
One rank per GPU (driver): python -m torch.distributed.run --nnodes=1
  --nproc-per-node N ... bench.py --gpus N --steps K --warmup W
"""
import argparse
import json
import os

import time


def weak_dims(n):
    """split factor per axis for n GPUs (largest-prime-first, smallest axis)"""
    from stencil_amd import prime_factors

    d = [1, 1, 1]
    for f in prime_factors(n):
        d[d.index(min(d))] *= f
    return sorted(d, reverse=True)  # x gets the largest factor


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--per-gpu", type=int, default=750, help="per-GPU edge length")
    ap.add_argument("--no-overlap", action="store_true")
    ap.add_argument("--halo-multiplier", type=int, default=1,
                    help="temporal blocking: exchange every m-th step with m-deep halos "
                    "(measured -7%% step time at m=2 on 1 GPU; default 1 = the "
                    "reference's per-step-exchange configuration)")
    ap.add_argument("--backend", default="native")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    dist = None
    ctrl = None  # gloo control group (barriers/reductions off the GPU)
    if world > 1:
        import torch
        import torch.distributed as dist_mod

        dist = dist_mod
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
        dist.init_process_group(backend="cpu:gloo,cuda:nccl")
        ctrl = dist.new_group(backend="gloo")
        n_gpus = world
        gpus = None  # DistributedDomain defaults to [LOCAL_RANK]
    else:
        n_gpus = args.gpus
        gpus = list(range(n_gpus)) if args.backend == "native" else [0] * n_gpus

    dims = weak_dims(n_gpus)
    size = tuple(args.per_gpu * d for d in dims)

    from stencil_amd.models.jacobi3d import Jacobi3D

    app = Jacobi3D(size, backend=args.backend, gpus=gpus,
                   halo_multiplier=args.halo_multiplier)
    app.realize()

    # check weak-scaling shape: every GPU must hold exactly per_gpu^3 cells
    for li in range(app.dd.num_local()):
        lo, hi = app.dd.local_rect(li)
        vol = (hi[0] - lo[0]) * (hi[1] - lo[1]) * (hi[2] - lo[2])
        assert vol == args.per_gpu ** 3, f"unequal split: {lo}..{hi}"

    def barrier_sync():
        if dist is not None:
            dist.barrier(group=ctrl)
        from stencil_amd import _C

        if args.backend == "native":
            _C.device_synchronize_all()

    for _ in range(args.warmup):
        app.step(overlap=not args.no_overlap)
    barrier_sync()

    t0 = time.perf_counter()
    if (getattr(app, "_graph", None) is not None
            or getattr(app, "_mr_graph", None) is not None) and not args.no_overlap:
        # graph mode: queue all K replays, one sync (the work is identical;
        # only the per-step host launch+wake latency is amortized)
        app.run(args.steps)
    else:
        for _ in range(args.steps):
            app.step(overlap=not args.no_overlap)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist is not None:
        import torch

        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX, group=ctrl)
        elapsed = float(t.item())

    total_cells = size[0] * size[1] * size[2]
    value = total_cells * args.steps / elapsed
    xbytes = app.dd.exchange_bytes_for_method(app.dd.methods)
    if dist is not None:
        import torch

        t = torch.tensor([xbytes], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=ctrl)
        xbytes = float(t.item())

    if rank == 0:
        out = {
            "metric": "jacobi3d_cell_updates_per_s",
            "value": value,
            "unit": "cells/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "jacobi3d",
                "grid_per_gpu": f"{args.per_gpu}^3",
                "global_grid": "x".join(str(s) for s in size),
                "radius": 1,
                "halo_multiplier": args.halo_multiplier,
                "overlap": not args.no_overlap,
                "step_graph": (getattr(app, "_graph", None) is not None
                               or getattr(app, "_mr_graph", None) is not None),
                "exchange_bytes_per_iter": xbytes,
                "parallelism": f"domain-decomposition {dims[0]}x{dims[1]}x{dims[2]} "
                + (
                    (
                        "multi-process-ipc-xgmi"
                        if getattr(app.dd.backend, "_ipc_active", False)
                        else "multi-process-rccl"
                    )
                    if world > 1
                    else "single-process-xgmi"
                ),
            },
        }
        print(json.dumps(out), flush=True)

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
