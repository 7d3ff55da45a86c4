#!/usr/bin/env python3
"""Pure exchange() timing, weak or strong scaling (reference:
bin/exchange_weak.cu, bin/exchange_strong.cu: 512^3/GPU weak or fixed
total, trimean seconds per method set)."""
import argparse
import os
import sys
import time

# NOTE: the engine's 512-block copy cap is kept even for exchange-only
# runs — measured FASTER than the natural full grid at 1024^3 (395 vs
# 306 GB/s faces; profiles/r2/r2_gpu11_exch.log): with one block per 512
# copy words the per-block tail and the scattered per-job addressing
# dominate, while 512 grid-striding blocks keep every HBM channel busy.

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

import stencil_amd as sa
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
    ap.add_argument("--size", type=int, default=512)
    ap.add_argument("--strong", action="store_true")
    ap.add_argument("--radius", type=int, default=1)
    ap.add_argument("--iters", type=int, default=30)
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

    size = (args.size,) * 3 if args.strong else tuple(args.size * d for d in weak_dims(n))
    dd = sa.DistributedDomain(*size, backend=args.backend)
    dd.set_radius(args.radius)
    if gpus is not None:
        dd.set_gpus(gpus)
    dd.add_data(np.float32, "q")
    dd.realize()

    stats = Statistics()
    for i in range(args.iters + 3):
        if world > 1:
            import torch.distributed as dist

            dist.barrier()
        t0 = time.perf_counter()
        dd.exchange()
        dt = time.perf_counter() - t0
        if i >= 3:
            stats.insert(dt)
        dd.swap()

    tm = stats.trimean()
    xbytes = dd.exchange_bytes_for_method(sa.Method.DEFAULT)
    if world > 1:
        import torch
        import torch.distributed as dist

        t = torch.tensor([tm, float(xbytes)], dtype=torch.float64)
        dist.all_reduce(t[:1], op=dist.ReduceOp.MAX)
        dist.all_reduce(t[1:], op=dist.ReduceOp.SUM)
        tm, xbytes = float(t[0]), float(t[1])
        dist.destroy_process_group()
    if rank == 0:
        mode = "strong" if args.strong else "weak"
        print(
            f"exchange,{mode},gpus={n},r={args.radius},{size[0]}x{size[1]}x{size[2]},"
            f"bytes={int(xbytes)},trimean_s={tm:.6f},GBs={xbytes / tm / 1e9:.2f}",
            flush=True,
        )


if __name__ == "__main__":
    main()
