#!/usr/bin/env python3
"""Halo-exchange latency/goodput benchmark over radius patterns
(reference: bin/bench_exchange.cu:126-195 — +x only, x only, faces,
face&edge, uniform r=2 on a 128^3-per-GPU domain; trimean seconds and B/s).

Single process multi-GPU: python benchmarks/bench_exchange.py --gpus 8
Multi-process (RCCL):     torchrun --nproc-per-node 8 benchmarks/bench_exchange.py
"""
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
from stencil_amd import _C
from stencil_amd.utils.statistics import Statistics


def radius_patterns():
    pats = {}
    r = _C.Radius.constant(0)
    r.set_dir(1, 0, 0, 2)
    pats["pos_x_only"] = r
    r = _C.Radius.constant(0)
    r.set_dir(1, 0, 0, 2)
    r.set_dir(-1, 0, 0, 2)
    pats["x_only"] = r
    pats["faces_only"] = _C.Radius.face_edge_corner(2, 0, 0)
    pats["faces_edges"] = _C.Radius.face_edge_corner(2, 2, 0)
    pats["uniform_2"] = _C.Radius.constant(2)
    return pats


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--size", type=int, default=128, help="edge length per GPU-ish")
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
        n = world
        gpus = None
    else:
        n = args.gpus
        gpus = list(range(n)) if args.backend == "native" else [0] * n

    if rank == 0:
        print("pattern,gpus,x,y,z,bytes,trimean_s,min_s,goodput_GBs", flush=True)
    for name, radius in radius_patterns().items():
        dd = sa.DistributedDomain(args.size, args.size, args.size, backend=args.backend)
        dd.set_radius(radius)
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
        xbytes = dd.exchange_bytes_for_method(sa.Method.DEFAULT)
        if world > 1:
            import torch
            import torch.distributed as dist

            t = torch.tensor([stats.trimean(), stats.min(), float(xbytes)], dtype=torch.float64)
            dist.all_reduce(t[:2], op=dist.ReduceOp.MAX)
            dist.all_reduce(t[2:], op=dist.ReduceOp.SUM)
            tm, mn, xbytes = float(t[0]), float(t[1]), float(t[2])
        else:
            tm, mn = stats.trimean(), stats.min()
        if rank == 0:
            gb = xbytes / tm / 1e9 if tm > 0 else 0.0
            print(
                f"{name},{n},{args.size},{args.size},{args.size},{int(xbytes)},{tm:.6f},{mn:.6f},{gb:.2f}",
                flush=True,
            )
        del dd

    if world > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
