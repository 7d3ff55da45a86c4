#!/usr/bin/env python3
"""Astaroth-class MHD benchmark: 256^3 points/GPU, fp64, radius 3, 8
fields, 3 RK3 substeps per iteration (reference: astaroth/astaroth.cu,
scripts/summit/512node_astaroth.sh). `--no-compute` measures the pure
exchange (3 exchanges/iter)."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from stencil_amd.models.astaroth import Astaroth, parse_conf
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
    ap.add_argument("--per-gpu", type=int, default=256)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--no-compute", action="store_true")
    ap.add_argument("--pipelined", action="store_true",
                    help="time iters as one pipelined run() block (graph replays "
                    "queue back-to-back; reports mean instead of trimean)")
    ap.add_argument("--no-overlap", action="store_true")
    ap.add_argument("--conf", default=None)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        import torch
        import torch.distributed as dist

        # modulo device_count so N-rank smokes on a 1-GPU box still run
        torch.cuda.set_device(
            int(os.environ.get("LOCAL_RANK", rank)) % max(1, torch.cuda.device_count())
        )
        dist.init_process_group(backend="cpu:gloo,cuda:nccl")
        n, gpus = world, None
    else:
        n, gpus = args.gpus, list(range(args.gpus))

    dims = weak_dims(n)
    size = tuple(args.per_gpu * d for d in dims)
    conf = parse_conf(args.conf) if args.conf else None
    app = Astaroth(size, conf=conf, gpus=gpus)
    app.realize()
    app.init_fields()

    stats = Statistics()
    if args.pipelined and world == 1 and not args.no_compute:
        import time as _t

        for _ in range(args.warmup):
            app.step(compute=True, overlap=False if args.no_overlap else None)
        t0 = _t.perf_counter()
        app.run(args.iters)
        mean = (_t.perf_counter() - t0) / args.iters
        cells = size[0] * size[1] * size[2]
        print(
            f"astaroth,pipelined,gpus=1,grid={size[0]}x{size[1]}x{size[2]},"
            f"mean_s={mean:.6f},Mcells_per_s={cells / mean / 1e6:.1f}",
            flush=True,
        )
        return
    for i in range(args.iters + args.warmup):
        if world > 1:
            import torch.distributed as dist

            dist.barrier()
        t0 = time.perf_counter()
        app.step(compute=not args.no_compute,
                 overlap=False if args.no_overlap else None)
        dt = time.perf_counter() - t0
        if i >= args.warmup:
            stats.insert(dt)

    tm = stats.trimean()
    if world > 1:
        import torch
        import torch.distributed as dist

        t = torch.tensor([tm], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        tm = float(t.item())
        dist.destroy_process_group()
    if rank == 0:
        cells = size[0] * size[1] * size[2]
        mode = "exchange-only" if args.no_compute else "full"
        print(
            f"astaroth,{mode},gpus={n},grid={size[0]}x{size[1]}x{size[2]},"
            f"trimean_s={tm:.6f},min_s={stats.min():.6f},Mcells_per_s={cells / tm / 1e6:.1f}",
            flush=True,
        )


if __name__ == "__main__":
    main()
