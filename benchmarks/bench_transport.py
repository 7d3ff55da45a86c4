#!/usr/bin/env python3
"""Raw transport benchmark on the library-shaped neighbor pattern
(reference: bin/bench_mpi.cu / bin/bench_alltoallv.cu): times bare RCCL
batched point-to-point transfers of the same buffers the halo exchange
would move, without pack/unpack -- isolates wire cost from kernel cost.
Run under torchrun with >= 2 ranks."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

import stencil_amd as sa
from stencil_amd.parallel.machine import Machine
from stencil_amd.parallel.planning import plan_exchange, wire_layout
from stencil_amd.utils.statistics import Statistics


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=512)
    ap.add_argument("--radius", type=int, default=1)
    ap.add_argument("--iters", type=int, default=30)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world < 2:
        print("bench_transport needs >= 2 ranks (torchrun)", file=sys.stderr)
        sys.exit(1)
    import torch
    import torch.distributed as dist

    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
    dist.init_process_group(backend="cpu:gloo,cuda:nccl")

    # plan only (no allocation of domains)
    dd = sa.DistributedDomain(args.size, args.size, args.size)
    dd.set_radius(args.radius)
    dd.do_placement()
    plan = plan_exchange(dd.placement, dd.radius, rank)

    es = [4]
    dev = f"cuda:{local_rank}"
    sends = [
        (torch.empty(wire_layout(s.messages, es)[0], dtype=torch.uint8, device=dev), s.peer_rank)
        for s in plan.sends
    ]
    recvs = [
        (torch.empty(wire_layout(r.messages, es)[0], dtype=torch.uint8, device=dev), r.peer_rank)
        for r in plan.recvs
    ]
    total = sum(t.numel() for t, _ in sends)

    # self/colocated/remote byte classification (reference bin/bench_mpi.cu
    # via its Machine model)
    machine = Machine.build(dd.comm, dd.gpus or [local_rank])
    by_class = {"self": 0, "colocated": 0, "remote": 0}
    for s_item in plan.sends:
        by_class[machine.classify(rank, s_item.peer_rank)] += wire_layout(
            s_item.messages, es
        )[0]
    for t_item in plan.translates:
        by_class["self"] += t_item.ext[0] * t_item.ext[1] * t_item.ext[2] * es[0]

    stats = Statistics()
    for i in range(args.iters + 3):
        dist.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        ops = [dist.P2POp(dist.isend, t, p) for t, p in sends]
        ops += [dist.P2POp(dist.irecv, t, p) for t, p in recvs]
        for w in dist.batch_isend_irecv(ops):
            w.wait()
        torch.cuda.synchronize()
        if i >= 3:
            stats.insert(time.perf_counter() - t0)

    tm = stats.trimean()
    agg = torch.tensor([tm, float(total)], dtype=torch.float64)
    dist.all_reduce(agg[:1], op=dist.ReduceOp.MAX)
    dist.all_reduce(agg[1:], op=dist.ReduceOp.SUM)
    if rank == 0:
        tm, tot = float(agg[0]), float(agg[1])
        print(
            f"transport,rccl_p2p,world={world},r={args.radius},size={args.size},"
            f"bytes={int(tot)},trimean_s={tm:.6f},GBs={tot / tm / 1e9:.2f}",
            flush=True,
        )
    cls_t = torch.tensor([by_class["self"], by_class["colocated"], by_class["remote"]],
                         dtype=torch.float64)
    dist.all_reduce(cls_t, op=dist.ReduceOp.SUM)
    if rank == 0:
        print(
            f"transport,classes,self_B={int(cls_t[0])},colocated_B={int(cls_t[1])},"
            f"remote_B={int(cls_t[2])},nodes={machine.num_nodes()},gpus={len(machine.gpus)}",
            flush=True,
        )
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
