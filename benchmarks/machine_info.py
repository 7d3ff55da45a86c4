#!/usr/bin/env python3
"""Dump the cluster/machine model (reference: bin/machine_info.cu,
include/stencil/machine.hpp): per-rank hostname + visible GPUs with name,
PCI address, memory, CU count, and the pairwise link distance matrix.
Run under torch.distributed for multi-rank output."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from stencil_amd import _C
from stencil_amd.parallel.comm import Comm
from stencil_amd.parallel.machine import Machine


def main():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        import torch.distributed as dist

        dist.init_process_group("gloo")
    comm = Comm()

    n = _C.device_count()
    gpus = []
    for d in range(n):
        gi = _C.gpu_info(d)
        gpus.append(
            dict(dev=d, name=gi.name, pci=gi.pci, mem_gb=gi.total_mem / 2**30, cus=gi.cu_count)
        )
    infos = comm.allgather_object({"host": comm.hostname, "gpus": gpus})
    machine = Machine.build(comm)
    if comm.rank == 0:
        for r, info in enumerate(infos):
            print(f"rank {r} host {info['host']}")
            for g in info["gpus"]:
                print(
                    f"  gpu {g['dev']}: {g['name']} pci={g['pci']} mem={g['mem_gb']:.0f}GB cus={g['cus']}"
                )
        print(f"machine: {machine.num_nodes()} node(s), {len(machine.gpus)} physical GPU(s)")
        for g in machine.gpus:
            print(f"  global gpu {g.index}: node {g.node} {g.name} pci={g.pci} ranks={g.ranks}")
        if n > 1:
            print("link distance matrix (gpu_distance):")
            for a in range(n):
                row = " ".join(f"{_C.gpu_distance(a, b):4.1f}" for b in range(n))
                print(f"  {a}: {row}")
    if world > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
