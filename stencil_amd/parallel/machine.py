"""Machine model: the cluster as every rank sees it.

MI355X-native equivalent of the reference's Machine (reference:
include/stencil/machine.hpp, src/machine.cpp:19-129 — hostname allgather
to node ids + CUDA device-UUID allgather to a global GPU list with owning
ranks). Here the per-GPU identity is the PCI bus id (unique per node) from
the native `gpu_info`, qualified by hostname."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Tuple

from .. import _C
from .comm import Comm


@dataclass
class MachineGpu:
    """one physical GPU of the job, with every rank that can drive it"""

    index: int  # global index in the machine
    node: int
    host: str
    pci: str
    name: str
    cu_count: int
    total_mem: int
    ranks: List[int] = field(default_factory=list)  # owning ranks
    cuda_of_rank: Dict[int, int] = field(default_factory=dict)  # rank -> local id


class Machine:
    """Global GPU inventory (reference Machine::build)."""

    def __init__(self, comm: Comm, gpus: List[MachineGpu], node_of_rank: List[int]):
        self.comm = comm
        self.gpus = gpus
        self._node_of_rank = node_of_rank
        self._by_key = {(g.host, g.pci): g for g in gpus}

    @classmethod
    def build(cls, comm: Comm, my_gpus: List[int] = None) -> "Machine":
        """collective: every rank reports its visible (or given) GPUs"""
        n = _C.device_count()
        if my_gpus is None:
            my_gpus = list(range(n))
        infos = []
        for cuda in my_gpus:
            if cuda < n:
                gi = _C.gpu_info(cuda)
                infos.append((cuda, gi.pci, gi.name, gi.cu_count, gi.total_mem))
            else:
                infos.append((cuda, f"cpu-stub-{cuda}", "no-gpu", 0, 0))
        gathered = comm.allgather_object({"host": comm.hostname, "gpus": infos})
        nodes = comm.node_of_rank()
        by_key: Dict[Tuple[str, str], MachineGpu] = {}
        order = []
        for rank, rec in enumerate(gathered):
            for cuda, pci, name, cu, mem in rec["gpus"]:
                key = (rec["host"], pci)
                if key not in by_key:
                    g = MachineGpu(len(order), nodes[rank], rec["host"], pci, name, cu, mem)
                    by_key[key] = g
                    order.append(g)
                by_key[key].ranks.append(rank)
                by_key[key].cuda_of_rank[rank] = cuda
        return cls(comm, order, nodes)

    def num_nodes(self) -> int:
        return len(set(self._node_of_rank))

    def node_of_rank(self, rank: int) -> int:
        return self._node_of_rank[rank]

    def gpus_of_rank(self, rank: int) -> List[MachineGpu]:
        return [g for g in self.gpus if rank in g.ranks]

    def classify(self, src_rank: int, dst_rank: int) -> str:
        """transfer class between two ranks (reference bench_mpi's
        self/colocated/remote split)"""
        if src_rank == dst_rank:
            return "self"
        if self._node_of_rank[src_rank] == self._node_of_rank[dst_rank]:
            return "colocated"
        return "remote"
