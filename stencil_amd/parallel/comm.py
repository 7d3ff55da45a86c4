"""Process-group wrapper for control-plane collectives.

MI355X-native replacement for the reference's MPI world / MpiTopology
(reference: include/stencil/mpi.hpp, mpi_topology.hpp): torch.distributed
(gloo for host objects, RCCL for device tensors) instead of MPI. Falls back
to single-rank stubs when torch.distributed is not initialized.
"""
from __future__ import annotations

import socket
from typing import Any, List


def _dist():
    import torch.distributed as dist

    return dist if (dist.is_available() and dist.is_initialized()) else None


class Comm:
    """Rank/world info + object collectives + colocated-rank discovery."""

    def __init__(self):
        d = _dist()
        if d is None:
            self.rank = 0
            self.world_size = 1
        else:
            self.rank = d.get_rank()
            self.world_size = d.get_world_size()
        self.hostname = socket.gethostname()
        self._hostnames = None

    def allgather_object(self, obj: Any) -> List[Any]:
        d = _dist()
        if d is None:
            return [obj]
        out = [None] * self.world_size
        d.all_gather_object(out, obj)
        return out

    def broadcast_object(self, obj: Any, src: int = 0) -> Any:
        d = _dist()
        if d is None:
            return obj
        box = [obj if self.rank == src else None]
        d.broadcast_object_list(box, src=src)
        return box[0]

    def barrier(self):
        d = _dist()
        if d is not None:
            d.barrier()

    def hostnames(self) -> List[str]:
        """hostname of every rank (cached; a collective on first call)"""
        if self._hostnames is None:
            self._hostnames = self.allgather_object(self.hostname)
        return self._hostnames

    def colocated_ranks(self) -> List[int]:
        """ranks sharing this rank's host (including self), ascending"""
        names = self.hostnames()
        return [r for r, h in enumerate(names) if h == self.hostname]

    def node_of_rank(self) -> List[int]:
        """node index (by first appearance) for every rank"""
        names = self.hostnames()
        order: List[str] = []
        for h in names:
            if h not in order:
                order.append(h)
        return [order.index(h) for h in names]
