"""Stencil neighbor topology (periodic boundary).

Reference: include/stencil/topology.hpp, src/topology.cpp — neighbor lookup
wraps the subdomain index into the partition grid.
"""
from __future__ import annotations

from typing import Tuple

Vec = Tuple[int, int, int]


class Topology:
    """Periodic 3D torus over the partition grid of extent `dim`."""

    def __init__(self, dim: Vec):
        self.dim = tuple(int(c) for c in dim)

    def get_neighbor(self, idx: Vec, direction: Vec) -> Vec:
        return tuple((idx[i] + direction[i]) % self.dim[i] for i in range(3))


DIRECTIONS = [
    (x, y, z)
    for z in (-1, 0, 1)
    for y in (-1, 0, 1)
    for x in (-1, 0, 1)
    if (x, y, z) != (0, 0, 0)
]


def dir_key(d: Vec):
    """deterministic sort key for directions (z, y, x lexicographic)"""
    return (d[2], d[1], d[0])
