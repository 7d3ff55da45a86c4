"""stencil_amd: an MI355X-native 3D stencil halo-exchange framework.

Built from scratch for AMD Instinct MI355X (gfx950 / CDNA4) with the same
capabilities as cwpearson/stencil: distributed 3D grids with named
quantities, arbitrary per-direction stencil radius, topology-aware
placement, batched halo exchange over xGMI (direct-write HIP kernels) and
RCCL (cross-process), and interior/exterior overlap of communication with
computation.
"""

try:
    from . import _C
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "stencil_amd native extension not built; run `python tools/build_native.py`"
    ) from e

from ._C import Radius, Rect3, Vec3, prime_factors
from .core import DataHandle, DistributedDomain, Method
from .parallel.placement import PlacementStrategy

__all__ = [
    "DistributedDomain",
    "DataHandle",
    "Method",
    "PlacementStrategy",
    "Radius",
    "Rect3",
    "Vec3",
    "prime_factors",
    "_C",
]

__version__ = "0.1.0"
