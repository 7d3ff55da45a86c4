"""stencil_amd: an MI355X-native 3D stencil halo-exchange framework.

Built from scratch for AMD Instinct MI355X (gfx950 / CDNA4) with the same
capabilities as cwpearson/stencil: distributed 3D grids with named
quantities, arbitrary per-direction stencil radius, topology-aware
placement, batched halo exchange over xGMI (direct-write HIP kernels) and
RCCL (cross-process), and interior/exterior overlap of communication with
computation.
"""

def _preload_torch_hip_runtime():
    """Bind to PyTorch's bundled HIP runtime BEFORE _C loads.

    PyTorch wheels ship their own libamdhip64/libhsa-runtime64 and load
    them by absolute path (rpath), so they load even when a copy with the
    same soname is already in the process. If _C loads /opt/rocm's runtime
    first and torch is imported later, the process ends up with TWO HSA
    runtimes; the kernel driver registers a process once, and the loser's
    hipGetDeviceCount sees 0 GPUs (observed under torchrun: torch saw the
    GPU, _C did not). dlopening torch's copy first registers its soname,
    so _C's DT_NEEDED resolves to the same runtime torch will use —
    one runtime regardless of import order. No-op when torch is absent
    (pure-native users get /opt/rocm's runtime as before)."""
    import ctypes
    import importlib.util
    import os

    if os.environ.get("STENCIL_AMD_NO_PRELOAD", "") == "1":
        return
    spec = importlib.util.find_spec("torch")
    if spec is None or not spec.origin:
        return
    lib = os.path.join(os.path.dirname(spec.origin), "lib", "libamdhip64.so")
    if os.path.exists(lib):
        try:
            ctypes.CDLL(lib, mode=ctypes.RTLD_GLOBAL)
        except OSError:  # pragma: no cover - fall back to /opt/rocm
            pass


_preload_torch_hip_runtime()

try:
    from . import _C
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "stencil_amd native extension not built; run `python tools/build_native.py`"
    ) from e

from ._C import Radius, Rect3, Vec3, prime_factors
from .core import DataHandle, DistributedDomain, Method
from .parallel.machine import Machine
from .parallel.placement import PlacementStrategy

__all__ = [
    "Machine",
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
