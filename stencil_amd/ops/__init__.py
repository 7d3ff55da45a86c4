"""Native compute ops (hand-written gfx950 HIP kernels, csrc/src/*.hip),
re-exported at package level. Each op launches on a domain's compute
stream of the given ExchangeEngine; regions are global-coordinate Rect3s.

- jacobi_step: 7-point fp32 Jacobi with hot/cold sphere sources
  (csrc/src/jacobi.hip, z-marching float4 kernel)
- mhd_div_pass / mhd_substep: separable-derivative 6th-order MHD
  (csrc/src/mhd.hip)
- fill_f32 / init_harmonic_f64: field initialization (csrc/src/init.hip)
- field_stats: min/max/RMS reduction (csrc/src/reductions.hip)
"""
from .._C import (  # noqa: F401
    FieldStats,
    MhdCoeffs,
    field_stats,
    fill_f32,
    init_harmonic_f64,
    jacobi_step,
    mhd_div_pass,
    mhd_substep,
)

__all__ = [
    "FieldStats",
    "MhdCoeffs",
    "field_stats",
    "fill_f32",
    "init_harmonic_f64",
    "jacobi_step",
    "mhd_div_pass",
    "mhd_substep",
]
