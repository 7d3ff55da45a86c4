"""Astaroth-class MHD mini-app: 8 fp64 fields, radius-3 (6th order), three
RK3 substeps per iteration, each overlapping interior compute with the halo
exchange (reference: astaroth/astaroth.cu:556-641 loop structure,
astaroth_utils.cu conf parser). The solver kernels are an independent
implementation of standard compressible resistive MHD (csrc/src/mhd.hip).
"""
from __future__ import annotations

import math

from typing import Dict, List, Optional

import numpy as np

from .. import _C
from ..core import DistributedDomain
from ..parallel.placement import PlacementStrategy

FIELDS = ["lnrho", "uux", "uuy", "uuz", "aax", "aay", "aaz", "ss"]

DEFAULT_CONF = {
    "nn": 256,  # per-GPU edge length (weak scaling)
    "dsx": 0.04908738521,  # 2*pi/128, astaroth.conf-style spacing
    "dsy": 0.04908738521,
    "dsz": 0.04908738521,
    "cs2": 1.0,
    "cp_inv": 1.0,
    "nu": 5e-3,
    "eta": 5e-3,
    "chi": 5e-4,
    "dt": 1e-4,
}


def parse_conf(path: str) -> Dict[str, float]:
    """`key = value` text config (reference: astaroth_utils.cu:23-48)"""
    conf = dict(DEFAULT_CONF)
    with open(path) as f:
        for line in f:
            line = line.split("#")[0].split("//")[0].strip()
            if not line or "=" not in line:
                continue
            k, v = (s.strip() for s in line.split("=", 1))
            conf[k] = float(v) if "." in v or "e" in v.lower() else int(v)
    return conf


def init_modes(size) -> List[tuple]:
    """deterministic smooth initial condition per field:
    (base, amp, mode vector (integers), phase)"""
    modes = []
    for i, name in enumerate(FIELDS):
        base = 0.0
        amp = 0.01 if name != "lnrho" else 0.02
        m = ((i % 3) + 1, ((i + 1) % 3) + 1, ((i + 2) % 3) + 1)
        phase = 0.3 * i
        modes.append((base, amp, m, phase))
    return modes


def harmonic_np(lo, hi, size, base, amp, m, phase):
    """NumPy mirror of init_harmonic_f64 over the box [lo, hi)"""
    zz, yy, xx = np.meshgrid(
        np.arange(lo[2], hi[2], dtype=np.float64),
        np.arange(lo[1], hi[1], dtype=np.float64),
        np.arange(lo[0], hi[0], dtype=np.float64),
        indexing="ij",
    )
    kx = 2 * math.pi * m[0] / size[0]
    ky = 2 * math.pi * m[1] / size[1]
    kz = 2 * math.pi * m[2] / size[2]
    return base + amp * np.sin(kx * xx + ky * yy + kz * zz + phase)


class Astaroth:
    def __init__(
        self,
        size,
        conf: Optional[Dict] = None,
        backend: str = "native",
        gpus: Optional[List[int]] = None,
        placement: PlacementStrategy = PlacementStrategy.NodeAware,
    ):
        self.conf = dict(DEFAULT_CONF)
        if conf:
            self.conf.update(conf)
        self.size = tuple(size)
        self.dd = DistributedDomain(*size, backend=backend)
        self.dd.set_radius(3)
        # exchange group 0 = the 8 physics fields (before the div pass),
        # group 1 = div u / div A (after it) -- each exchange moves only
        # what changed
        self.dd.set_exchange_groups([list(range(8)), [8, 9]])
        self.dd.set_placement(placement)
        if gpus is not None:
            self.dd.set_gpus(gpus)
        self.handles = [self.dd.add_data(np.float64, n) for n in FIELDS]
        # auxiliary exchanged quantities of the separable-derivative
        # scheme (csrc/src/mhd.hip): div u and div A, recomputed and
        # re-exchanged every substep
        self.dd.add_data(np.float64, "divu")
        self.dd.add_data(np.float64, "diva")
        self.cf = _C.MhdCoeffs()
        for k in ("dsx", "dsy", "dsz", "cs2", "cp_inv", "nu", "eta", "chi"):
            setattr(self.cf, k, float(self.conf[k]))

    def realize(self):
        self.dd.realize()
        self.interiors = self.dd.get_interior()
        self.exteriors = self.dd.get_exterior()
        # whole-substep hipGraph fast path (world=1 single-domain): replays
        # captured [X1 -> div -> X2 -> scalar||momentum -> table swap]
        # graphs per substep, removing the ~0.45 ms/substep host gap
        # measured at 256^3 (profiles/astaroth_256_kernel_stats.csv).
        # Only valid for the conf dt it was captured with.
        import os

        self._graph = None
        self._mr_graph = None
        self._graph_dt = None
        graphs_on = os.environ.get("STENCIL_AMD_STEP_GRAPH", "1") != "0"
        staged_local = any(getattr(self.dd.backend, "_staged_local", []))
        if (
            self.dd.backend_kind == "native"
            and self.dd.comm.world_size == 1
            and self.dd.num_local() == 1
            and graphs_on
            and not staged_local
        ):
            lo, hi = self.dd.local_rect(0)
            rect = _C.Rect3(_C.Vec3(*lo), _C.Vec3(*hi))
            self._graph_dt = float(self.conf["dt"])
            self._graph = _C.mhd_graph_create(
                self.dd.backend.engine, 0, rect, self._graph_dt, self.cf
            )
        elif (
            self.dd.backend_kind == "native"
            and self.dd.comm.world_size > 1
            and self.dd.num_local() == 1
            and graphs_on
            and getattr(self.dd.backend, "_ipc_active", False)
            and not any(getattr(self.dd.backend, "_has_wire", [True]))
            and not staged_local
        ):
            # multi-rank substep graphs (single-node 1-rank/GPU shape):
            # per substep, three graphs around the two colo barriers
            # (csrc/src/mhd.hip mhd_mr_graph_create)
            ilo, ihi = self.interiors[0]
            self._graph_dt = float(self.conf["dt"])
            self._mr_graph = _C.mhd_mr_graph_create(
                self.dd.backend.engine, 0,
                _C.Rect3(_C.Vec3(*ilo), _C.Vec3(*ihi)),
                [_C.Rect3(_C.Vec3(*blo), _C.Vec3(*bhi)) for blo, bhi in self.exteriors[0]],
                self._graph_dt, self.cf,
            )

    def _mr_barrier(self, sync_first: bool):
        b = self.dd.backend
        if b._colo_wire is not None:
            b._colo_wire.barrier(_C.mhd_mr_graph_stream(self._mr_graph))
        else:
            import torch.distributed as dist

            _C.mhd_mr_graph_sync(self._mr_graph)
            dist.barrier(group=b._colo_group)

    def _mr_iter(self, sync: bool = True):
        """one queued RK3 iteration through the multi-rank graphs"""
        b = self.dd.backend
        for _s in range(3):
            _C.mhd_mr_phase1(self._mr_graph)
            self._mr_barrier(True)
            _C.mhd_mr_phase2(self._mr_graph)
            self._mr_barrier(True)
            _C.mhd_mr_phase3(self._mr_graph)
            b._colo_parity[0] ^= 1  # keep the eager path's mirrors in sync
            b._colo_parity[1] ^= 1
        if sync or b._colo_wire is None:
            _C.mhd_mr_graph_sync(self._mr_graph)


    def init_fields(self, kind: str = "harmonic"):
        """device-side initial conditions on every interior.
        kind='harmonic' (default): per-field sinusoidal modes;
        kind='explosion': radial gaussian velocity bump at the domain
        center (reference astaroth.cu radial_explosion_init_kernel),
        other fields zero."""
        eng = self.dd.backend.engine
        if kind == "explosion":
            cx, cy, cz = (s / 2.0 for s in self.size)
            sigma = min(self.size) / 8.0
            for li in range(self.dd.num_local()):
                lo, hi = self.dd.local_rect(li)
                rect = _C.Rect3(_C.Vec3(*lo), _C.Vec3(*hi))
                for qi, name in enumerate(FIELDS):
                    amp = 0.1 if name.startswith("uu") else 0.0
                    _C.init_radial_f64(eng, li, qi, rect, 0.0, amp, cx, cy, cz, sigma, False)
            self.dd.backend.sync_compute()
            return
        for li in range(self.dd.num_local()):
            lo, hi = self.dd.local_rect(li)
            for qi, (base, amp, m, phase) in enumerate(init_modes(self.size)):
                kx = 2 * math.pi * m[0] / self.size[0]
                ky = 2 * math.pi * m[1] / self.size[1]
                kz = 2 * math.pi * m[2] / self.size[2]
                _C.init_harmonic_f64(
                    eng,
                    li,
                    qi,
                    _C.Rect3(_C.Vec3(*lo), _C.Vec3(*hi)),
                    base,
                    amp,
                    kx,
                    ky,
                    kz,
                    phase,
                    False,
                )
        self.dd.backend.sync_compute()

    def _substep(self, s: int, dt: float, compute: bool, overlap: bool):
        """separable-derivative substep with two exchanges, both hidden
        under compute when `overlap`:

          div(interior)            || exchange #1 (field halos)
          div(exterior shells)
          main(interior)           || exchange #2 (div halos)
          main(exterior shells)

        Safe because interior cells read only +-3 neighborhoods inside the
        compute region (never halos), and the cells an exchange sends (the
        outermost radius-deep shells) lie in the exterior region, which is
        only written after the exchange that could read it completes."""
        dd = self.dd
        eng = dd.backend.engine

        def rect(lo, hi):
            return _C.Rect3(_C.Vec3(*lo), _C.Vec3(*hi))

        if not compute:
            dd.exchange(group=0)
            dd.exchange(group=1)
            dd.backend.sync_compute()
            dd.swap()
            return
        if overlap:
            for li in range(dd.num_local()):
                _C.mhd_div_pass(eng, li, rect(*self.interiors[li]), self.cf)
            dd.exchange(group=0)  # field halos; overlaps div(interior)
            for li in range(dd.num_local()):
                for box in self.exteriors[li]:
                    _C.mhd_div_pass(eng, li, rect(*box), self.cf)
            dd.backend.sync_compute()  # div complete before X2 reads its edges
            for li in range(dd.num_local()):
                _C.mhd_substep(eng, li, rect(*self.interiors[li]), s, dt, self.cf)
            dd.exchange(group=1)  # div halos only; overlaps main(interior)
            for li in range(dd.num_local()):
                for box in self.exteriors[li]:
                    _C.mhd_substep(eng, li, rect(*box), s, dt, self.cf)
        else:
            dd.exchange(group=0)
            for li in range(dd.num_local()):
                _C.mhd_div_pass(eng, li, rect(*dd.local_rect(li)), self.cf)
            dd.backend.sync_compute()
            dd.exchange(group=1)
            for li in range(dd.num_local()):
                _C.mhd_substep(eng, li, rect(*dd.local_rect(li)), s, dt, self.cf)
        dd.backend.sync_compute()
        dd.swap()

    def step(self, dt: Optional[float] = None, compute: bool = True,
             overlap: Optional[bool] = None):
        """overlap=None auto-selects: the interior/exterior split pays only
        when the exchange is worth hiding (multi-rank); at world=1 the
        extra thin-slab launches cost more than the 0.2 ms exchange
        (measured 12.7 vs 10.8 ms/iter at 256^3)"""
        if overlap is None:
            overlap = self.dd.comm.world_size > 1
        dt = self.conf["dt"] if dt is None else dt
        if (
            self._graph is not None
            and compute
            and not overlap
            and dt == self._graph_dt
        ):
            _C.mhd_graph_iter(self._graph, 1)
            _C.mhd_graph_sync(self._graph)
            return
        if self._mr_graph is not None:
            if compute and dt == self._graph_dt:
                self._mr_iter()
                return
            # parity cannot be resynced after an eager substep; degrade
            self._mr_graph = None
        for s in range(3):
            self._substep(s, dt, compute, overlap)

    def run(self, n: int):
        """n full RK3 iterations with one host sync at the end (graph
        mode queues 3n substep replays back-to-back)"""
        if self._graph is not None and self.dd.comm.world_size == 1:
            _C.mhd_graph_iter(self._graph, n)
            _C.mhd_graph_sync(self._graph)
            return
        if self._mr_graph is not None:
            for _ in range(n):
                self._mr_iter(sync=False)
            _C.mhd_mr_graph_sync(self._mr_graph)
            return
        for _ in range(n):
            self.step()

    def read_field(self, li: int, name: str) -> np.ndarray:
        lo, hi = self.dd.local_rect(li)
        return self.dd.read_global(li, lo, hi, self.handles[FIELDS.index(name)])
