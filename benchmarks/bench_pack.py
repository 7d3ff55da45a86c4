#!/usr/bin/env python3
"""Pack/unpack kernel throughput per halo direction on a radius-3 domain
(reference: bin/bench_pack.cu). One GPU."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

from stencil_amd import _C
from stencil_amd.utils.statistics import Statistics


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=512)
    ap.add_argument("--radius", type=int, default=3)
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()

    sz = _C.Vec3(args.size, args.size, args.size)
    dom = _C.LocalDomain(sz, _C.Vec3(0, 0, 0), 0)
    dom.set_radius(_C.Radius.constant(args.radius))
    dom.add_data(4, "q")
    dom.realize()

    print("dir,bytes,pack_GBs,unpack_GBs", flush=True)
    for dz in (-1, 0, 1):
        for dy in (-1, 0, 1):
            for dx in (-1, 0, 1):
                if (dx, dy, dz) == (0, 0, 0):
                    continue
                d = _C.Vec3(dx, dy, dz)
                nd = _C.Vec3(-dx, -dy, -dz)
                ext = dom.halo_extent(nd)
                nbytes = 4 * ext.flatten()
                eng = _C.ExchangeEngine([dom])
                buf = eng.create_buffer(0, nbytes)
                eng.add_pack(0, buf, 0, dom.halo_pos(d, False), ext, 0)
                eng.add_unpack(0, buf, 0, dom.halo_pos(nd, True), ext, 0)
                eng.finalize()
                # warmup
                eng.launch_packs()
                eng.launch_unpacks()
                eng.sync_packs()
                sp, su = Statistics(), Statistics()
                for _ in range(args.iters):
                    t0 = time.perf_counter()
                    eng.launch_packs()
                    eng.sync_packs()
                    t1 = time.perf_counter()
                    eng.launch_unpacks()
                    eng.sync_packs()
                    t2 = time.perf_counter()
                    sp.insert(t1 - t0)
                    su.insert(t2 - t1)
                print(
                    f"({dx} {dy} {dz}),{nbytes},{nbytes / sp.trimean() / 1e9:.2f},{nbytes / su.trimean() / 1e9:.2f}",
                    flush=True,
                )
                del eng


if __name__ == "__main__":
    main()
