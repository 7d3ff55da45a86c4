#!/usr/bin/env python3
"""Communication/computation overlap study (reference:
bin/measure_buf_exchange.cu): jacobi3d iteration time with and without
interior/exterior overlap, plus pure-exchange and pure-compute baselines.
overlap efficiency = (compute + exchange - overlapped) / exchange."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from stencil_amd.models.jacobi3d import Jacobi3D
from stencil_amd.utils.statistics import Statistics


def timed(fn, iters, warmup=3):
    for _ in range(warmup):
        fn()
    s = Statistics()
    for _ in range(iters):
        t0 = time.perf_counter()
        fn()
        s.insert(time.perf_counter() - t0)
    return s.trimean()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--size", type=int, default=512)
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()

    app = Jacobi3D((args.size,) * 3, gpus=list(range(args.gpus)))
    app.realize()

    t_overlap = timed(lambda: app.step(overlap=True), args.iters)
    t_seq = timed(lambda: app.step(overlap=False), args.iters)

    def exchange_only():
        app.dd.exchange()
        app.dd.swap()

    t_x = timed(exchange_only, args.iters)

    def compute_only():
        for li in range(app.dd.num_local()):
            lo, hi = app.dd.local_rect(li)
            app.dd.backend.jacobi_step(li, app.h.index, lo, hi, app.compute_lo, app.compute_hi)
        app.dd.backend.sync_compute()
        app.dd.swap()

    t_c = timed(compute_only, args.iters)

    hidden = max(0.0, (t_c + t_x) - t_overlap)
    eff = hidden / t_x if t_x > 0 else 0.0
    print(
        f"overlap_study,gpus={args.gpus},size={args.size},overlap_s={t_overlap:.6f},"
        f"sequential_s={t_seq:.6f},exchange_s={t_x:.6f},compute_s={t_c:.6f},"
        f"exchange_hidden={eff * 100:.0f}%",
        flush=True,
    )


if __name__ == "__main__":
    main()
