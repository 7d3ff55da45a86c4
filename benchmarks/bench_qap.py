#!/usr/bin/env python3
"""QAP solver quality/runtime on random, matched, and block-diagonal
matrices (reference: bin/bench_qap.cu). CPU-only."""
import argparse
import os
import random
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from stencil_amd import _C


def mat(n, fn):
    m = _C.SqMat(n)
    for i in range(n):
        for j in range(n):
            m.set(i, j, fn(i, j))
    return m


def identity_cost(w, d, n):
    return _C.qap_cost(w, d, list(range(n)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=8)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    rng = random.Random(args.seed)
    n = args.n

    cases = {
        "random": (
            mat(n, lambda i, j: 0 if i == j else rng.random()),
            mat(n, lambda i, j: 0 if i == j else rng.random()),
        ),
        "matched": (
            mat(n, lambda i, j: 0 if i == j else (i + j) % 5 + 1),
            mat(n, lambda i, j: 0 if i == j else 1.0 / ((i + j) % 5 + 1)),
        ),
        "block_diag": (
            mat(n, lambda i, j: 10.0 if (i // 2 == j // 2 and i != j) else 0.1),
            mat(n, lambda i, j: 0 if i == j else (1.0 if abs(i - j) == 1 else 4.0)),
        ),
    }
    print("case,n,identity_cost,solved_cost,improvement,solve_s", flush=True)
    for name, (w, d) in cases.items():
        t0 = time.perf_counter()
        f = _C.qap_solve(w, d)
        dt = time.perf_counter() - t0
        c0 = identity_cost(w, d, n)
        c1 = _C.qap_cost(w, d, f)
        print(f"{name},{n},{c0:.3f},{c1:.3f},{c0 / max(c1, 1e-12):.2f}x,{dt:.4f}", flush=True)


if __name__ == "__main__":
    main()
