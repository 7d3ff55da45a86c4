#!/usr/bin/env python3
"""xGMI link bandwidth sweep between GPU pairs (reference: bin/pingpong.cu
node-pair MPI sweep 2^0..2^27 B). Single process, hipMemcpyPeerAsync; with
--all-pairs sweeps every device pair."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from stencil_amd import _C
from stencil_amd.utils.statistics import Statistics


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--src", type=int, default=0)
    ap.add_argument("--dst", type=int, default=1)
    ap.add_argument("--all-pairs", action="store_true")
    ap.add_argument("--max-log2", type=int, default=27)
    ap.add_argument("--iters", type=int, default=10)
    args = ap.parse_args()

    n = _C.device_count()
    pairs = (
        [(a, b) for a in range(n) for b in range(n) if a != b]
        if args.all_pairs
        else [(args.src, args.dst)]
    )
    print("src,dst,bytes,GBs", flush=True)
    for src, dst in pairs:
        if not _C.ExchangeEngine.can_access_peer(src, dst):
            print(f"{src},{dst},-,no-peer-access", flush=True)
            continue
        for lg in range(10, args.max_log2 + 1, 2):
            nbytes = 1 << lg
            gbs = _C.peer_copy_bandwidth(src, dst, nbytes, args.iters)
            print(f"{src},{dst},{nbytes},{gbs:.2f}", flush=True)


if __name__ == "__main__":
    main()
