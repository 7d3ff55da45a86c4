#!/bin/bash
# Torchrun-free multi-process launcher for the pure-C++ binaries: one
# process per GPU, control plane = FileBootstrap (shared dir), data plane
# = RcclWire over xGMI. Usage: tools/run_native_mp.sh <world> <binary> [args...]
set -e
WORLD=$1; shift
BIN=$1; shift
DIR=$(mktemp -d /tmp/stencil_boot.XXXXXX)
trap 'rm -rf "$DIR"' EXIT
pids=()
for r in $(seq 0 $((WORLD-1))); do
  STENCIL_RANK=$r STENCIL_WORLD=$WORLD STENCIL_LOCAL_RANK=$r \
    STENCIL_BOOTSTRAP_DIR=$DIR "$BIN" "$@" &
  pids+=($!)
done
rc=0
for p in "${pids[@]}"; do wait "$p" || rc=1; done
exit $rc
