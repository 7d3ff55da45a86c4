#!/bin/bash
# Single-node scaling curves (reference: scripts/summit/1node_jacobi3d.sh):
# the full measurement battery for a 1-8 GPU MI355X node. Shapes follow
# BASELINE.json configs 3-5.
set -e
cd "$(dirname "$0")/.."
STEPS=${STEPS:-30}
WARMUP=${WARMUP:-5}
MAXN=${MAXN:-8}
NS="1"; [ "$MAXN" -ge 2 ] && NS="1 2"; [ "$MAXN" -ge 4 ] && NS="1 2 4"; [ "$MAXN" -ge 8 ] && NS="1 2 4 8"

for N in $NS; do
  echo "=== jacobi3d weak 750^3/GPU, $N GPUs (1 rank/GPU, IPC + native RCCL wire) ==="
  if [ "$N" = 1 ]; then
    python bench.py --gpus 1 --steps $STEPS --warmup $WARMUP
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29641 \
      bench.py --gpus $N --steps $STEPS --warmup $WARMUP
  fi
done

for N in $NS; do
  [ "$N" = 1 ] && continue
  echo "=== jacobi3d weak, $N GPUs, SINGLE process (direct xGMI stores) ==="
  python bench.py --gpus $N --steps $STEPS --warmup $WARMUP
done

for N in $NS; do
  [ "$N" = 1 ] && continue
  echo "=== exchange weak 512^3/GPU, $N GPUs ==="
  python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
    --master-addr 127.0.0.1 --master-port 29642 \
    benchmarks/exchange_scaling.py --iters 30
done

echo "=== bench_exchange 1024^3 single GPU reference (BASELINE config 3 shape) ==="
python benchmarks/bench_exchange.py --size 512 --iters 20 || true

for N in $NS; do
  echo "=== jacobi3d STRONG 2048^3 r=2, $N GPUs (BASELINE config 4) ==="
  if [ "$N" = 1 ]; then
    python benchmarks/jacobi3d.py --size 2048 --radius 2 --strong --iters 20
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29644 \
      benchmarks/jacobi3d.py --size 2048 --radius 2 --strong --iters 20
  fi
done

for N in $NS; do
  echo "=== astaroth weak 256^3/GPU, $N GPUs ==="
  if [ "$N" = 1 ]; then
    python benchmarks/astaroth.py --per-gpu 256 --iters 20 --warmup 3
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29643 \
      benchmarks/astaroth.py --gpus $N --per-gpu 256 --iters 20 --warmup 3
  fi
done

echo "=== alltoallv A/B: direct-store kernels vs SDMA mesh vs RCCL ($MAXN GPUs) ==="
build/bench_alltoallv 16777216 20 "$MAXN" || true
build/bench_alltoallv 1048576 50 "$MAXN" || true

echo "=== pure C++ orchestrator: single process $MAXN GPUs ==="
build/jacobi3d_native 750 $STEPS "$MAXN" || true

if [ "$MAXN" -ge 2 ]; then
  echo "=== pure C++ orchestrator: $MAXN processes over FileBootstrap + RcclWire ==="
  tools/run_native_mp.sh "$MAXN" build/jacobi3d_native 1024 $STEPS || true
  tools/run_native_mp.sh "$MAXN" build/astaroth_native 512 10 || true
fi
