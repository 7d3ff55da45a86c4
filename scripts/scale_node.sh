#!/bin/bash
# Single-node scaling curves (reference: scripts/summit/1node_jacobi3d.sh):
# runs the flagship jacobi3d weak-scaling bench and the exchange benchmark
# at 1/2/4/8 GPUs, one rank per GPU over IPC/RCCL.
set -e
cd "$(dirname "$0")/.."
STEPS=${STEPS:-30}
WARMUP=${WARMUP:-5}
for N in 1 2 4 8; do
  echo "=== jacobi3d weak, $N GPUs ==="
  if [ "$N" = 1 ]; then
    python bench.py --gpus 1 --steps $STEPS --warmup $WARMUP
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29641 \
      bench.py --gpus $N --steps $STEPS --warmup $WARMUP
  fi
done
for N in 2 4 8; do
  echo "=== exchange weak 512^3/GPU, $N GPUs ==="
  python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
    --master-addr 127.0.0.1 --master-port 29642 \
    benchmarks/exchange_scaling.py --iters 30
done
