#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu7.log
: > $LOG

echo "=== extended probe ===" >> $LOG
timeout 200 ./build/jacobi_probe 752 10 >> $LOG 2>&1

echo "=== jacobi (2-stream exterior) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 20 --warmup 4 >> $LOG 2>&1

echo "=== astaroth (32x4x2 + 2-stream) ===" >> $LOG
timeout 300 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 >> $LOG 2>&1

echo "=== gpu tests guard ===" >> $LOG
timeout 700 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG
tail -4 $LOG
