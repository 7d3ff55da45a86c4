#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu8.log
: > $LOG
echo "=== jacobi (uniform ptrs) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 20 --warmup 4 >> $LOG 2>&1
echo "=== astaroth (uniform ptrs) ===" >> $LOG
timeout 300 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 >> $LOG 2>&1
echo "=== numerics guard ===" >> $LOG
timeout 600 python -m pytest tests/test_gpu_mhd.py tests/test_gpu_native.py -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG
tail -4 $LOG
