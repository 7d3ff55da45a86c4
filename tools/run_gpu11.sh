#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu11.log
: > $LOG
echo "=== jacobi (kernarg bases) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 25 --warmup 5 >> $LOG 2>&1
timeout 200 python bench.py --gpus 1 --steps 25 --warmup 5 >> $LOG 2>&1
echo "=== astaroth (kernarg bases) ===" >> $LOG
timeout 300 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 >> $LOG 2>&1
echo "=== numerics guard ===" >> $LOG
timeout 700 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG
tail -4 $LOG
