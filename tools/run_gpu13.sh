#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu13.log
: > $LOG
echo "=== gpu tests (staged colo) ===" >> $LOG
timeout 700 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG
echo "=== jacobi regression check ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 20 --warmup 4 >> $LOG 2>&1
echo "=== astaroth regression check ===" >> $LOG
timeout 250 python benchmarks/astaroth.py --gpus 1 --iters 4 --warmup 1 >> $LOG 2>&1
tail -4 $LOG
