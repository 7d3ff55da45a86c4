#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu10.log
: > $LOG
echo "=== jacobi A (z-spread, default prio) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 25 --warmup 5 >> $LOG 2>&1
timeout 200 python bench.py --gpus 1 --steps 25 --warmup 5 >> $LOG 2>&1
echo "=== probe again ===" >> $LOG
timeout 200 ./build/jacobi_probe 752 12 >> $LOG 2>&1
tail -6 $LOG
