#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu6.log
: > $LOG

echo "=== jacobi ablation probe 752 ===" >> $LOG
timeout 200 ./build/jacobi_probe 752 10 >> $LOG 2>&1

echo "=== mhd block sweep ===" >> $LOG
for B in 64x4x1 32x8x1 16x4x4 32x4x2 16x16x1 64x2x2; do
  echo "--- $B ---" >> $LOG
  STENCIL_MHD_BLOCK=$B timeout 250 python benchmarks/astaroth.py --gpus 1 --iters 4 --warmup 1 >> $LOG 2>&1
done

echo "=== quick numerics guard (mhd + jacobi) ===" >> $LOG
timeout 400 python -m pytest tests/test_gpu_mhd.py tests/test_gpu_native.py -q -m gpu >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG
tail -3 $LOG
