#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu33.log
: > $LOG
echo "=== MHD ychunk sweep @256^3 (within-box A/B) ===" >> $LOG
for yc in 0 8 16 32 64; do
  STENCIL_MHD_YCHUNK=$yc timeout 150 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 6 --warmup 2 2>&1 | grep astaroth, | sed "s/^/yc=$yc /" >> $LOG
done
echo "=== ychunk numerics check ===" >> $LOG
STENCIL_MHD_YCHUNK=16 timeout 300 python -m pytest tests/test_gpu_mhd.py::test_mhd_matches_numpy_reference -x -q >> $LOG 2>&1
echo "=== best-at-640 recheck with ychunk ===" >> $LOG
for yc in 0 16; do
  STENCIL_MHD_YCHUNK=$yc timeout 200 python benchmarks/astaroth.py --gpus 1 --per-gpu 640 --iters 3 --warmup 1 2>&1 | grep astaroth, | sed "s/^/yc=$yc /" >> $LOG
done
cat $LOG
