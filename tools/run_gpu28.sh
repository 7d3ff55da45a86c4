#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu28.log
: > $LOG
echo "=== strong 2048 r=2 refresh (vecAll) ===" >> $LOG
timeout 300 python benchmarks/jacobi3d.py --gpus 1 --strong --size 2048 --radius 2 --iters 10 2>&1 | grep jacobi3d, >> $LOG
echo "=== 1500^3 refresh ===" >> $LOG
timeout 300 python benchmarks/jacobi3d.py --gpus 1 --size 1500 --iters 8 2>&1 | grep jacobi3d, >> $LOG
echo "=== JAC_BLOCK sweep under vecAll @750 ===" >> $LOG
for blk in 64x4 64x2 32x8 128x2 32x4 256x1; do
  STENCIL_JAC_BLOCK=$blk timeout 150 python benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 20 2>&1 | grep jacobi3d, | sed "s/^/blk=$blk /" >> $LOG
done
echo "=== r=2 / r=3 weak 750 refresh ===" >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --radius 2 --iters 15 2>&1 | grep jacobi3d, >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --radius 3 --iters 15 2>&1 | grep jacobi3d, >> $LOG
echo "=== halo multiplier refresh (graph path excluded, m>1 eager) ===" >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --halo-multiplier 2 --iters 15 2>&1 | grep jacobi3d, >> $LOG
cat $LOG
