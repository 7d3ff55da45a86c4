#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu20b.log
: > $LOG
for i in 1 2; do
  timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 30 2>&1 | grep jacobi3d, | sed 's/^/ovl   /' >> $LOG
  timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 30 --no-overlap 2>&1 | grep jacobi3d, | sed 's/^/noovl /' >> $LOG
done
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 1024 --iters 15 2>&1 | grep jacobi3d, | sed 's/^/ovl   /' >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 1024 --iters 15 --no-overlap 2>&1 | grep jacobi3d, | sed 's/^/noovl /' >> $LOG
timeout 300 python benchmarks/jacobi3d.py --gpus 1 --strong --size 2048 --radius 2 --iters 10 2>&1 | grep jacobi3d, | sed 's/^/ovl   /' >> $LOG
timeout 300 python benchmarks/jacobi3d.py --gpus 1 --strong --size 2048 --radius 2 --iters 10 --no-overlap 2>&1 | grep jacobi3d, | sed 's/^/noovl /' >> $LOG
STENCIL_AMD_GRAPHS=1 timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 30 --no-overlap 2>&1 | grep jacobi3d, | sed 's/^/noovl+g /' >> $LOG
cat $LOG
