#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu20.log
: > $LOG
echo "=== jacobi overlap vs no-overlap (weak 750/1024, strong 2048 r2) ===" >> $LOG
for i in 1 2; do
  timeout 200 python benchmarks/jacobi3d.py --gpus 1 --per-gpu 750 --iters 30 --warmup 5 2>&1 | grep jacobi | sed 's/^/ovl   /' >> $LOG
  timeout 200 python benchmarks/jacobi3d.py --gpus 1 --per-gpu 750 --iters 30 --warmup 5 --no-overlap 2>&1 | grep jacobi | sed 's/^/noovl /' >> $LOG
done
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --per-gpu 1024 --iters 15 --warmup 3 2>&1 | grep jacobi | sed 's/^/ovl   /' >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --per-gpu 1024 --iters 15 --warmup 3 --no-overlap 2>&1 | grep jacobi | sed 's/^/noovl /' >> $LOG
timeout 300 python benchmarks/jacobi3d.py --gpus 1 --strong --size 2048 --radius 2 --iters 10 --warmup 2 2>&1 | grep jacobi | sed 's/^/ovl   /' >> $LOG
timeout 300 python benchmarks/jacobi3d.py --gpus 1 --strong --size 2048 --radius 2 --iters 10 --warmup 2 --no-overlap 2>&1 | grep jacobi | sed 's/^/noovl /' >> $LOG
echo "=== graphs on, both modes, 750 ===" >> $LOG
STENCIL_AMD_GRAPHS=1 timeout 200 python benchmarks/jacobi3d.py --gpus 1 --per-gpu 750 --iters 30 --warmup 5 2>&1 | grep jacobi | sed 's/^/ovl+g   /' >> $LOG
STENCIL_AMD_GRAPHS=1 timeout 200 python benchmarks/jacobi3d.py --gpus 1 --per-gpu 750 --iters 30 --warmup 5 --no-overlap 2>&1 | grep jacobi | sed 's/^/noovl+g /' >> $LOG
cat $LOG
