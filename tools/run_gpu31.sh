#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu31.log
: > $LOG
echo "=== full GPU suite ===" >> $LOG
timeout 900 python -m pytest tests/ -m gpu -q >> $LOG 2>&1
echo "=== smoke ===" >> $LOG
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $LOG 2>&1
echo "=== bench default ===" >> $LOG
timeout 200 python bench.py 2>&1 | grep -E '^\{' >> $LOG
echo "=== 1500^3 / 2500^3 LDS refresh ===" >> $LOG
timeout 300 python benchmarks/jacobi3d.py --gpus 1 --size 1500 --iters 8 2>&1 | grep jacobi3d, >> $LOG
timeout 400 python bench.py --gpus 1 --per-gpu 2500 --steps 5 --warmup 1 2>&1 | grep -E '^\{' >> $LOG
echo "=== astaroth final ===" >> $LOG
timeout 200 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 8 --warmup 2 2>&1 | grep astaroth, >> $LOG
echo "=== kernel-stats profile (LDS graph mode) ===" >> $LOG
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof31 -o jac -- python /root/repo/bench.py --gpus 1 --steps 10 --warmup 2 >> $LOG 2>&1
tail -12 $LOG
