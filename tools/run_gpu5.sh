#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu5.log
: > $LOG

echo "=== gpu tests (IPC fix) ===" >> $LOG
timeout 700 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG

echo "=== jacobi (XCD swizzle) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 20 --warmup 4 >> $LOG 2>&1

echo "=== astaroth 256^3 (3-pass) ===" >> $LOG
timeout 400 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 >> $LOG 2>&1

echo "=== astaroth kernel stats ===" >> $LOG
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof5 -o ast -- \
  python /root/repo/benchmarks/astaroth.py --gpus 1 --iters 2 --warmup 1 >> $LOG 2>&1

echo "=== jacobi kernel stats ===" >> $LOG
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof5 -o jac -- \
  python /root/repo/bench.py --gpus 1 --steps 8 --warmup 2 >> $LOG 2>&1

echo "=== jacobi pmc FETCH/WRITE ===" >> $LOG
timeout 300 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE --output-format csv -d /root/repo/gpurun_out/prof5pmc -o jpmc -- \
  python /root/repo/bench.py --gpus 1 --steps 2 --warmup 1 --per-gpu 384 >> $LOG 2>&1
echo "pmc exit: $?" >> $LOG

cd /root/repo
echo "=== machine info ===" >> $LOG
timeout 120 python benchmarks/machine_info.py >> $LOG 2>&1
echo "=== native example ===" >> $LOG
timeout 120 ./build/jacobi3d_native 384 10 >> $LOG 2>&1
tail -4 $LOG
