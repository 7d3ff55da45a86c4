#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu9.log
: > $LOG
echo "=== jacobi (z-spread tails + prio comm) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 25 --warmup 5 >> $LOG 2>&1
echo "=== jacobi no-overlap (sanity) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 10 --warmup 2 --no-overlap >> $LOG 2>&1
echo "=== astaroth ===" >> $LOG
timeout 300 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 >> $LOG 2>&1
echo "=== exchange weak 512 r1/r2 ===" >> $LOG
timeout 200 python benchmarks/exchange_scaling.py --gpus 1 --size 512 --radius 1 --iters 20 >> $LOG 2>&1
timeout 200 python benchmarks/exchange_scaling.py --gpus 1 --size 512 --radius 2 --iters 20 >> $LOG 2>&1
echo "=== numerics guard ===" >> $LOG
timeout 600 python -m pytest tests/test_gpu_mhd.py tests/test_gpu_native.py -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG
echo "=== jacobi kernel stats (final check) ===" >> $LOG
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof9 -o jac -- \
  python /root/repo/bench.py --gpus 1 --steps 8 --warmup 2 >> $LOG 2>&1
tail -4 $LOG
