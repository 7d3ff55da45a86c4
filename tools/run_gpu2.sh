#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo
LOG=gpurun_out/gpu2.log
: > $LOG

echo "=== gpu tests ===" >> $LOG
timeout 900 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG

echo "=== bench jacobi 750^3 (v4 kernel) ===" >> $LOG
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 >> $LOG 2>&1

echo "=== astaroth 256^3 ===" >> $LOG
timeout 600 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 >> $LOG 2>&1
echo "=== astaroth 256^3 no-compute ===" >> $LOG
timeout 300 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 --no-compute >> $LOG 2>&1

echo "=== bench_exchange 128^3 ===" >> $LOG
timeout 300 python benchmarks/bench_exchange.py --gpus 1 --iters 20 >> $LOG 2>&1

echo "=== rocprof kernel stats (jacobi bench) ===" >> $LOG
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof2 -o jstats -- \
  python /root/repo/bench.py --gpus 1 --steps 10 --warmup 2 >> $LOG 2>&1
echo "rocprof exit: $?" >> $LOG
cd /root/repo
find gpurun_out/prof2 -name "*.csv" | head >> $LOG
tail -3 $LOG
