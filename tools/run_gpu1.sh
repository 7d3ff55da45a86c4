#!/bin/bash
# First GPU validation pass: gpu tests, smoke, bench, rocprof profile.
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo

echo "=== gpu tests ===" > gpurun_out/gpu1.log
timeout 600 python -m pytest tests -m gpu -x -q >> gpurun_out/gpu1.log 2>&1
echo "pytest exit: $?" >> gpurun_out/gpu1.log

echo "=== smoke ===" >> gpurun_out/gpu1.log
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> gpurun_out/gpu1.log 2>&1
echo "smoke exit: $?" >> gpurun_out/gpu1.log

echo "=== bench 750^3 ===" >> gpurun_out/gpu1.log
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 5 >> gpurun_out/gpu1.log 2>&1
echo "bench exit: $?" >> gpurun_out/gpu1.log

echo "=== rocprof ===" >> gpurun_out/gpu1.log
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -- \
  python /root/repo/bench.py --gpus 1 --steps 10 --warmup 2 >> /root/repo/gpurun_out/gpu1.log 2>&1
echo "rocprof exit: $?" >> /root/repo/gpurun_out/gpu1.log
tail -5 /root/repo/gpurun_out/gpu1.log
