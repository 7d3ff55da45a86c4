#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu3.log
: > $LOG

echo "=== gpu tests ===" >> $LOG
timeout 900 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG

echo "=== bench jacobi 750^3 (z-march kernel) ===" >> $LOG
timeout 300 python bench.py --gpus 1 --steps 30 --warmup 5 >> $LOG 2>&1

echo "=== rocprof kernel stats ===" >> $LOG
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof3 -o jk -- \
  python /root/repo/bench.py --gpus 1 --steps 10 --warmup 2 >> $LOG 2>&1
echo "rocprof exit: $?" >> $LOG
echo "=== rocprof pmc (jacobi) ===" >> $LOG
timeout 600 rocprofv3 --pmc SQ_WAVES FETCH_SIZE WRITE_SIZE --output-format csv -d /root/repo/gpurun_out/prof3pmc -o jpmc -- \
  python /root/repo/bench.py --gpus 1 --steps 5 --warmup 1 >> $LOG 2>&1
echo "pmc exit: $?" >> $LOG
find /root/repo/gpurun_out/prof3* -name "*.csv" >> $LOG 2>&1
tail -5 $LOG
