#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu15.log
: > $LOG
echo "=== full gpu tests ===" >> $LOG
timeout 700 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG
echo "=== smoke ===" >> $LOG
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $LOG 2>&1
echo "=== mhd block sweep (post-separable) ===" >> $LOG
for B in 32x4x2 64x4x1 32x8x1 16x4x4 64x2x2; do
  STENCIL_MHD_BLOCK=$B timeout 150 python benchmarks/astaroth.py --gpus 1 --iters 4 --warmup 1 2>/dev/null | sed "s/^/[$B] /" >> $LOG
done
echo "=== jacobi final ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 25 --warmup 5 >> $LOG 2>&1
echo "=== profiles ===" >> $LOG
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof15 -o ast -- \
  python /root/repo/benchmarks/astaroth.py --gpus 1 --iters 3 --warmup 1 >> $LOG 2>&1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof15 -o jac -- \
  python /root/repo/bench.py --gpus 1 --steps 8 --warmup 2 >> $LOG 2>&1
echo done >> $LOG
tail -3 $LOG
