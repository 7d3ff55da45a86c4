#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu12.log
: > $LOG
echo "=== full gpu tests ===" >> $LOG
timeout 700 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG
echo "=== smoke ===" >> $LOG
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $LOG 2>&1
echo "=== bench jacobi (final) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 30 --warmup 5 >> $LOG 2>&1
echo "=== astaroth final + 384 ===" >> $LOG
timeout 300 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 >> $LOG 2>&1
timeout 300 python benchmarks/astaroth.py --gpus 1 --per-gpu 384 --iters 3 --warmup 1 >> $LOG 2>&1
timeout 200 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 --no-compute >> $LOG 2>&1
echo "=== bench_exchange ===" >> $LOG
timeout 200 python benchmarks/bench_exchange.py --gpus 1 --iters 20 >> $LOG 2>&1
echo "=== bench_pack r3 512 ===" >> $LOG
timeout 300 python benchmarks/bench_pack.py --size 512 --radius 3 --iters 30 >> $LOG 2>&1
echo "=== overlap study 750 ===" >> $LOG
timeout 200 python benchmarks/overlap_study.py --gpus 1 --size 750 --iters 10 >> $LOG 2>&1
echo "=== kernel stats (final) ===" >> $LOG
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof12 -o jac -- \
  python /root/repo/bench.py --gpus 1 --steps 8 --warmup 2 >> $LOG 2>&1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof12 -o ast -- \
  python /root/repo/benchmarks/astaroth.py --gpus 1 --iters 2 --warmup 1 >> $LOG 2>&1
echo done >> $LOG
tail -3 $LOG
