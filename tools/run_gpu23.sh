#!/bin/bash
set -x
cd /tmp
export TMPDIR=/tmp
export PYTHONPATH=/root/repo
OUT=/root/repo/gpurun_out
LOG=$OUT/gpu23.log
: > $LOG
echo "=== kernel stats: jacobi graph mode 750^3 ===" >> $LOG
timeout 300 rocprofv3 --kernel-trace --stats -d $OUT/prof23j -o jac -- python /root/repo/bench.py --gpus 1 --steps 10 --warmup 2 >> $LOG 2>&1
echo "=== kernel stats: astaroth graph mode 256^3 ===" >> $LOG
timeout 300 rocprofv3 --kernel-trace --stats -d $OUT/prof23a -o ast -- python /root/repo/benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 4 --warmup 1 >> $LOG 2>&1
echo "=== PMC single-counter: FETCH_SIZE jacobi ===" >> $LOG
timeout 300 rocprofv3 --pmc FETCH_SIZE -d $OUT/pmc23jf -o jf -- python /root/repo/benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 3 >> $LOG 2>&1
echo "pmc_fetch_exit=$?" >> $LOG
echo "=== PMC single-counter: WRITE_SIZE jacobi ===" >> $LOG
timeout 300 rocprofv3 --pmc WRITE_SIZE -d $OUT/pmc23jw -o jw -- python /root/repo/benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 3 >> $LOG 2>&1
echo "pmc_write_exit=$?" >> $LOG
echo "=== PMC single-counter: FETCH_SIZE astaroth ===" >> $LOG
timeout 300 rocprofv3 --pmc FETCH_SIZE -d $OUT/pmc23af -o af -- python /root/repo/benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 2 --warmup 0 >> $LOG 2>&1
echo "pmc_ast_exit=$?" >> $LOG
ls -R $OUT/prof23j $OUT/prof23a $OUT/pmc23jf 2>/dev/null | head -20 >> $LOG
tail -20 $LOG
