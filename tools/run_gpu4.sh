#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu4.log
: > $LOG

echo "=== gpu tests ===" >> $LOG
timeout 700 python -m pytest tests -m gpu -q >> $LOG 2>&1
echo "pytest exit: $?" >> $LOG

echo "=== jacobi block sweep ===" >> $LOG
for B in 64x4 32x8 16x16 128x2; do
  echo "--- block $B ---" >> $LOG
  STENCIL_JAC_BLOCK=$B timeout 200 python bench.py --gpus 1 --steps 15 --warmup 3 2>>$LOG | python -c "import json,sys; d=json.load(sys.stdin); print('$B', round(d['value']/1e9,1),'Gcells/s', round(d['ms_per_step'],3),'ms')" >> $LOG 2>&1
done

echo "=== jacobi graphs mode ===" >> $LOG
STENCIL_AMD_GRAPHS=1 timeout 200 python bench.py --gpus 1 --steps 15 --warmup 3 >> $LOG 2>&1

echo "=== astaroth 256^3 (split kernels) ===" >> $LOG
timeout 400 python benchmarks/astaroth.py --gpus 1 --iters 5 --warmup 1 >> $LOG 2>&1

echo "=== astaroth kernel stats ===" >> $LOG
cd /tmp && export TMPDIR=/tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof4 -o ast -- \
  python /root/repo/benchmarks/astaroth.py --gpus 1 --iters 2 --warmup 1 >> $LOG 2>&1
echo "ast rocprof exit: $?" >> $LOG

echo "=== jacobi pmc (256^3) ===" >> $LOG
timeout 400 rocprofv3 --pmc SQ_WAVES FETCH_SIZE WRITE_SIZE --output-format csv -d /root/repo/gpurun_out/prof4pmc -o jpmc -- \
  python /root/repo/bench.py --gpus 1 --steps 3 --warmup 1 --per-gpu 256 >> $LOG 2>&1
echo "pmc exit: $?" >> $LOG

cd /root/repo
echo "=== overlap study 512^3 ===" >> $LOG
timeout 200 python benchmarks/overlap_study.py --gpus 1 --size 512 --iters 10 >> $LOG 2>&1
tail -3 $LOG
