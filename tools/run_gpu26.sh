#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu26.log
: > $LOG
echo "=== numerics: graph(vecAll) vs eager vs torch ===" >> $LOG
timeout 300 python -m pytest tests/test_gpu_native.py -x -q >> $LOG 2>&1
echo "=== bench 750 within-box A/B: vecAll-graph vs eager ===" >> $LOG
for i in 1 2; do
  timeout 200 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | grep -E '^\{' | python -c "import json,sys; d=json.load(sys.stdin); print('graph+vecAll', round(d['ms_per_step'],4), round(d['value']/1e9,1))" >> $LOG
  STENCIL_AMD_STEP_GRAPH=0 timeout 200 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | grep -E '^\{' | python -c "import json,sys; d=json.load(sys.stdin); print('eager       ', round(d['ms_per_step'],4), round(d['value']/1e9,1))" >> $LOG
done
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 512 --iters 40 2>&1 | grep jacobi3d, >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 1024 --iters 15 2>&1 | grep jacobi3d, >> $LOG
cat $LOG
