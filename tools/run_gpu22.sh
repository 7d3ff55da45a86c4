#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu22.log
: > $LOG
echo "=== graph tests (jacobi + mhd) ===" >> $LOG
timeout 400 python -m pytest tests/test_gpu_native.py::test_jacobi_step_graph_matches_eager tests/test_gpu_mhd.py -x -q >> $LOG 2>&1
echo "=== astaroth graph on/off A/B 256^3 ===" >> $LOG
for i in 1 2; do
  timeout 200 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 8 --warmup 2 2>&1 | grep astaroth, | sed 's/^/graph /' >> $LOG
  STENCIL_AMD_STEP_GRAPH=0 timeout 200 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 8 --warmup 2 2>&1 | grep astaroth, | sed 's/^/eager /' >> $LOG
done
echo "=== astaroth graph 512 ===" >> $LOG
timeout 250 python benchmarks/astaroth.py --gpus 1 --per-gpu 512 --iters 4 --warmup 1 2>&1 | grep astaroth, | sed 's/^/graph /' >> $LOG
echo "=== astaroth graph 128 (small-grid regime) ===" >> $LOG
timeout 150 python benchmarks/astaroth.py --gpus 1 --per-gpu 128 --iters 15 --warmup 3 2>&1 | grep astaroth, | sed 's/^/graph /' >> $LOG
STENCIL_AMD_STEP_GRAPH=0 timeout 150 python benchmarks/astaroth.py --gpus 1 --per-gpu 128 --iters 15 --warmup 3 2>&1 | grep astaroth, | sed 's/^/eager /' >> $LOG
cat $LOG
