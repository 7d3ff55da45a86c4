#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu21.log
: > $LOG
echo "=== graph numerics test ===" >> $LOG
timeout 300 python -m pytest tests/test_gpu_native.py::test_jacobi_step_graph_matches_eager tests/test_gpu_native.py::test_jacobi_matches_torch_reference -x -q >> $LOG 2>&1
echo "=== bench.py 750 (graph default) ===" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | grep -E "^\{" >> $LOG
echo "=== jacobi3d.py graph on/off A/B ===" >> $LOG
for i in 1 2; do
  timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 30 2>&1 | grep jacobi3d, | sed 's/^/graph /' >> $LOG
  STENCIL_AMD_STEP_GRAPH=0 timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 30 2>&1 | grep jacobi3d, | sed 's/^/eager /' >> $LOG
done
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 1024 --iters 15 2>&1 | grep jacobi3d, | sed 's/^/graph /' >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 512 --iters 40 2>&1 | grep jacobi3d, | sed 's/^/graph /' >> $LOG
STENCIL_AMD_STEP_GRAPH=0 timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 512 --iters 40 2>&1 | grep jacobi3d, | sed 's/^/eager /' >> $LOG
cat $LOG
