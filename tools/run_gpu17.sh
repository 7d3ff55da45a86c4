#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu17.log
: > $LOG
echo "=== 4-rank astaroth (grouped IPC flow) ===" >> $LOG
STENCIL_AMD_WIRE=cpu timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29791 benchmarks/astaroth.py --per-gpu 96 --iters 3 --warmup 1 2>&1 | grep -E "astaroth," >> $LOG
echo "=== big-memory jacobi 2500^3 (125 GB of buffers) ===" >> $LOG
timeout 400 python bench.py --gpus 1 --per-gpu 2500 --steps 5 --warmup 1 >> $LOG 2>&1
echo "=== astaroth 640^3 ===" >> $LOG
timeout 400 python benchmarks/astaroth.py --gpus 1 --per-gpu 640 --iters 3 --warmup 1 >> $LOG 2>&1
echo "=== astaroth no-compute final ===" >> $LOG
timeout 200 python benchmarks/astaroth.py --gpus 1 --iters 6 --warmup 1 --no-compute >> $LOG 2>&1
tail -6 $LOG
