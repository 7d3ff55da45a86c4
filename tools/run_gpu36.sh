#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu36.log
: > $LOG
echo "=== FULL GPU suite (30 tests) ===" >> $LOG
timeout 900 python -m pytest tests/ -m gpu -q >> $LOG 2>&1
echo "=== smoke ===" >> $LOG
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $LOG 2>&1
echo "=== bench default ===" >> $LOG
timeout 200 python bench.py 2>&1 | grep -E '^\{' >> $LOG
echo "=== astaroth ===" >> $LOG
timeout 150 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 8 --warmup 2 2>&1 | grep astaroth, >> $LOG
echo "=== 8-rank-on-1-GPU driver-shape smoke ===" >> $LOG
STENCIL_AMD_WIRE=cpu timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 --master-port 29796 bench.py --gpus 8 --per-gpu 128 --steps 5 --warmup 1 2>&1 | grep -E '^\{' >> $LOG
tail -12 $LOG
