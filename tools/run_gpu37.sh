#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu37.log
: > $LOG
echo "=== bench_exchange finals (128 + 1024) ===" >> $LOG
timeout 200 python benchmarks/bench_exchange.py --gpus 1 --iters 30 >> $LOG 2>&1
timeout 300 python benchmarks/bench_exchange.py --gpus 1 --size 1024 --iters 20 >> $LOG 2>&1
echo "=== exchange_scaling weak 512 ===" >> $LOG
timeout 200 python benchmarks/exchange_scaling.py --gpus 1 --size 512 --iters 30 >> $LOG 2>&1
echo "=== bench_pack final ===" >> $LOG
timeout 200 python benchmarks/bench_pack.py --size 512 --radius 3 >> $LOG 2>&1
echo "=== 3000^3 jacobi (216 GB resident) ===" >> $LOG
timeout 500 python bench.py --gpus 1 --per-gpu 3000 --steps 4 --warmup 1 2>&1 | grep -E '^\{' >> $LOG
echo "=== 4-rank astaroth full defaults ===" >> $LOG
timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29797 benchmarks/astaroth.py --per-gpu 96 --iters 3 --warmup 1 2>&1 | grep astaroth, >> $LOG
tail -30 $LOG
