#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu18.log
: > $LOG
echo "=== 4-rank astaroth grouped (post set_device fix) ===" >> $LOG
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29791 benchmarks/astaroth.py --per-gpu 96 --iters 3 --warmup 1 >> $LOG 2>&1
echo "exit=$?" >> $LOG
echo "=== MHD block sweep @640^3 ===" >> $LOG
for blk in 64x2x2 32x4x2 64x4x1 128x2x1 64x2x4 32x2x4; do
  STENCIL_MHD_BLOCK=$blk timeout 200 python benchmarks/astaroth.py --gpus 1 --per-gpu 640 --iters 3 --warmup 1 2>&1 | sed "s/^/blk=$blk /" >> $LOG
done
echo "=== MHD block sweep @256^3 (confirm default) ===" >> $LOG
for blk in 64x2x2 64x2x4 32x2x4; do
  STENCIL_MHD_BLOCK=$blk timeout 150 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 6 --warmup 2 2>&1 | sed "s/^/blk=$blk /" >> $LOG
done
echo "=== bench_exchange 1024^3 r=1 (BASELINE target 3 config, 1 GPU) ===" >> $LOG
timeout 300 python benchmarks/bench_exchange.py --gpus 1 --size 1024 --iters 20 >> $LOG 2>&1
tail -40 $LOG
