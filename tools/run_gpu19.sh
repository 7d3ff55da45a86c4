#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu19.log
: > $LOG
echo "=== gloo barrier latency on box (world=8, no GPU) ===" >> $LOG
cat > /tmp/bar_bench.py <<'PYEOF'
import os, time, torch.distributed as dist
rank = int(os.environ["RANK"])
dist.init_process_group("gloo")
for _ in range(20): dist.barrier()
t0 = time.perf_counter(); N = 300
for _ in range(N): dist.barrier()
dt = (time.perf_counter() - t0) / N
if rank == 0: print(f"gloo barrier world={dist.get_world_size()}: {dt*1e6:.1f} us", flush=True)
dist.destroy_process_group()
PYEOF
for w in 2 4 8; do
  timeout 120 python -m torch.distributed.run --nnodes=1 --nproc-per-node $w --master-addr 127.0.0.1 --master-port 2982$w /tmp/bar_bench.py 2>&1 | grep "gloo barrier" >> $LOG
done
echo "=== 4-rank astaroth grouped (preload fix) ===" >> $LOG
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29791 benchmarks/astaroth.py --per-gpu 96 --iters 3 --warmup 1 2>&1 | grep -E "astaroth,|Error|error|Traceback" >> $LOG
echo "exit=$?" >> $LOG
echo "=== MHD 512^3 + 640^3 block refinement ===" >> $LOG
for blk in 64x2x2 32x2x4; do
  STENCIL_MHD_BLOCK=$blk timeout 200 python benchmarks/astaroth.py --gpus 1 --per-gpu 512 --iters 4 --warmup 1 2>&1 | sed "s/^/blk=$blk /" | grep astaroth >> $LOG
done
for blk in 32x2x4 32x2x8 48x2x4 32x4x4 16x2x4 32x2x2 ; do
  STENCIL_MHD_BLOCK=$blk timeout 200 python benchmarks/astaroth.py --gpus 1 --per-gpu 640 --iters 3 --warmup 1 2>&1 | sed "s/^/blk=$blk /" | grep astaroth >> $LOG
done
echo "=== 256^3 re-check best-of ===" >> $LOG
for blk in 64x2x2 32x2x4; do
  STENCIL_MHD_BLOCK=$blk timeout 150 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 8 --warmup 2 2>&1 | sed "s/^/blk=$blk /" | grep astaroth >> $LOG
done
cat $LOG
