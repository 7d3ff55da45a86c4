#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu32.log
: > $LOG
echo "=== jacobi numerics suite (incl 2-domain overlap/no-overlap/torch) ===" >> $LOG
timeout 400 python -m pytest tests/test_gpu_native.py tests/test_gpu_mp.py -x -q >> $LOG 2>&1
echo "=== eager bench (mode 1 now) vs graph ===" >> $LOG
STENCIL_AMD_STEP_GRAPH=0 timeout 200 python bench.py --gpus 1 --steps 25 --warmup 4 2>&1 | grep -E '^\{' | python -c "import json,sys; d=json.load(sys.stdin); print('eager-m1', round(d['ms_per_step'],4), round(d['value']/1e9,1))" >> $LOG
timeout 200 python bench.py --gpus 1 --steps 25 --warmup 4 2>&1 | grep -E '^\{' | python -c "import json,sys; d=json.load(sys.stdin); print('graph   ', round(d['ms_per_step'],4), round(d['value']/1e9,1))" >> $LOG
echo "=== 4-rank IPC (mode 2) + 2-rank wire (mode 1) smokes ===" >> $LOG
timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29794 bench.py --gpus 4 --per-gpu 192 --steps 8 --warmup 2 2>&1 | grep -E '^\{' >> $LOG
STENCIL_AMD_IPC=0 timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29795 bench.py --gpus 2 --per-gpu 256 --steps 8 --warmup 2 2>&1 | grep -E '^\{' >> $LOG
echo "=== multi-rank numerics: 2-rank wire jacobi vs torch ===" >> $LOG
timeout 400 python -m pytest tests/test_gpu_mp_mhd.py -x -q >> $LOG 2>&1
cat $LOG
