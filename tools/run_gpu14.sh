#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu14.log
: > $LOG
echo "=== jacobi 1024^3 and 1500^3 per GPU ===" >> $LOG
timeout 200 python bench.py --gpus 1 --per-gpu 1024 --steps 15 --warmup 3 >> $LOG 2>&1
timeout 300 python bench.py --gpus 1 --per-gpu 1500 --steps 10 --warmup 2 >> $LOG 2>&1
echo "=== astaroth 512^3 ===" >> $LOG
timeout 400 python benchmarks/astaroth.py --gpus 1 --per-gpu 512 --iters 3 --warmup 1 >> $LOG 2>&1
echo "=== stability: jacobi 200 iters + value check ===" >> $LOG
timeout 400 python - >> $LOG 2>&1 <<'PYEOF'
import numpy as np
from stencil_amd.models.jacobi3d import Jacobi3D

app = Jacobi3D((384, 384, 384), gpus=[0])
app.realize()
for i in range(200):
    app.step()
lo, hi = app.dd.local_rect(0)
arr = app.dd.read_global(0, lo, hi, app.h)
assert np.isfinite(arr).all() and arr.min() >= 0 and arr.max() <= 1
print("jacobi 200-iter soak ok", arr.min(), arr.max(), arr.mean())
PYEOF
echo "=== stability: astaroth 30 iters finite ===" >> $LOG
timeout 400 python - >> $LOG 2>&1 <<'PYEOF'
import numpy as np
from stencil_amd.models.astaroth import Astaroth, FIELDS

app = Astaroth((128, 128, 128), gpus=[0])
app.realize()
app.init_fields()
for i in range(30):
    app.step()
for n in FIELDS:
    a = app.read_field(0, n)
    assert np.isfinite(a).all(), n
print("astaroth 30-iter soak ok")
PYEOF
echo "=== 4-rank IPC bench 256^3/rank ===" >> $LOG
export STENCIL_AMD_WIRE=cpu
timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29761 bench.py --gpus 4 --per-gpu 256 --steps 10 --warmup 2 >> $LOG 2>&1
tail -8 $LOG
