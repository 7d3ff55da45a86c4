#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu16.log
: > $LOG
echo "=== 8-rank-on-1-GPU bench smoke (driver-width rendezvous) ===" >> $LOG
STENCIL_AMD_WIRE=cpu timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 --master-port 29781 bench.py --gpus 8 --per-gpu 128 --steps 5 --warmup 1 2>&1 | grep -E "^\{" >> $LOG
echo "=== jacobi r=2 / r=3 weak 750 ===" >> $LOG
timeout 200 python benchmarks/jacobi3d.py --size 750 --radius 2 --iters 15 >> $LOG 2>&1
timeout 200 python benchmarks/jacobi3d.py --size 750 --radius 3 --iters 15 >> $LOG 2>&1
echo "=== native C++ example 512^3 ===" >> $LOG
timeout 200 ./build/jacobi3d_native 512 30 >> $LOG 2>&1
echo "=== long soaks ===" >> $LOG
timeout 500 python - >> $LOG 2>&1 <<'PYEOF'
import numpy as np
from stencil_amd.models.jacobi3d import Jacobi3D

app = Jacobi3D((256, 256, 256), gpus=[0], halo_multiplier=2)
app.realize()
for i in range(1000):
    app.step()
lo, hi = app.dd.local_rect(0)
a = app.dd.read_global(0, lo, hi, app.h)
assert np.isfinite(a).all() and 0 <= a.min() and a.max() <= 1
print("jacobi m=2 1000-iter soak ok", a.mean())
PYEOF
timeout 500 python - >> $LOG 2>&1 <<'PYEOF'
import numpy as np
from stencil_amd.models.astaroth import Astaroth, FIELDS

app = Astaroth((128, 128, 128), gpus=[0])
app.realize()
app.init_fields()
for i in range(300):
    app.step()
for n in FIELDS:
    assert np.isfinite(app.read_field(0, n)).all(), n
print("astaroth 300-iter soak ok")
PYEOF
echo "=== final exchange/overlap numbers ===" >> $LOG
timeout 200 python benchmarks/exchange_scaling.py --gpus 1 --size 512 --radius 1 --iters 20 >> $LOG 2>&1
timeout 200 python benchmarks/overlap_study.py --gpus 1 --size 750 --iters 10 >> $LOG 2>&1
tail -8 $LOG
