"""2-rank staged-IPC debug: dump staging buffer contents after exchange."""
import multiprocessing as mp
import os
import sys


def worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        sys.path.insert(0, "/root/repo")
        sys.path.insert(0, "/root/repo/tests")
        import numpy as np

        import stencil_amd as sa
        from util import fill_interiors, ripple_block

        dd = sa.DistributedDomain(12, 10, 8, backend="native")
        dd.set_radius(2)
        dd.set_gpus([0])
        h = dd.add_data(np.float32, "q")
        dd.realize()
        b = dd.backend
        out = [f"rank{rank} ipc={b._ipc_active} staging_recv={[(k, v[1]) for k, v in b._staging_recv.items()]}"]
        fill_interiors(dd, h)
        dd.exchange()
        for skey, (buf, total) in b._staging_recv.items():
            raw = np.frombuffer(b.engine.buffer_to_host(buf), dtype=np.float32)
            half0 = raw[: total // 4]
            half1 = raw[total // 4 : 2 * total // 4]
            out.append(
                f"rank{rank} staging{skey}: total={total} "
                f"h0[nonzero]={np.count_nonzero(half0)}/{half0.size} h0[:6]={half0[:6]} "
                f"h1[nonzero]={np.count_nonzero(half1)}/{half1.size}"
            )
        # check -x halo
        lo, hi = dd.local_rect(0)
        r = dd.radius
        flo = (lo[0] - 2, lo[1] - 2, lo[2] - 2)
        got = dd.read_global(0, flo, (flo[0] + 2, flo[1] + 6, flo[2] + 1), h)
        out.append(f"rank{rank} -x halo sample={got.ravel()[:8]}")
        q.put((rank, "\n".join(out)))
        dist.destroy_process_group()
    except Exception as e:
        import traceback

        q.put((rank, f"FAIL {e}\n{traceback.format_exc()}"))


if __name__ == "__main__":
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, 2, 29833, q)) for r in range(2)]
    for p in procs:
        p.start()
    for _ in procs:
        rank, msg = q.get(timeout=300)
        print(msg, flush=True)
    for p in procs:
        p.join(timeout=30)
