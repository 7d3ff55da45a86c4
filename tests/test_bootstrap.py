"""FileBootstrap (the C++ multi-process control plane) protocol test on
CPU: N OS processes allgather payloads through a shared directory."""
import multiprocessing as mp

from stencil_amd import _C


def _worker(dir_, rank, world, q):
    b = _C.FileBootstrap(dir_, rank, world)
    got = b.allgather("uid", f"payload-{rank}".encode())
    # a second phase must not collide with the first
    got2 = b.allgather("slots", bytes([rank]) * 4)
    q.put((rank, [bytes(g) for g in got], [bytes(g) for g in got2]))


def test_file_bootstrap_allgather(tmp_path):
    world = 4
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(str(tmp_path), r, world, q)) for r in range(world)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=60) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    expect1 = [f"payload-{r}".encode() for r in range(world)]
    expect2 = [bytes([r]) * 4 for r in range(world)]
    for rank, got1, got2 in results:
        assert got1 == expect1
        assert got2 == expect2
