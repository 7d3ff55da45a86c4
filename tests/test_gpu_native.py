"""Native (HIP) backend tests on a real MI355X. These are the GPU analogs
of the reference's test_cuda tier: full-region ripple exchange through the
batched translate kernels (incl. the same-GPU multi-subdomain trick,
reference test_exchange.cu:52), pack/unpack roundtrips, and jacobi
numerics against the plain-PyTorch fp32 reference."""
import numpy as np
import pytest

import stencil_amd as sa
from stencil_amd import _C

from util import check_full_regions, fill_interiors, ripple_block

pytestmark = pytest.mark.gpu


def make_dd(size, radius, n_domains=1):
    dd = sa.DistributedDomain(*size, backend="native")
    dd.set_radius(radius)
    dd.set_gpus([0] * n_domains)  # same-GPU fake-multi-GPU
    return dd


@pytest.mark.parametrize("r", [1, 2])
@pytest.mark.parametrize("n_domains", [1, 2, 8])
def test_ripple_exchange_native(r, n_domains):
    dd = make_dd((12, 10, 8), r, n_domains)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h)
    dd.exchange()
    check_full_regions(dd, h)


def test_ripple_asymmetric_native():
    r = _C.Radius.constant(1)
    r.set_dir(1, 0, 0, 2)
    dd = make_dd((12, 10, 8), 0, 2)
    dd.set_radius(r)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h)
    dd.exchange()
    check_full_regions(dd, h)


def test_multi_quantity_mixed_sizes():
    dd = make_dd((10, 10, 10), 1, 2)
    h32 = dd.add_data(np.float32, "a")
    h64 = dd.add_data(np.float64, "b")
    dd.realize()
    fill_interiors(dd, h32)
    for li in range(dd.num_local()):
        lo, hi = dd.local_rect(li)
        dd.write_global(li, lo, ripple_block(lo, hi, dd.size, 2.0).astype(np.float64), h64)
    dd.exchange()
    check_full_regions(dd, h32)
    # fp64 full-region check
    for li in range(dd.num_local()):
        from util import full_region_of

        flo, fhi = full_region_of(dd, li)
        got = dd.read_global(li, flo, fhi, h64)
        want = ripple_block(flo, fhi, dd.size, 2.0).astype(np.float64)
        assert np.array_equal(got, want)


def test_exchange_after_swap_native():
    dd = make_dd((8, 8, 8), 1, 2)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h, scale=3.0)
    dd.swap()
    fill_interiors(dd, h, scale=5.0)
    dd.exchange()
    check_full_regions(dd, h, scale=5.0)
    dd.swap()
    for li in range(dd.num_local()):
        lo, hi = dd.local_rect(li)
        assert np.array_equal(dd.read_global(li, lo, hi, h), ripple_block(lo, hi, dd.size, 3.0))


def test_pack_unpack_roundtrip():
    """engine-level: pack a halo region to a buffer, scribble the region,
    unpack, verify restoration"""
    dd = make_dd((8, 6, 4), 1, 1)
    h = dd.add_data(np.float32, "q")
    # use a raw engine (the dd one is finalized by realize)
    dom = _C.LocalDomain(_C.Vec3(8, 6, 4), _C.Vec3(0, 0, 0), 0)
    dom.set_radius(_C.Radius.constant(1))
    dom.add_data(4, "q")
    dom.realize()
    eng = _C.ExchangeEngine([dom])
    ext = _C.Vec3(8, 6, 1)  # +z face interior
    pos = dom.halo_pos(_C.Vec3(0, 0, 1), False)
    nbytes = 8 * 6 * 4
    buf = eng.create_buffer(0, nbytes)
    eng.add_pack(0, buf, 0, pos, ext, 0)
    hpos = dom.halo_pos(_C.Vec3(0, 0, 1), True)
    eng.add_unpack(0, buf, 0, hpos, ext, 0)
    eng.finalize()
    data = np.arange(8 * 6, dtype=np.float32).reshape(1, 6, 8)
    dom.region_from_host(data.tobytes(), pos, ext, 0)
    eng.launch_packs()
    eng.sync_packs()
    packed = np.frombuffer(eng.buffer_to_host(buf), dtype=np.float32).reshape(1, 6, 8)
    assert np.array_equal(packed, data)
    eng.launch_unpacks()
    eng.sync_packs()
    halo = np.frombuffer(dom.region_to_host(hpos, ext, 0), dtype=np.float32).reshape(1, 6, 8)
    assert np.array_equal(halo, data)


def _run_jacobi(backend, size, steps, n_domains):
    from stencil_amd.models.jacobi3d import Jacobi3D

    app = Jacobi3D(size, backend=backend, gpus=[0] * n_domains)
    app.realize()
    fill_interiors(app.dd, app.h)
    for _ in range(steps):
        app.step()
    out = []
    for li in range(app.dd.num_local()):
        lo, hi = app.dd.local_rect(li)
        out.append((lo, app.dd.read_global(li, lo, hi, app.h)))
    return out


def test_jacobi_matches_torch_reference():
    """HIP jacobi kernel vs the plain-PyTorch fp32 reference, 3 steps with
    exchanges, 2 subdomains"""
    size = (20, 16, 12)
    native = _run_jacobi("native", size, 3, 2)
    ref = _run_jacobi("torch", size, 3, 2)
    assert len(native) == len(ref)
    for (lo_n, a), (lo_t, b) in zip(native, ref):
        assert lo_n == lo_t
        np.testing.assert_allclose(a, b, rtol=1e-6, atol=1e-6)


def test_jacobi_overlap_equals_no_overlap():
    from stencil_amd.models.jacobi3d import Jacobi3D

    outs = []
    for overlap in (True, False):
        app = Jacobi3D((16, 16, 16), backend="native", gpus=[0, 0])
        app.realize()
        fill_interiors(app.dd, app.h)
        for _ in range(2):
            app.step(overlap=overlap)
        outs.append(
            [
                app.dd.read_global(li, *app.dd.local_rect(li), app.h)
                for li in range(app.dd.num_local())
            ]
        )
    for a, b in zip(outs[0], outs[1]):
        np.testing.assert_array_equal(a, b)


def test_field_stats_matches_numpy():
    dd = make_dd((14, 12, 10), 1, 1)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h, scale=2.5)
    lo, hi = dd.local_rect(0)
    st = _C.field_stats(
        dd.backend.engine, 0, 0, _C.Rect3(_C.Vec3(*lo), _C.Vec3(*hi))
    )
    arr = dd.read_global(0, lo, hi, h).astype(np.float64)
    assert st.min == pytest.approx(arr.min())
    assert st.max == pytest.approx(arr.max())
    assert st.rms == pytest.approx(np.sqrt((arr ** 2).mean()), rel=1e-12)


def test_halo_multiplier_native_matches_plain():
    from stencil_amd.models.jacobi3d import Jacobi3D

    size = (18, 15, 12)
    outs = []
    for mult in (1, 2):
        app = Jacobi3D(size, backend="native", gpus=[0, 0], halo_multiplier=mult)
        app.realize()
        fill_interiors(app.dd, app.h)
        for _ in range(4):
            app.step()
        outs.append(
            [
                app.dd.read_global(li, *app.dd.local_rect(li), app.h)
                for li in range(app.dd.num_local())
            ]
        )
    for a, b in zip(outs[0], outs[1]):
        np.testing.assert_array_equal(a, b)


@pytest.mark.parametrize("seed", [0, 3, 5, 8])
def test_fuzz_ripple_native(seed):
    """randomized radii/quantities through the native engine (mirrors the
    CPU fuzz tier)"""
    import random

    from test_fuzz_exchange import random_radius
    from util import check_valid_regions

    rng = random.Random(seed)
    size = tuple(rng.randint(6, 24) for _ in range(3))
    n_dom = rng.choice([1, 2, 3, 4])
    radius = random_radius(rng)
    max_r = max(
        radius.dir(x, y, z) for x in (-1, 0, 1) for y in (-1, 0, 1) for z in (-1, 0, 1)
    )
    size = tuple(max(s, max_r * n_dom * 2 + n_dom) for s in size)
    dd = sa.DistributedDomain(*size, backend="native")
    dd.set_radius(radius)
    dd.set_gpus([0] * n_dom)
    handles = []
    for qi in range(rng.randint(1, 3)):
        dtype = rng.choice([np.float32, np.float64])
        handles.append((dd.add_data(dtype, f"q{qi}"), dtype, 1.0 + qi))
    dd.realize()
    for h, dtype, scale in handles:
        for li in range(dd.num_local()):
            lo, hi = dd.local_rect(li)
            dd.write_global(li, lo, ripple_block(lo, hi, dd.size, scale).astype(dtype), h)
    dd.exchange()
    for h, dtype, scale in handles:
        check_valid_regions(dd, h, scale)


def test_jacobi_step_graph_matches_eager(monkeypatch):
    """the whole-step hipGraph path (auto-active for single-process
    single-domain) must be bitwise-identical to the eager path and match
    the torch reference"""
    from stencil_amd.models.jacobi3d import Jacobi3D

    size = (24, 18, 14)
    outs = {}
    for mode in ("graph", "eager"):
        monkeypatch.setenv("STENCIL_AMD_STEP_GRAPH", "1" if mode == "graph" else "0")
        app = Jacobi3D(size, backend="native", gpus=[0])
        app.realize()
        fill_interiors(app.dd, app.h)  # match _run_jacobi's initial state
        if mode == "graph":
            assert app._graph is not None, "graph path did not activate"
        else:
            assert app._graph is None
        for _ in range(4):
            app.step()
        lo, hi = app.dd.local_rect(0)
        outs[mode] = app.dd.read_global(0, lo, hi, app.h)
    np.testing.assert_array_equal(outs["graph"], outs["eager"])
    ref = _run_jacobi("torch", size, 4, 1)
    np.testing.assert_allclose(outs["graph"], ref[0][1], rtol=1e-6, atol=1e-6)


def test_rccl_wire_self_roundtrip():
    """native RcclWire sanity on one GPU: a world-1 communicator moving a
    device buffer to itself through grouped ncclSend/ncclRecv posted on
    the engine pack stream (the stream-ordering contract exchange_end
    relies on), plus the device barrier. Multi-rank matching is covered
    by the plan-parity CPU tests + the driver's multi-GPU run (RCCL
    refuses two ranks on one device, so N>1 cannot run here)."""
    dom = _C.LocalDomain(_C.Vec3(8, 8, 8), _C.Vec3(0, 0, 0), 0)
    dom.set_radius(_C.Radius.constant(0))
    dom.add_data(4, "q")
    dom.realize()
    eng = _C.ExchangeEngine([dom])
    nbytes = 4096
    src = eng.create_buffer(0, nbytes)
    dst = eng.create_buffer(0, nbytes)
    payload = bytes(range(256)) * (nbytes // 256)
    eng.buffer_from_host(src, payload)
    eng.buffer_from_host(dst, b"\0" * nbytes)

    uid = _C.RcclWire.unique_id()
    w = _C.RcclWire(0, 0, 1, uid)
    w.add_send(0, eng.buffer_ptr(src), nbytes, 0, 0)
    w.add_recv(0, eng.buffer_ptr(dst), nbytes, 0, 0)
    w.finalize()
    stream = eng.pack_stream_handle(0)
    w.post(0, stream)
    w.barrier(stream)
    eng.sync_packs()
    assert eng.buffer_to_host(dst) == payload


def test_rccl_wire_two_transfers_tag_order():
    """two self-transfers in one group must match by tag order on both
    sides (the pair_seq_tags contract): payloads land in the buffers the
    tags name, not swapped."""
    dom = _C.LocalDomain(_C.Vec3(8, 8, 8), _C.Vec3(0, 0, 0), 0)
    dom.set_radius(_C.Radius.constant(0))
    dom.add_data(4, "q")
    dom.realize()
    eng = _C.ExchangeEngine([dom])
    n = 1024
    bufs = [eng.create_buffer(0, n) for _ in range(4)]  # s0 s1 d0 d1
    pay = [bytes([i]) * n for i in (1, 2)]
    eng.buffer_from_host(bufs[0], pay[0])
    eng.buffer_from_host(bufs[1], pay[1])
    uid = _C.RcclWire.unique_id()
    w = _C.RcclWire(0, 0, 1, uid)
    # register recvs before sends and with shuffled tag order: finalize()
    # must sort both sides into the same pairing
    w.add_recv(0, eng.buffer_ptr(bufs[3]), n, 0, 1)
    w.add_recv(0, eng.buffer_ptr(bufs[2]), n, 0, 0)
    w.add_send(0, eng.buffer_ptr(bufs[1]), n, 0, 1)
    w.add_send(0, eng.buffer_ptr(bufs[0]), n, 0, 0)
    w.finalize()
    w.post(0, eng.pack_stream_handle(0))
    eng.sync_packs()
    assert eng.buffer_to_host(bufs[2]) == pay[0]
    assert eng.buffer_to_host(bufs[3]) == pay[1]


@pytest.mark.parametrize("r", [1, 2])
def test_staged_local_translates(r, monkeypatch):
    """STENCIL_AMD_STAGE_LOCAL=all forces every same-process translate
    through the staged pack -> dst-device buffer -> unpack path (the
    route thin cross-GPU faces take on a real multi-GPU node); results
    must match the direct-write path bitwise."""
    monkeypatch.setenv("STENCIL_AMD_STAGE_LOCAL", "all")
    dd = make_dd((12, 10, 8), r, 2)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    assert any(dd.backend._staged_local), "staged-local path did not engage"
    fill_interiors(dd, h)
    dd.exchange()
    check_full_regions(dd, h)
    dd.swap()
    fill_interiors(dd, h, scale=3.0)
    dd.exchange()
    check_full_regions(dd, h, scale=3.0)


def test_read_global_out_buffer_reuse():
    """read_global(out=...) fills the caller's array in place (checkpoint
    loops skip fresh-page allocation); contents match a fresh read."""
    dd = make_dd((16, 12, 10), 1, 1)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h, scale=5.0)
    lo, hi = dd.local_rect(0)
    fresh = dd.read_global(0, lo, hi, h)
    out = np.zeros_like(fresh)
    got = dd.read_global(0, lo, hi, h, out=out)
    assert got is out
    np.testing.assert_array_equal(out, fresh)


def test_domain_lifecycle_no_leak():
    """create/exchange/destroy five DistributedDomains in one process:
    destructors (domains, engine buffers, streams, IPC-free path) must
    return the VRAM -- guards the teardown paths the long-running apps
    never exercise"""
    import gc

    def one_round():
        dd = make_dd((64, 48, 32), 2, 2)
        h = dd.add_data(np.float32, "q")
        dd.realize()
        fill_interiors(dd, h)
        dd.exchange()
        check_full_regions(dd, h)
        dd.swap()

    one_round()  # warm pools (pinned bounce etc.)
    gc.collect()
    free0, total = _C.device_mem_info(0)
    for _ in range(5):
        one_round()
        gc.collect()
    free1, _ = _C.device_mem_info(0)
    leaked = free0 - free1
    assert leaked < 64 << 20, f"leaked {leaked/2**20:.1f} MiB over 5 domain lifecycles"
