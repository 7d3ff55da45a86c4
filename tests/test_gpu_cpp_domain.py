"""The pure-C++ orchestrator (distributed.hpp) under pytest: the bound
CppDistributedDomain must reproduce the Python path's exchange semantics
cell-for-cell (same planner, same engine). Complements examples/*.cpp
(the no-Python proof) with an assertable ripple check."""
import numpy as np
import pytest

from stencil_amd import _C

pytestmark = pytest.mark.gpu


def ripple_value(x, y, z, size, scale=1.0):
    ripple = [0.0, 1.0, 2.0, 1.0]
    return scale * (
        (x % size[0]) + ripple[x % 4] + 2 * ((y % size[1]) + ripple[y % 4])
        + 4 * ((z % size[2]) + ripple[z % 4])
    )


@pytest.mark.parametrize("r,n_domains", [(1, 1), (2, 2), (1, 4)])
def test_cpp_domain_ripple_exchange(r, n_domains):
    size = (12, 10, 8)
    dd = _C.CppDistributedDomain(*size)
    dd.set_radius(_C.Radius.constant(r))
    qi = dd.add_data(4, "q")
    dd.set_gpus([0] * n_domains)  # same-GPU fake-multi-GPU
    dd.realize()
    assert dd.num_local() == n_domains
    assert dd.bytes_translate() > 0

    # fill every interior with the deterministic ripple
    for li in range(dd.num_local()):
        rect = dd.local_rect(li)
        lo = rect.lo.tuple()
        hi = rect.hi.tuple()
        ext = tuple(hi[i] - lo[i] for i in range(3))
        arr = np.empty((ext[2], ext[1], ext[0]), dtype=np.float32)
        for z in range(ext[2]):
            for y in range(ext[1]):
                for x in range(ext[0]):
                    arr[z, y, x] = ripple_value(lo[0] + x, lo[1] + y, lo[2] + z, size)
        dom = dd.domain(li)
        # interior position in allocation coords = halo offset
        pos = _C.Vec3(r, r, r)
        dom.region_from_host(arr.tobytes(), pos, _C.Vec3(*ext), qi)

    dd.exchange()

    # verify the FULL region (incl. halos) of every domain against the
    # periodic-wrapped analytic value
    for li in range(dd.num_local()):
        rect = dd.local_rect(li)
        lo = rect.lo.tuple()
        hi = rect.hi.tuple()
        fext = tuple(hi[i] - lo[i] + 2 * r for i in range(3))
        dom = dd.domain(li)
        raw = dom.region_to_host(_C.Vec3(0, 0, 0), _C.Vec3(*fext), qi)
        got = np.frombuffer(raw, dtype=np.float32).reshape(fext[2], fext[1], fext[0])
        for z in range(fext[2]):
            for y in range(fext[1]):
                for x in range(fext[0]):
                    gx, gy, gz = lo[0] + x - r, lo[1] + y - r, lo[2] + z - r
                    want = ripple_value(gx % size[0], gy % size[1], gz % size[2], size)
                    assert got[z, y, x] == want, (li, (x, y, z), got[z, y, x], want)
    # swap + second exchange exercises buffer alternation
    dd.swap()
    assert "plan" in dd.setup_times()


def test_cpp_domain_exchange_groups():
    """quantity-group exchanges through the C++ orchestrator: group 0
    moves q0 only; q1's halos stay stale until group 1 is exchanged"""
    size = (10, 8, 8)
    r = 1
    dd = _C.CppDistributedDomain(*size)
    dd.set_radius(_C.Radius.constant(r))
    q0 = dd.add_data(4, "a")
    q1 = dd.add_data(4, "b")
    dd.set_exchange_groups([[0], [1]])
    dd.set_gpus([0, 0])
    dd.realize()

    for li in range(dd.num_local()):
        rect = dd.local_rect(li)
        lo, hi = rect.lo.tuple(), rect.hi.tuple()
        ext = tuple(hi[i] - lo[i] for i in range(3))
        for qi, scale in ((q0, 1.0), (q1, 2.0)):
            arr = np.zeros((ext[2], ext[1], ext[0]), dtype=np.float32)
            for z in range(ext[2]):
                for y in range(ext[1]):
                    for x in range(ext[0]):
                        arr[z, y, x] = scale * ripple_value(lo[0] + x, lo[1] + y, lo[2] + z, size)
            dd.domain(li).region_from_host(arr.tobytes(), _C.Vec3(r, r, r), _C.Vec3(*ext), qi)

    dd.exchange(0)  # q0 halos only

    def halo_filled(li, qi, scale):
        rect = dd.local_rect(li)
        lo, hi = rect.lo.tuple(), rect.hi.tuple()
        fext = tuple(hi[i] - lo[i] + 2 * r for i in range(3))
        raw = dd.domain(li).region_to_host(_C.Vec3(0, 0, 0), _C.Vec3(*fext), qi)
        got = np.frombuffer(raw, dtype=np.float32).reshape(fext[2], fext[1], fext[0])
        # check one -x halo plane cell
        gx, gy, gz = lo[0] - 1, lo[1], lo[2]
        want = scale * ripple_value(gx % size[0], gy % size[1], gz % size[2], size)
        return got[r, r, 0] == want

    assert halo_filled(0, q0, 1.0)
    assert not halo_filled(0, q1, 2.0)  # group 1 not exchanged yet (halo is 0)
    dd.exchange(1)
    assert halo_filled(0, q1, 2.0)
