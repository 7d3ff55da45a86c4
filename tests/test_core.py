"""Foundation-layer tests (no GPU): Vec3, Radius, prime factors, partition
exactness on uneven splits, QAP. Mirrors the reference's test_cpu tier
(test/test_cpu_partition.cpp, test_cpu_qap.cpp)."""
import pytest

from stencil_amd import _C


def V(x, y, z):
    return _C.Vec3(x, y, z)


class TestVec3:
    def test_ops(self):
        a, b = V(1, 2, 3), V(4, 5, 6)
        assert (a + b).tuple() == (5, 7, 9)
        assert (b - a).tuple() == (3, 3, 3)
        assert (a * b).tuple() == (4, 10, 18)
        assert a != b and a == V(1, 2, 3)

    def test_neq_all_components(self):
        # the reference's Dim3 operator!= only compared z (dim3.hpp:195)
        assert V(0, 0, 1) != V(1, 0, 1)
        assert V(0, 1, 0) != V(0, 0, 0)

    def test_wrap(self):
        assert V(-1, 10, 3).wrap(V(10, 10, 10)).tuple() == (9, 0, 3)
        assert V(-11, 21, -20).wrap(V(10, 10, 10)).tuple() == (9, 1, 0)


class TestRadius:
    def test_constant(self):
        r = _C.Radius.constant(2)
        assert r.dir(1, 0, 0) == 2 and r.dir(1, 1, 1) == 2 and r.dir(0, 0, 0) == 0

    def test_face_edge_corner(self):
        r = _C.Radius.face_edge_corner(3, 2, 1)
        assert r.dir(1, 0, 0) == 3 and r.dir(0, -1, 0) == 3
        assert r.dir(1, 1, 0) == 2 and r.dir(0, -1, 1) == 2
        assert r.dir(1, -1, 1) == 1
        assert r.dir(0, 0, 0) == 0

    def test_asymmetric(self):
        r = _C.Radius.constant(1)
        r.set_dir(1, 0, 0, 2)
        assert r.x(1) == 2 and r.x(-1) == 1


def test_prime_factors():
    assert _C.prime_factors(12) == [3, 2, 2]
    assert _C.prime_factors(7) == [7]
    assert _C.prime_factors(1) == []
    assert _C.prime_factors(8) == [2, 2, 2]


class TestRankPartition:
    def test_even(self):
        p = _C.RankPartition(V(100, 100, 100), 8)
        assert p.dim().tuple() == (2, 2, 2)
        for i in range(8):
            idx = p.dimensionize(i)
            assert p.subdomain_size(idx).tuple() == (50, 50, 50)
            assert p.linearize(idx) == i

    def test_uneven_covers_exactly(self):
        # sizes and origins must tile the global grid exactly
        for size, n in [((10, 10, 10), 4), ((7, 5, 3), 6), ((100, 1, 1), 3), ((13, 17, 19), 8)]:
            p = _C.RankPartition(V(*size), n)
            d = p.dim().tuple()
            assert d[0] * d[1] * d[2] == n
            total = 0
            for z in range(d[2]):
                for y in range(d[1]):
                    for x in range(d[0]):
                        idx = V(x, y, z)
                        s = p.subdomain_size(idx).tuple()
                        o = p.subdomain_origin(idx).tuple()
                        assert all(c >= 1 for c in s), (size, n, s)
                        total += s[0] * s[1] * s[2]
                        # origin = sum of sizes of preceding subdomains per axis
                        for ax in range(3):
                            acc = 0
                            for k in range((x, y, z)[ax]):
                                kk = [x, y, z]
                                kk[ax] = k
                                acc += p.subdomain_size(V(*kk)).tuple()[ax]
                            assert o[ax] == acc, (size, n, (x, y, z), ax)
            assert total == size[0] * size[1] * size[2]


class TestNodePartition:
    def test_two_level(self):
        r = _C.Radius.constant(1)
        p = _C.NodePartition(V(512, 512, 512), r, 2, 4)
        sd, nd = p.sys_dim().tuple(), p.node_dim().tuple()
        assert sd[0] * sd[1] * sd[2] == 2
        assert nd[0] * nd[1] * nd[2] == 4
        d = p.dim().tuple()
        assert d == tuple(sd[i] * nd[i] for i in range(3))

    def test_radius_weighted_split(self):
        # huge +-x radius makes x-splits expensive: expect no x split
        r = _C.Radius.constant(1)
        r.set_dir(1, 0, 0, 50)
        r.set_dir(-1, 0, 0, 50)
        p = _C.NodePartition(V(64, 64, 64), r, 1, 4)
        assert p.dim().tuple()[0] == 1

    def test_uneven_tiles(self):
        r = _C.Radius.constant(2)
        p = _C.NodePartition(V(100, 90, 80), r, 1, 8)
        d = p.dim().tuple()
        total = 0
        for z in range(d[2]):
            for y in range(d[1]):
                for x in range(d[0]):
                    total += _mul(p.subdomain_size(V(x, y, z)).tuple())
        assert total == 100 * 90 * 80


def _mul(t):
    return t[0] * t[1] * t[2]


class TestQap:
    def _mat(self, vals):
        n = len(vals)
        m = _C.SqMat(n)
        for i in range(n):
            for j in range(n):
                m.set(i, j, vals[i][j])
        return m

    def test_identity_optimal(self):
        # w and d already aligned: identity must be among the optima
        w = self._mat([[0, 10, 0], [10, 0, 1], [0, 1, 0]])
        d = self._mat([[0, 1, 5], [1, 0, 5], [5, 5, 0]])
        f = _C.qap_solve(w, d)
        assert _C.qap_cost(w, d, f) <= _C.qap_cost(w, d, [0, 1, 2])

    def test_finds_better_than_worst(self):
        # heavy comm pair (0,1) should land on the fast link pair
        w = self._mat([[0, 100, 0, 0], [100, 0, 0, 0], [0, 0, 0, 1], [0, 0, 1, 0]])
        d = self._mat(
            [[0, 9, 9, 9], [9, 0, 9, 9], [9, 9, 0, 1], [9, 9, 1, 0]]
        )
        f = _C.qap_solve(w, d)
        # 0 and 1 must map onto {2,3} (the distance-1 pair)
        assert {f[0], f[1]} == {2, 3}
        assert _C.qap_cost(w, d, f) == 100 * 1 * 2 + 1 * 9 * 2

    def test_exhaustive_matches_bruteforce(self):
        import itertools

        w = self._mat([[0, 3, 1], [2, 0, 4], [1, 1, 0]])
        d = self._mat([[0, 2, 7], [3, 0, 1], [6, 2, 0]])
        best = min(
            (_C.qap_cost(w, d, list(p)) for p in itertools.permutations(range(3))),
        )
        f = _C.qap_solve(w, d)
        assert _C.qap_cost(w, d, f) == pytest.approx(best)


def test_data_handle_preserves_dtype():
    """add_data keeps the numpy dtype (not just element size), so
    read_global on int32/float16 quantities does not silently
    reinterpret bytes (round-1 advisor finding)"""
    import numpy as np

    from stencil_amd import DistributedDomain

    dd = DistributedDomain(8, 8, 8, backend="torch")
    hi32 = dd.add_data(np.int32, "ids")
    hf32 = dd.add_data(np.float32, "temp")
    hf16 = dd.add_data(np.float16, "half")
    assert hi32.dtype == np.dtype(np.int32) and hi32.elem_size == 4
    assert hf32.dtype == np.dtype(np.float32)
    assert hf16.dtype == np.dtype(np.float16) and hf16.elem_size == 2
    assert dd.data_handle(0).dtype == np.dtype(np.int32)
    dd.set_radius(1)
    dd.set_gpus([0])
    dd.realize()
    lo, hi = dd.local_rect(0)
    arr = np.arange(8 * 8 * 8, dtype=np.int32).reshape(8, 8, 8)
    dd.write_global(0, lo, arr, hi32)
    back = dd.read_global(0, lo, hi, hi32)
    assert back.dtype == np.int32
    np.testing.assert_array_equal(back, arr)


def test_add_data_rejects_bad_size():
    import pytest

    from stencil_amd import DistributedDomain

    dd = DistributedDomain(8, 8, 8, backend="torch")
    with pytest.raises(ValueError):
        dd.add_data(3, "bad")


def test_machine_model_single_rank():
    """Machine.build: global GPU list with owning ranks (reference
    src/machine.cpp:19-129); on a no-GPU box the stub entries still give
    a consistent inventory and self/colocated/remote classification."""
    from stencil_amd.parallel.comm import Comm
    from stencil_amd.parallel.machine import Machine

    comm = Comm()
    m = Machine.build(comm, [0, 1])
    assert m.num_nodes() == 1
    assert len(m.gpus) == 2
    assert m.gpus[0].ranks == [0] and m.gpus[0].cuda_of_rank[0] == 0
    assert m.gpus_of_rank(0) == m.gpus
    assert m.classify(0, 0) == "self"
