"""Halo geometry tests (no GPU): halo_pos/halo_extent/halo_coords for
faces/edges/corners with symmetric and asymmetric radii. Mirrors the
reference's test_cuda_local_domain.cu cases (which only need geometry, so
they run CPU-side here)."""
from stencil_amd import _C


def V(x, y, z):
    return _C.Vec3(x, y, z)


SZ = V(30, 40, 50)


class TestSymmetric:
    r = _C.Radius.constant(2)

    def ext(self, d):
        return _C.halo_extent(V(*d), SZ, self.r).tuple()

    def pos(self, d, halo):
        return _C.halo_pos(V(*d), SZ, self.r, halo).tuple()

    def test_extents(self):
        assert self.ext((1, 0, 0)) == (2, 40, 50)
        assert self.ext((-1, 0, 0)) == (2, 40, 50)
        assert self.ext((0, 1, 0)) == (30, 2, 50)
        assert self.ext((1, 1, 0)) == (2, 2, 50)
        assert self.ext((1, -1, 1)) == (2, 2, 2)
        assert self.ext((0, 0, 0)) == (30, 40, 50)

    def test_halo_positions(self):
        # -x halo starts at allocation x=0; +x halo just past the interior
        assert self.pos((-1, 0, 0), True) == (0, 2, 2)
        assert self.pos((1, 0, 0), True) == (32, 2, 2)
        assert self.pos((0, 0, 0), True) == (2, 2, 2)

    def test_interior_positions(self):
        # the interior cells adjacent to each boundary (what a send packs)
        assert self.pos((-1, 0, 0), False) == (2, 2, 2)
        assert self.pos((1, 0, 0), False) == (30, 2, 2)
        assert self.pos((1, 1, 1), False) == (30, 40, 50)

    def test_pack_region_is_last_interior_cells(self):
        # dir=+x sends halo_extent(-x).x = 2 cells starting at alloc x=30:
        # interior spans [2, 32) so the last 2 interior cells are [30, 32)
        p = self.pos((1, 0, 0), False)
        e = _C.halo_extent(V(-1, 0, 0), SZ, self.r).tuple()
        assert p[0] + e[0] == 2 + 30  # == interior end


class TestAsymmetric:
    """+x radius 2, -x radius 1 (the reference's uncentered-kernel case)"""

    def setup_method(self):
        self.r = _C.Radius.constant(1)
        self.r.set_dir(1, 0, 0, 2)

    def test_raw_and_positions(self):
        d = _C.LocalDomain(SZ, V(0, 0, 0), 0)
        d.set_radius(self.r)
        assert d.raw_size().tuple() == (33, 42, 52)
        # -x halo is 1 deep, +x halo is 2 deep
        assert d.halo_extent(V(-1, 0, 0)).tuple() == (1, 40, 50)
        assert d.halo_extent(V(1, 0, 0)).tuple() == (2, 40, 50)
        # +x halo starts after -x halo (1) + interior (30)
        assert d.halo_pos(V(1, 0, 0), True).tuple() == (31, 1, 1)
        # a +x send packs halo_extent(-x)=1 cell starting at alloc x=30
        assert d.halo_pos(V(1, 0, 0), False).tuple() == (30, 1, 1)
        # a -x send packs halo_extent(+x)=2 cells starting at interior begin
        assert d.halo_pos(V(-1, 0, 0), False).tuple() == (1, 1, 1)

    def test_halo_coords_global(self):
        d = _C.LocalDomain(SZ, V(100, 0, 0), 0)
        d.set_radius(self.r)
        hc = d.halo_coords(V(-1, 0, 0), True)
        assert hc.lo.tuple() == (99, 0, 0)
        assert hc.hi.tuple() == (100, 40, 50)
        hc = d.halo_coords(V(1, 0, 0), True)
        assert hc.lo.tuple() == (130, 0, 0)
        assert hc.hi.tuple() == (132, 40, 50)


class TestFullAndCompute:
    def test_regions(self):
        r = _C.Radius.constant(3)
        d = _C.LocalDomain(V(10, 10, 10), V(20, 30, 40), 0)
        d.set_radius(r)
        cr = d.compute_region()
        assert cr.lo.tuple() == (20, 30, 40) and cr.hi.tuple() == (30, 40, 50)
        fr = d.full_region()
        assert fr.lo.tuple() == (17, 27, 37) and fr.hi.tuple() == (33, 43, 53)
        d.add_data(4, "q")
        assert d.halo_bytes(V(1, 0, 0), 0) == 4 * 3 * 10 * 10
