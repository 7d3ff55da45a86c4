"""Shared test helpers: the ripple-field full-region exchange verification
(modeled on the reference's flagship test, test/test_cuda_mpi_exchange.cu:
13-33, 126-190: deterministic global field, exchange, then verify EVERY
cell of the full region including halos against the periodic-wrapped
analytic value)."""
import numpy as np


def ripple(cx, cy, cz, size):
    """deterministic field value at global (possibly out-of-range) coords,
    periodic-wrapped into the global compute region"""
    x = np.mod(cx, size[0])
    y = np.mod(cy, size[1])
    z = np.mod(cz, size[2])
    return ((x % 7 + 1) * 100 + (y % 5 + 1) * 10 + (z % 3 + 1)).astype(np.float32)


def ripple_block(lo, hi, size, scale=1.0):
    """ripple over the box [lo, hi) as a (z, y, x) float32 array"""
    zz, yy, xx = np.meshgrid(
        np.arange(lo[2], hi[2]), np.arange(lo[1], hi[1]), np.arange(lo[0], hi[0]), indexing="ij"
    )
    return (ripple(xx, yy, zz, size) * scale).astype(np.float32)


def fill_interiors(dd, handle, scale=1.0):
    for li in range(dd.num_local()):
        lo, hi = dd.local_rect(li)
        block = ripple_block(lo, hi, dd.size, scale).astype(handle.dtype)
        dd.write_global(li, lo, block, handle)


def full_region_of(dd, li):
    lo, hi = dd.local_rect(li)
    r = dd.radius
    flo = (lo[0] - r.x(-1), lo[1] - r.y(-1), lo[2] - r.z(-1))
    fhi = (hi[0] + r.x(1), hi[1] + r.y(1), hi[2] + r.z(1))
    return flo, fhi


def check_valid_regions(dd, handle, scale=1.0):
    """verify the interior plus every halo side whose direction has a
    nonzero radius (sides with radius 0 are never exchanged and hold
    undefined data -- legal for sparse per-direction radii)"""
    import itertools

    r = dd.radius
    for li in range(dd.num_local()):
        lo, hi = dd.local_rect(li)
        regions = [(lo, hi)]
        for d in itertools.product((-1, 0, 1), repeat=3):
            if d == (0, 0, 0) or r.dir(*d) == 0:
                continue
            ext = [0, 0, 0]
            glo = [0, 0, 0]
            for i in range(3):
                face = (r.x, r.y, r.z)[i]
                if d[i] == 0:
                    glo[i], ext[i] = lo[i], hi[i] - lo[i]
                elif d[i] < 0:
                    ext[i] = face(-1)
                    glo[i] = lo[i] - ext[i]
                else:
                    ext[i] = face(1)
                    glo[i] = hi[i]
            if ext[0] * ext[1] * ext[2] == 0:
                continue
            regions.append((tuple(glo), tuple(glo[i] + ext[i] for i in range(3))))
        for (glo, ghi) in regions:
            got = dd.read_global(li, glo, ghi, handle)
            want = ripple_block(glo, ghi, dd.size, scale).astype(got.dtype)
            if not np.array_equal(got, want):
                bad = np.argwhere(got != want)
                i = tuple(bad[0])
                raise AssertionError(
                    f"domain {li} region {glo}..{ghi}: {len(bad)} mismatches; first {i} "
                    f"got {got[i]} want {want[i]}"
                )


def check_full_regions(dd, handle, scale=1.0):
    """verify every cell of every local domain's full region (incl. halos)"""
    for li in range(dd.num_local()):
        flo, fhi = full_region_of(dd, li)
        got = dd.read_global(li, flo, fhi, handle)
        want = ripple_block(flo, fhi, dd.size, scale)
        if not np.array_equal(got, want):
            bad = np.argwhere(got != want)
            i = tuple(bad[0])
            raise AssertionError(
                f"domain {li}: {len(bad)} mismatches; first at (z,y,x)={i} "
                f"got {got[i]} want {want[i]} (full region {flo}..{fhi})"
            )
