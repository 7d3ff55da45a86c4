"""Full-region ripple exchange verification on the torch reference backend
(CPU, single process). Covers self-wrap, multi-subdomain, symmetric radii
1 and 2, asymmetric radius, multiple quantities, and swap semantics."""
import numpy as np
import pytest

import stencil_amd as sa
from stencil_amd import _C

from util import check_full_regions, fill_interiors


def make_dd(size, radius, n_domains=1):
    dd = sa.DistributedDomain(*size, backend="torch")
    dd.set_radius(radius)
    dd.set_gpus([0] * n_domains)
    return dd


@pytest.mark.parametrize("r", [1, 2])
@pytest.mark.parametrize("n_domains", [1, 2, 4, 8])
def test_ripple_exchange(r, n_domains):
    dd = make_dd((12, 10, 8), r, n_domains)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h)
    dd.exchange()
    check_full_regions(dd, h)


def test_ripple_asymmetric():
    r = _C.Radius.constant(1)
    r.set_dir(1, 0, 0, 2)
    dd = make_dd((12, 10, 8), 0, 2)
    dd.set_radius(r)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h)
    dd.exchange()
    check_full_regions(dd, h)


def test_multiple_quantities_exchange():
    dd = make_dd((10, 10, 10), 1, 2)
    h1 = dd.add_data(np.float32, "a")
    h2 = dd.add_data(np.float32, "b")
    dd.realize()
    fill_interiors(dd, h1, scale=1.0)
    fill_interiors(dd, h2, scale=2.0)
    dd.exchange()
    check_full_regions(dd, h1, scale=1.0)
    check_full_regions(dd, h2, scale=2.0)


def test_exchange_after_swap():
    """halos must follow the curr buffer across swap()"""
    dd = make_dd((8, 8, 8), 1, 2)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h, scale=3.0)
    dd.swap()  # ripple data now in next
    fill_interiors(dd, h, scale=5.0)  # new data in curr
    dd.exchange()
    check_full_regions(dd, h, scale=5.0)
    dd.swap()
    # old interiors intact in (now) curr
    for li in range(dd.num_local()):
        lo, hi = dd.local_rect(li)
        got = dd.read_global(li, lo, hi, h)
        from util import ripple_block

        assert np.array_equal(got, ripple_block(lo, hi, dd.size, 3.0))


def test_interior_exterior_cover():
    dd = make_dd((16, 12, 10), 2, 2)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    interiors = dd.get_interior()
    exteriors = dd.get_exterior()
    for li in range(dd.num_local()):
        lo, hi = dd.local_rect(li)
        vol = (hi[0] - lo[0]) * (hi[1] - lo[1]) * (hi[2] - lo[2])
        (ilo, ihi) = interiors[li]
        ivol = max(0, ihi[0] - ilo[0]) * max(0, ihi[1] - ilo[1]) * max(0, ihi[2] - ilo[2])
        evol = sum(
            (b[1][0] - b[0][0]) * (b[1][1] - b[0][1]) * (b[1][2] - b[0][2])
            for b in exteriors[li]
        )
        assert ivol + evol == vol
        # exterior boxes must not overlap the interior or each other
        cells = np.zeros((hi[2] - lo[2], hi[1] - lo[1], hi[0] - lo[0]), dtype=np.int32)
        cells[
            ilo[2] - lo[2] : ihi[2] - lo[2],
            ilo[1] - lo[1] : ihi[1] - lo[1],
            ilo[0] - lo[0] : ihi[0] - lo[0],
        ] += 1
        for b in exteriors[li]:
            cells[
                b[0][2] - lo[2] : b[1][2] - lo[2],
                b[0][1] - lo[1] : b[1][1] - lo[1],
                b[0][0] - lo[0] : b[1][0] - lo[0],
            ] += 1
        assert (cells == 1).all()


def test_paraview_dump(tmp_path):
    dd = make_dd((6, 5, 4), 1, 1)
    h = dd.add_data(np.float32, "temp")
    dd.realize()
    fill_interiors(dd, h)
    dd.write_paraview(str(tmp_path / "pv"))
    files = list(tmp_path.glob("pv*.txt"))
    assert len(files) == 1
    lines = files[0].read_text().strip().splitlines()
    assert lines[0] == "Z,Y,X,temp"
    assert len(lines) == 1 + 6 * 5 * 4


def test_checkpoint_roundtrip(tmp_path):
    from util import ripple_block

    dd = make_dd((8, 8, 8), 1, 2)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h, scale=7.0)
    dd.save_checkpoint(str(tmp_path / "ck"))

    dd2 = make_dd((8, 8, 8), 1, 2)
    h2 = dd2.add_data(np.float32, "q")
    dd2.realize()
    dd2.load_checkpoint(str(tmp_path / "ck"))
    for li in range(dd2.num_local()):
        lo, hi = dd2.local_rect(li)
        assert np.array_equal(dd2.read_global(li, lo, hi, h2), ripple_block(lo, hi, dd2.size, 7.0))


def test_methods_restriction():
    import stencil_amd as sa

    dd = make_dd((8, 8, 8), 1, 2)
    dd.set_methods(sa.Method.DIRECT_KERNEL)  # same-rank only: fine
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h)
    dd.exchange()
    check_full_regions(dd, h)


def test_exchange_begin_end():
    dd = make_dd((8, 8, 8), 1, 2)
    h = dd.add_data(np.float32, "q")
    dd.realize()
    fill_interiors(dd, h)
    dd.exchange_begin()
    dd.exchange_end()
    check_full_regions(dd, h)


@pytest.mark.parametrize("m", [2, 3])
def test_halo_multiplier_matches_plain(m):
    """temporal blocking (exchange every m-th step with m-deep halos) must
    reproduce the every-step-exchange results exactly"""
    from stencil_amd.models.jacobi3d import Jacobi3D

    size = (18, 15, 12)
    outs = []
    for mult in (1, m):
        app = Jacobi3D(size, backend="torch", gpus=[0, 0], halo_multiplier=mult)
        app.realize()
        fill_interiors(app.dd, app.h)
        for _ in range(2 * m):
            app.step()
        outs.append(
            [
                app.dd.read_global(li, *app.dd.local_rect(li), app.h)
                for li in range(app.dd.num_local())
            ]
        )
    for a, b in zip(outs[0], outs[1]):
        np.testing.assert_array_equal(a, b)


def test_halo_multiplier_with_radius2():
    from stencil_amd.models.jacobi3d import Jacobi3D

    size = (20, 16, 12)
    outs = []
    for mult in (1, 2):
        app = Jacobi3D(size, backend="torch", gpus=[0, 0], radius=2, halo_multiplier=mult)
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


def test_exchange_groups_selective():
    """exchange(group) must move only that group's quantities"""
    import stencil_amd as sa

    dd = make_dd((10, 8, 8), 1, 2)
    dd.set_exchange_groups([[0], [1]])
    ha = dd.add_data(np.float32, "a")
    hb = dd.add_data(np.float32, "b")
    dd.realize()
    fill_interiors(dd, ha, scale=1.0)
    fill_interiors(dd, hb, scale=2.0)
    dd.exchange(group=0)
    check_full_regions(dd, ha, scale=1.0)  # a's halos are fresh
    # b's halos are still zero (only the interior was written)
    from util import full_region_of

    lo, hi = dd.local_rect(0)
    flo, fhi = full_region_of(dd, 0)
    got = dd.read_global(0, flo, fhi, hb)
    assert got[0, 0, 0] == 0.0  # a corner halo cell untouched
    dd.exchange(group=1)
    check_full_regions(dd, hb, scale=2.0)


def test_plan_files_and_timers(tmp_path):
    """Aux-subsystem parity (reference SETUP_STATS/EXCHANGE_STATS +
    --plan-file machinery, src/stencil.cu realize): plan/matrix files are
    written under the output prefix and the always-on timers accumulate."""
    dd = make_dd((12, 10, 8), 1, 2)
    dd.set_output_prefix(str(tmp_path / "run_"))
    h = dd.add_data(np.float32, "q")
    dd.realize()
    plans = list(tmp_path.glob("run_plan_*.txt"))
    assert len(plans) == 1, plans
    txt = plans[0].read_text()
    assert "q" in txt or "quantit" in txt or len(txt) > 0
    assert "direct_kernel" in txt and "bytes=" in txt
    # rank x rank matrix: single process -> 1x1 (all traffic is direct)
    mats = list(tmp_path.glob("run_mat*.txt"))
    assert len(mats) == 1
    mat = np.atleast_2d(np.loadtxt(str(mats[0])))
    assert mat.shape == (1, 1)

    for k in ("topo", "placement", "realize", "plan", "create"):
        assert k in dd.setup_times and dd.setup_times[k] >= 0.0

    fill_interiors(dd, h)
    assert dd.time_exchange == 0.0
    dd.exchange()
    dd.swap()
    assert dd.time_exchange > 0.0
    assert dd.time_swap > 0.0


def test_exchange_groups_validation_and_omission():
    import stencil_amd as sa

    dd = make_dd((10, 8, 8), 1, 1)
    with pytest.raises(ValueError):
        dd.set_exchange_groups([[0, 1], [1]])  # overlapping

    # a quantity left out of every group never moves; grouped ones do
    dd2 = make_dd((10, 8, 8), 1, 1)
    ha = dd2.add_data(np.float32, "a")
    hb = dd2.add_data(np.float32, "b")
    dd2.set_exchange_groups([[ha.index]])
    dd2.realize()
    fill_interiors(dd2, ha)
    fill_interiors(dd2, hb)
    dd2.exchange(group=0)
    from util import check_valid_regions

    check_valid_regions(dd2, ha)
    lo, hi = dd2.local_rect(0)
    flo, fhi, full_b = (
        tuple(c - 1 for c in lo),
        tuple(c + 1 for c in hi),
        None,
    )
    full_b = dd2.read_global(0, flo, fhi, hb)
    assert (full_b[0, :, :] == 0).all()  # b's halo untouched (zero-init)
