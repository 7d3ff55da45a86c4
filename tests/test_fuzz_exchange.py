"""Randomized exchange configurations (CPU, torch backend): random grid
sizes, per-direction radii, quantity sets, and subdomain counts, each
verified with the full-region ripple check. Catches convention bugs the
hand-picked cases miss (the reference's test matrix was fixed)."""
import random

import numpy as np
import pytest

import stencil_amd as sa
from stencil_amd import _C

from util import check_valid_regions, fill_interiors, ripple_block


def random_radius(rng):
    r = _C.Radius.constant(rng.choice([0, 1, 2]))
    # sprinkle asymmetric per-direction radii
    for _ in range(rng.randint(0, 6)):
        d = (rng.randint(-1, 1), rng.randint(-1, 1), rng.randint(-1, 1))
        if d == (0, 0, 0):
            continue
        r.set_dir(*d, rng.choice([0, 1, 2, 3]))
    # ensure at least one direction is active
    if all(r.dir(x, y, z) == 0 for x in (-1, 0, 1) for y in (-1, 0, 1) for z in (-1, 0, 1)):
        r.set_dir(1, 0, 0, 1)
    return r


@pytest.mark.parametrize("seed", range(12))
def test_fuzz_ripple(seed):
    rng = random.Random(seed)
    size = tuple(rng.randint(6, 24) for _ in range(3))
    n_dom = rng.choice([1, 2, 3, 4])
    radius = random_radius(rng)
    # subdomains must be at least as large as the deepest radius they
    # exchange; keep sizes generous relative to radii
    max_r = max(
        radius.dir(x, y, z) for x in (-1, 0, 1) for y in (-1, 0, 1) for z in (-1, 0, 1)
    )
    size = tuple(max(s, max_r * n_dom * 2 + n_dom) for s in size)

    dd = sa.DistributedDomain(*size, backend="torch")
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


@pytest.mark.parametrize("seed", range(6))
def test_fuzz_exchange_groups(seed):
    """random quantity partitions into 2 exchange groups; exchanging both
    groups must fill every fillable halo of every quantity"""
    rng = random.Random(1000 + seed)
    size = tuple(rng.randint(8, 20) for _ in range(3))
    n_dom = rng.choice([1, 2, 3])
    radius = random_radius(rng)
    max_r = max(
        radius.dir(x, y, z) for x in (-1, 0, 1) for y in (-1, 0, 1) for z in (-1, 0, 1)
    )
    size = tuple(max(s, max_r * n_dom * 2 + n_dom) for s in size)
    nq = rng.randint(2, 4)
    qs = list(range(nq))
    rng.shuffle(qs)
    cut = rng.randint(1, nq - 1)
    groups = [sorted(qs[:cut]), sorted(qs[cut:])]

    dd = sa.DistributedDomain(*size, backend="torch")
    dd.set_radius(radius)
    dd.set_gpus([0] * n_dom)
    dd.set_exchange_groups(groups)
    handles = [(dd.add_data(np.float32, f"q{qi}"), 1.0 + qi) for qi in range(nq)]
    dd.realize()
    for h, scale in handles:
        for li in range(dd.num_local()):
            lo, hi = dd.local_rect(li)
            dd.write_global(li, lo, ripple_block(lo, hi, dd.size, scale), h)
    dd.exchange(group=0)
    dd.exchange(group=1)
    for h, scale in handles:
        check_valid_regions(dd, h, scale)


@pytest.mark.parametrize("seed", range(20, 28))
def test_fuzz_multi_step_swap(seed):
    """randomized MULTI-STEP campaigns: exchange / swap / re-fill cycles
    with mixed dtypes and asymmetric radii — catches swap-parity and
    stale-pointer bugs that single-exchange fuzz cannot"""
    rng = random.Random(seed)
    n_dom = rng.choice([1, 2, 4])
    radius = random_radius(rng)
    max_r = max(
        radius.dir(x, y, z) for x in (-1, 0, 1) for y in (-1, 0, 1) for z in (-1, 0, 1)
    )
    size = tuple(max(rng.randint(8, 20), max_r * n_dom * 2 + n_dom) for _ in range(3))

    dd = sa.DistributedDomain(*size, backend="torch")
    dd.set_radius(radius)
    dd.set_gpus([0] * n_dom)
    handles = []
    for qi in range(rng.randint(1, 3)):
        dtype = rng.choice([np.float32, np.float64])
        handles.append((dd.add_data(dtype, f"q{qi}"), dtype))
    dd.realize()

    for step in range(rng.randint(2, 4)):
        scale = 1.0 + step
        for h, dtype in handles:
            for li in range(dd.num_local()):
                lo, hi = dd.local_rect(li)
                dd.write_global(li, lo, ripple_block(lo, hi, dd.size, scale).astype(dtype), h)
        dd.exchange()
        for h, dtype in handles:
            check_valid_regions(dd, h, scale)
        dd.swap()
