"""Seeded fuzz campaigns through the NATIVE engine on hardware: random
sizes, radii (incl. asymmetric), quantity sets, subdomain counts and
exchange groups, each verified with the full-region ripple check across
multiple exchange/swap rounds. The GPU analog of the CPU fuzz tier
(test_fuzz_exchange.py runs the torch oracle)."""
import random

import numpy as np
import pytest

import stencil_amd as sa
from stencil_amd import _C

from util import check_valid_regions, fill_interiors

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("seed", [11, 29, 53])
def test_native_fuzz_campaign(seed):
    rng = random.Random(seed)
    for case in range(4):
        size = tuple(rng.randint(8, 24) for _ in range(3))
        r = _C.Radius.constant(rng.randint(1, 3))
        for _ in range(rng.randint(0, 3)):
            d = (rng.randint(-1, 1), rng.randint(-1, 1), rng.randint(-1, 1))
            if d != (0, 0, 0):
                r.set_dir(*d, rng.randint(0, 3))
        n_local = rng.choice([1, 2, 4])
        nq = rng.randint(1, 3)
        dd = sa.DistributedDomain(*size, backend="native")
        dd.set_radius(r)
        dd.set_gpus([0] * n_local)
        hs = [
            dd.add_data(rng.choice([np.float32, np.float64]), f"q{i}") for i in range(nq)
        ]
        if nq >= 2 and rng.random() < 0.5:
            dd.set_exchange_groups([[0], list(range(1, nq))])
        dd.realize()
        groups = range(len(dd.exchange_groups)) if dd.exchange_groups else [0]
        for h in hs:
            fill_interiors(dd, h, scale=1.0 + h.index)
        for _round in range(rng.randint(1, 3)):
            for g in groups:
                dd.exchange(group=g)
            for h in hs:
                # valid-regions check: randomly sparsified radii leave
                # radius-0 sides legitimately unexchanged
                check_valid_regions(dd, h, scale=1.0 + h.index)
            dd.swap()
            for h in hs:
                fill_interiors(dd, h, scale=1.0 + h.index)
