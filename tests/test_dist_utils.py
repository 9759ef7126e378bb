"""Unit tests for the distributed chunking/gather helpers and logging
names (rank-partition correctness is the foundation of world-size
invariance; e2e coverage is in test_distributed.py)."""

import math

import torch

from rlr_amd.options import default_args
from rlr_amd.parallel.dist import all_gather_updates, chunk_bounds
from rlr_amd.utils.logging import run_name
from rlr_amd.utils.rng import derive_seed, sample_agents


def test_chunk_bounds_partition():
    """Every (n, world) partitions [0, n) exactly, in order, with a
    constant per-rank slot count."""
    for n in (1, 2, 7, 10, 33, 40, 41):
        for world in (1, 2, 3, 4, 8):
            c = math.ceil(n / world)
            covered = []
            for r in range(world):
                lo, hi, chunk = chunk_bounds(n, world, r)
                assert chunk == c
                assert 0 <= lo <= hi <= n
                covered.extend(range(lo, hi))
            assert covered == list(range(n)), (n, world)


def test_chunk_bounds_more_ranks_than_items():
    lo, hi, c = chunk_bounds(3, 8, 7)
    assert lo == hi == 3 and c == 1
    # ranks past the items hold zero valid rows but the same chunk size
    assert chunk_bounds(3, 8, 2)[0] == 2


def test_all_gather_updates_single_process():
    local = torch.arange(12.0).reshape(4, 3)
    out = all_gather_updates(local, [2], 4)
    assert torch.equal(out, local[:2])


def test_sample_agents_shared_and_deterministic():
    a = sample_agents(42, 7, 3383, 0.01)
    b = sample_agents(42, 7, 3383, 0.01)
    assert list(a) == list(b)
    assert len(a) == max(1, int(0.01 * 3383))
    assert len(set(a)) == len(a)  # no replacement
    c = sample_agents(42, 8, 3383, 0.01)
    assert list(a) != list(c)  # round enters the stream
    full = sample_agents(42, 1, 10, 1.0)
    assert sorted(full) == list(range(10))


def test_derive_seed_stable_and_distinct():
    assert derive_seed(42, 'init') == derive_seed(42, 'init')
    assert derive_seed(42, 'init') != derive_seed(43, 'init')
    assert derive_seed(42, 'init') != derive_seed(42, 'noise')
    assert derive_seed(42, 'dropout', 1, 2) != derive_seed(42, 'dropout', 2, 1)


def test_run_name_reference_fields():
    """The reference's run-name embeds these fields (federated.py:27-30);
    dashboards parse them."""
    args = default_args(clip=3.0, noise=0.1, aggr='comed', num_corrupt=4,
                       robustLR_threshold=8, pattern_type='square')
    name = run_name(args)
    for frag in ('clip_val:3.0', 'noise_std:0.1', 'aggr:comed',
                 'num_cor:4', 'thrs_robustLR:8', 'pttrn:square'):
        assert frag in name, (frag, name)
