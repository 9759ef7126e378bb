"""Trojan pattern coordinate tests — all patterns x datasets x DBA
quadrants, against hand-derived pixel sets from the reference
(utils.py:181-284)."""

import numpy as np
import pytest
import torch

from rlr_amd.data.poison import pattern_spec, apply_pattern_, poison_dataset
from rlr_amd.options import default_args


def coords_set(spec):
    return set(map(tuple, spec.coords.tolist()))


def test_cifar_full_plus():
    spec = pattern_spec('cifar10', 'plus', agent_idx=-1)
    expect = {(i, 5) for i in range(5, 12)} | {(8, i) for i in range(2, 9)}
    assert coords_set(spec) == expect
    assert spec.value == 0


@pytest.mark.parametrize("agent_idx,expect", [
    (0, {(i, 5) for i in range(5, 9)}),            # upper vertical
    (1, {(i, 5) for i in range(9, 12)}),           # lower vertical
    (2, {(8, i) for i in range(2, 7)}),            # left horizontal
    (3, {(8, i) for i in range(5, 9)}),            # right horizontal
    (4, {(i, 5) for i in range(5, 9)}),            # agent_idx % 4 wraps
])
def test_cifar_dba_quadrants(agent_idx, expect):
    spec = pattern_spec('cifar10', 'plus', agent_idx=agent_idx)
    assert coords_set(spec) == expect


def test_dba_union_is_full_plus():
    """The 4 DBA quarters tile the full plus exactly (no overlap gaps
    beyond the shared center column/row cells)."""
    full = coords_set(pattern_spec('cifar10', 'plus', -1))
    union = set()
    for q in range(4):
        union |= coords_set(pattern_spec('cifar10', 'plus', q))
    assert union == full


def test_fmnist_square():
    spec = pattern_spec('fmnist', 'square')
    assert coords_set(spec) == {(i, j) for i in range(21, 26)
                                for j in range(21, 26)}
    assert spec.value == 255


def test_fmnist_plus():
    spec = pattern_spec('fmnist', 'plus')
    expect = {(i, 5) for i in range(5, 10)} | {(7, i) for i in range(3, 8)}
    assert coords_set(spec) == expect
    assert spec.value == 255


def test_fedemnist_square_and_plus():
    sq = pattern_spec('fedemnist', 'square')
    assert coords_set(sq) == {(i, j) for i in range(21, 26)
                              for j in range(21, 26)}
    assert sq.value == 0.0
    pl = pattern_spec('fedemnist', 'plus')
    expect = {(i, 8) for i in range(8, 13)} | {(10, i) for i in range(6, 11)}
    assert coords_set(pl) == expect


def test_watermark_masks():
    for data, mode in [('fmnist', 'add_wrap_u8'), ('fedemnist', 'sub_float')]:
        for name in ('copyright', 'apple'):
            spec = pattern_spec(data, name)
            assert spec.mode == mode
            assert spec.mask.shape == (28, 28)
            assert spec.mask.dtype == np.uint8
            assert spec.mask.max() == 255  # pattern present


def test_apply_pattern_uint8_hw():
    raw = torch.zeros(4, 28, 28, dtype=torch.uint8)
    apply_pattern_(raw, pattern_spec('fmnist', 'square'))
    assert (raw[:, 21:26, 21:26] == 255).all()
    assert raw.sum() == 4 * 25 * 255


def test_apply_pattern_uint8_wraps():
    """uint8 + mask wraps modulo 256 (reference numpy semantics)."""
    raw = torch.full((1, 28, 28), 200, dtype=torch.uint8)
    spec = pattern_spec('fmnist', 'copyright')
    apply_pattern_(raw, spec)
    m = torch.as_tensor(spec.mask)
    expect = (200 + m.to(torch.int32)) % 256
    assert torch.equal(raw[0].to(torch.int32), expect)


def test_apply_pattern_cifar_channels():
    raw = torch.full((2, 32, 32, 3), 100, dtype=torch.uint8)
    apply_pattern_(raw, pattern_spec('cifar10', 'plus', -1))
    assert (raw[:, 5:12, 5, :] == 0).all()
    assert (raw[:, 8, 2:9, :] == 0).all()
    assert (raw[:, 0, 0, :] == 100).all()


def test_poison_dataset_relabels_and_fraction(tiny_sizes):
    from rlr_amd.data import get_datasets
    args = default_args(synthetic=True, poison_frac=0.5, base_class=5,
                        target_class=7, pattern_type='square', data='fmnist')
    train, _ = get_datasets('fmnist', args)
    base_idxs = (train.targets == 5).nonzero().flatten().tolist()
    n_base = len(base_idxs)
    poisoned = poison_dataset(train, args, agent_idx=0)
    assert len(poisoned) == n_base // 2
    assert (train.targets[poisoned] == 7).all()
    # pattern present on poisoned images
    assert (train.data[poisoned][:, 21:26, 21:26] == 255).all()


def test_poison_deterministic_across_calls(tiny_sizes):
    from rlr_amd.data import get_datasets
    args = default_args(synthetic=True, poison_frac=0.3, data='fmnist',
                        pattern_type='square')
    t1, _ = get_datasets('fmnist', args)
    t2, _ = get_datasets('fmnist', args)
    p1 = poison_dataset(t1, args, agent_idx=2)
    p2 = poison_dataset(t2, args, agent_idx=2)
    assert p1 == p2


def test_unsupported_pattern_raises():
    """Explicit validation improvement over the reference (PARITY.md):
    utils.py:187 silently relabels without writing a pattern for
    unsupported (dataset, pattern_type) combos; this build raises."""
    import pytest
    with pytest.raises(ValueError):
        pattern_spec('cifar10', 'square', agent_idx=-1)
    with pytest.raises(ValueError):
        pattern_spec('fmnist', 'nosuch')
    with pytest.raises(ValueError):
        pattern_spec('nosuchdata', 'plus')
