"""Checkpoint format + resume determinism: run 4 rounds straight vs
2 rounds -> checkpoint -> resume 2 more; final global models must be
bit-identical (the format is world-size independent by construction)."""

import os

import torch

from rlr_amd.federated import run
from rlr_amd.options import default_args


def _args(tmp, **over):
    base = dict(num_agents=2, rounds=4, snap=1, local_ep=1, bs=64,
                synthetic=True, no_tb=True, data='fmnist',
                ckpt_dir=str(tmp))
    base.update(over)
    return default_args(**base)


def test_resume_bitwise_identical(tmp_path, tiny_sizes):
    a_full = _args(tmp_path / 'full')
    h_full = run(a_full)

    a_half = _args(tmp_path / 'half', rounds=2)
    run(a_half)
    ck = os.path.join(str(tmp_path / 'half'), 'round_000002.pt')
    assert os.path.exists(ck)

    a_res = _args(tmp_path / 'resumed', rounds=4, resume=ck)
    h_res = run(a_res)

    assert torch.equal(h_full['final_params'], h_res['final_params'])


def test_checkpoint_contents(tmp_path, tiny_sizes):
    a = _args(tmp_path, rounds=1)
    run(a)
    state = torch.load(os.path.join(str(tmp_path), 'round_000001.pt'),
                       weights_only=False)
    assert state['version'] == 1
    assert state['round'] == 1
    assert state['params'].dtype == torch.float32
    assert state['params'].shape == (1_199_882,)
    assert state['args']['data'] == 'fmnist'
    assert 'seed' in state


def test_resume_refuses_config_mismatch(tmp_path, tiny_sizes):
    """Resuming with a different model/dataset must fail loudly, not
    silently corrupt the weights (checkpoint.load_checkpoint guard)."""
    import pytest
    from rlr_amd.federated import build_world
    from rlr_amd.utils import load_checkpoint

    a = _args(tmp_path, rounds=1)
    run(a)
    ck = os.path.join(str(tmp_path), 'round_000001.pt')

    b = default_args(num_agents=2, rounds=2, snap=1, local_ep=1, bs=64,
                     synthetic=True, no_tb=True, data='cifar10')
    world = build_world(b)
    with pytest.raises(ValueError, match='refusing to resume'):
        load_checkpoint(ck, world['global_model'], expect_args=b)
    # param-count guard fires even without expect_args
    with pytest.raises(ValueError, match='params'):
        load_checkpoint(ck, world['global_model'])
