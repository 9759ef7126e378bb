"""Unit tests for the aggregation rules against tiny closed-form cases
(reference semantics: src/aggregation.py:48-75)."""

import pytest
import torch

from rlr_amd.aggregation import Aggregation
from rlr_amd.options import default_args


def make_agg(**over):
    args = default_args(no_tb=True, **over)
    sizes = {i: 10 * (i + 1) for i in range(8)}
    return Aggregation(sizes, 6, None, args), args


def test_rlr_vote_hand_built():
    # theta=3, 4 agents; per-coordinate |sum of signs|
    agg, _ = make_agg(robustLR_threshold=3, aggr='avg')
    U = torch.tensor([
        [+1.0, -2.0, +0.5, 0.0, +1.0, -1.0],
        [+2.0, -1.0, -0.5, 0.0, +1.0, -1.0],
        [+0.1, -5.0, +0.5, 0.0, -1.0, -1.0],
        [+3.0, -0.1, -0.5, 0.0, -1.0, +1.0],
    ], dtype=torch.float64)
    lr = agg.compute_robustLR(U)
    # |sums|: 4, 4, 0, 0, 0, 2  -> >=3: +1 else -1
    assert torch.equal(lr, torch.tensor([1., 1., -1., -1., -1., -1.],
                                        dtype=torch.float64))


def test_rlr_vote_threshold_boundary():
    agg, _ = make_agg(robustLR_threshold=2, aggr='avg')
    U = torch.tensor([[+1.0, +1.0], [+1.0, -1.0]], dtype=torch.float64)
    lr = agg.compute_robustLR(U)
    # sums: 2 (== theta -> +), 0 (< theta -> -)
    assert lr.tolist() == [1.0, -1.0]


def test_fedavg_weighting():
    agg, _ = make_agg(aggr='avg')
    U = torch.tensor([[1.0] * 6, [4.0] * 6], dtype=torch.float64)
    out = agg.agg_avg(U, [0, 1])  # sizes 10, 20
    expect = (10 * 1.0 + 20 * 4.0) / 30
    assert torch.allclose(out, torch.full((6,), expect, dtype=torch.float64))


@pytest.mark.parametrize("k", [3, 4, 5, 8])
def test_comed_matches_torch_median(k):
    agg, _ = make_agg(aggr='comed')
    U = torch.randn(k, 101, dtype=torch.float64)
    out = agg.agg_comed(U)
    assert torch.equal(out, torch.median(U, dim=0).values)


def test_sign_aggregate():
    agg, _ = make_agg(aggr='sign')
    U = torch.tensor([[+5.0, -1.0, 0.0], [+1.0, -2.0, 0.0],
                      [-9.0, +7.0, 0.0]], dtype=torch.float64)
    out = agg.agg_sign(U)
    assert out.tolist() == [1.0, -1.0, 0.0]


def test_apply_update_fp64_to_fp32():
    """theta <- float32(theta + lr*agg) (reference aggregation.py:38-40)."""
    from rlr_amd.flatmodel import FlatParamModel
    agg, args = make_agg(aggr='avg')

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.p = torch.nn.Parameter(torch.zeros(6))

    gm = FlatParamModel(M(), 'cpu')
    gm.flat_params.fill_(1.0)
    delta = torch.full((6,), 0.25, dtype=torch.float64)
    agg._apply(gm, None, delta)
    assert gm.flat_params.dtype == torch.float32
    assert torch.allclose(gm.flat_params, torch.full((6,), 1.25))


def test_rlr_applied_sign_flip():
    """End-to-end: coords below threshold move AGAINST the aggregate."""
    from rlr_amd.flatmodel import FlatParamModel
    agg, args = make_agg(aggr='avg', robustLR_threshold=2)

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.p = torch.nn.Parameter(torch.zeros(2))

    gm = FlatParamModel(M(), 'cpu')
    # both agents agree on coord 0 (+), disagree on coord 1
    U = torch.tensor([[+1.0, +1.0], [+1.0, -1.0]], dtype=torch.float64)
    agg.aggregate_updates(gm, U, cur_round=1, agent_ids=[0, 1])
    # coord0: lr=+1, avg=+1 -> +1 ; coord1: lr=-1, avg=(10*1-20*1)/30=-1/3 -> +1/9... sign matters
    assert gm.flat_params[0] > 0
    # coord1 avg = (10*1 + 20*(-1))/30 = -1/3; lr=-1 -> +1/3 (ascent against it)
    assert gm.flat_params[1] == pytest.approx(1 / 3, rel=1e-5)


def test_noise_deterministic_per_round():
    agg, _ = make_agg(aggr='avg', noise=0.5, clip=2.0)
    n1 = agg._noise(3, torch.device('cpu'), torch.float64)
    n2 = agg._noise(3, torch.device('cpu'), torch.float64)
    n3 = agg._noise(4, torch.device('cpu'), torch.float64)
    assert torch.equal(n1, n2)
    assert not torch.equal(n1, n3)
    assert n1.shape == (6,)


def test_dict_api_parity():
    """The reference dict-based call path still works."""
    from rlr_amd.flatmodel import FlatParamModel
    agg, _ = make_agg(aggr='avg')

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.p = torch.nn.Parameter(torch.zeros(6))

    gm = FlatParamModel(M(), 'cpu')
    d = {0: torch.ones(6, dtype=torch.float64),
         1: torch.full((6,), 4.0, dtype=torch.float64)}
    agg.aggregate_updates(gm, d, cur_round=1)
    expect = (10 * 1.0 + 20 * 4.0) / 30
    assert torch.allclose(gm.flat_params, torch.full((6,), expect))
