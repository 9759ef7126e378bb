"""Model-zoo e2e on GPU: full FL rounds for CIFAR10 (CNN + ResNet18/BN
buffer aggregation) and the non-IID Fed-EMNIST path."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = 'cuda:0'


@pytest.fixture(autouse=True)
def small_sizes(monkeypatch):
    import rlr_amd.data.datasets as D
    monkeypatch.setitem(D.DEFAULT_SIZES, 'cifar10', (2000, 400))
    monkeypatch.setitem(D.DEFAULT_SIZES, 'fedemnist', (8, 120))


def _args(**over):
    from rlr_amd.options import default_args
    base = dict(num_agents=2, rounds=2, snap=2, local_ep=1, bs=128,
                synthetic=True, no_tb=True, device=DEV)
    base.update(over)
    return default_args(**base)


def test_cifar_cnn_fl_rounds():
    from rlr_amd.federated import run
    # clean run (with K=2 agents a 50%-weight corrupt agent hijacks the
    # model and RLR theta=2 flips every disputed coordinate — both are
    # degenerate by design; attack/defense semantics are covered at K=10
    # in test_e2e_gpu and the DBA pattern in test_kernels_gpu)
    h = run(_args(data='cifar10', rounds=10, snap=5, local_ep=2))
    assert torch.isfinite(h['final_params']).all()
    # the deep CIFAR CNN's short-run accuracy is chaotic (fp32 reorder
    # flips it between ~0.1 and ~0.2); pin the robust signal: val CE loss
    # dips clearly below the 10-class chance level ln(10)=2.303
    assert min(h['val_loss']) < 2.25, h['val_loss']


def test_resnet18_fl_rounds_with_bn_buffers():
    from rlr_amd.federated import run
    h = run(_args(data='cifar10', model='resnet18', bs=64, rounds=2))
    assert torch.isfinite(h['final_params']).all()
    from rlr_amd.flatmodel import FlatParamModel
    from rlr_amd.models import get_model
    gm = FlatParamModel(get_model('cifar10', 'resnet18'), 'cpu')
    assert h['final_params'].numel() == gm.n_params


def test_fedemnist_fl_rounds():
    from rlr_amd.federated import run
    h = run(_args(data='fedemnist', num_agents=8, agent_frac=0.5,
                  num_corrupt=2, poison_frac=0.5, robustLR_threshold=2,
                  bs=32, pattern_type='square'))
    assert torch.isfinite(h['final_params']).all()


def test_comed_and_sign_aggregation_gpu():
    from rlr_amd.federated import run
    for aggr in ('comed', 'sign'):
        h = run(_args(data='cifar10', aggr=aggr, rounds=1, snap=1,
                      server_lr=0.01))
        assert torch.isfinite(h['final_params']).all(), aggr
