"""End-to-end GPU tests: full FL rounds through the HIP op path on one
MI355X (config 2 of BASELINE.json at reduced size), determinism, and the
model zoo forward/backward."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = 'cuda:0'


def _args(**over):
    from rlr_amd.options import default_args
    base = dict(num_agents=4, rounds=2, snap=1, local_ep=1, bs=128,
                synthetic=True, no_tb=True, data='fmnist', num_corrupt=1,
                poison_frac=0.5, robustLR_threshold=3, device=DEV)
    base.update(over)
    return default_args(**base)


@pytest.fixture(autouse=True)
def small_sizes(monkeypatch):
    import rlr_amd.data.datasets as D
    monkeypatch.setitem(D.DEFAULT_SIZES, 'fmnist', (4000, 500))
    monkeypatch.setitem(D.DEFAULT_SIZES, 'cifar10', (4000, 500))


def test_native_ext_loaded():
    from rlr_amd.ops import ext
    assert ext().__file__.endswith('_hip.so')


def test_models_fwd_bwd_gpu():
    from rlr_amd.models import CNN_MNIST, CNN_CIFAR, ResNet18
    from rlr_amd.ops import functional as Fo
    for model, shape in [(CNN_MNIST(), (8, 1, 28, 28)),
                         (CNN_CIFAR(), (8, 3, 32, 32)),
                         (ResNet18(), (8, 3, 32, 32))]:
        m = model.to(DEV)
        m.train()
        m.set_dropout_seed(7)
        x = torch.randn(*shape, device=DEV)
        y = torch.randint(0, 10, (shape[0],), device=DEV)
        loss = Fo.cross_entropy(m(x), y)
        loss.backward()
        assert torch.isfinite(loss).item()
        for p in m.parameters():
            assert p.grad is not None and torch.isfinite(p.grad).all()


def test_fl_round_runs_and_learns():
    from rlr_amd.federated import run
    args = _args(rounds=4, snap=4, robustLR_threshold=0, num_corrupt=0,
                 poison_frac=0.0)
    h = run(args)
    assert h['val_acc'][-1] > 0.5, h['val_acc']


def test_gpu_run_deterministic():
    """Identical seeds -> bitwise identical global model (no atomics on the
    training or aggregation paths)."""
    from rlr_amd.federated import run
    h1 = run(_args())
    h2 = run(_args())
    assert torch.equal(h1['final_params'], h2['final_params'])


def test_single_step_matches_cpu_reference():
    """One fwd+bwd+clipped-SGD step on the HIP path vs the identical step
    on CPU eager (dropout off).  Multi-step trajectories are chaotic under
    fp32 reordering (relu boundary flips), so parity is pinned per-step and
    per-kernel (test_kernels_gpu), not over whole local epochs."""
    from rlr_amd.flatmodel import FlatParamModel
    from rlr_amd.models import get_model
    from rlr_amd.ops import flat as flat_ops
    from rlr_amd.ops import functional as Fo
    from rlr_amd.utils.rng import derive_seed

    torch.manual_seed(0)
    x_cpu = torch.randn(64, 1, 28, 28)
    y_cpu = torch.randint(0, 10, (64,))
    upds = {}
    for dev in (DEV, 'cpu'):
        torch.manual_seed(derive_seed(42, 'init'))
        gm = FlatParamModel(get_model('fmnist'), dev)
        gm.model.p_drop = 0.0
        gm.train()
        theta0 = gm.flat_params.clone()
        gm.zero_grad()
        loss = Fo.cross_entropy(gm(x_cpu.to(dev)), y_cpu.to(dev))
        loss.backward()
        flat_ops.clipped_sgd_step_(gm.flat_params, gm.flat_grads,
                                   gm.momentum, 0.1, 0.9, 10.0)
        upds[dev] = (gm.flat_params - theta0).cpu()
    d = (upds[DEV] - upds['cpu']).abs().max().item()
    assert d < 1e-4, f"single-step GPU vs CPU divergence {d}"


def test_defense_semantics_gpu():
    """Short GPU run: RLR suppresses the backdoor (north-star semantics on
    the real HIP path)."""
    from rlr_amd.federated import run
    import rlr_amd.data.datasets as D
    D.DEFAULT_SIZES['fmnist'] = (6000, 500)
    poisons = {}
    for thr in (0, 5):
        args = _args(num_agents=10, num_corrupt=2, poison_frac=1.0,
                     rounds=15, snap=3, local_ep=2, bs=128,
                     pattern_type='square', robustLR_threshold=thr)
        h = run(args)
        poisons[thr] = h['poison_acc']
    assert max(poisons[0][-3:]) > 0.4, poisons
    assert max(poisons[5][-2:]) < 0.2, poisons


def test_gpu_resume_bitwise(tmp_path):
    """checkpoint -> resume on GPU (graphs + streams on) is bitwise equal
    to the uninterrupted run."""
    from rlr_amd.federated import run
    import os
    h_full = run(_args(rounds=4, snap=2, ckpt_dir=str(tmp_path / 'a')))
    run(_args(rounds=2, snap=2, ckpt_dir=str(tmp_path / 'b')))
    ck = os.path.join(str(tmp_path / 'b'), 'round_000002.pt')
    h_res = run(_args(rounds=4, snap=2, resume=ck,
                      ckpt_dir=str(tmp_path / 'c')))
    assert torch.equal(h_full['final_params'], h_res['final_params'])
