"""Agent semantics (reference agent.py:33-64): fused clip+SGD step parity
with torch's clip_grad_norm_ + optim.SGD, per-batch PGD placement, fp64
update dtype, momentum reset, and global-model restoration."""

import torch
import pytest

from rlr_amd.agent import Agent
from rlr_amd.data import get_datasets, distribute_data
from rlr_amd.flatmodel import FlatParamModel
from rlr_amd.models import get_model
from rlr_amd.options import default_args
from rlr_amd.ops import flat as flat_ops
from rlr_amd.utils.rng import derive_seed, np_rng


def test_fused_step_matches_torch():
    torch.manual_seed(0)
    n = 1000
    p_ref = torch.randn(n)
    g = torch.randn(n) * 5
    v_ref = torch.randn(n).abs()

    # torch reference: clip_grad_norm_ + SGD(momentum)
    p_t = torch.nn.Parameter(p_ref.clone())
    p_t.grad = g.clone()
    opt = torch.optim.SGD([p_t], lr=0.1, momentum=0.9)
    opt.state[p_t] = {'momentum_buffer': v_ref.clone()}
    torch.nn.utils.clip_grad_norm_([p_t], 2.0)
    opt.step()

    # ours
    p, gg, v = p_ref.clone(), g.clone(), v_ref.clone()
    flat_ops.clipped_sgd_step_(p, gg, v, 0.1, 0.9, 2.0)
    assert torch.allclose(p, p_t.detach(), atol=1e-6)


def test_fused_step_no_clip_needed():
    """Norm below max_norm -> plain SGD step."""
    p = torch.ones(10)
    g = torch.full((10,), 0.01)
    v = torch.zeros(10)
    flat_ops.clipped_sgd_step_(p, g, v, 0.5, 0.0, 10.0)
    assert torch.allclose(p, torch.ones(10) - 0.5 * 0.01, atol=1e-7)


def test_pgd_projection_math():
    p = torch.zeros(4)
    theta0 = torch.zeros(4)
    p[0] = 10.0  # ||update|| = 10
    flat_ops.pgd_project_(p, theta0, 2.0)
    assert torch.allclose(p, torch.tensor([2.0, 0, 0, 0]))
    # inside the ball: untouched
    p2 = torch.tensor([0.5, 0, 0, 0.])
    flat_ops.pgd_project_(p2, theta0, 2.0)
    assert torch.allclose(p2, torch.tensor([0.5, 0, 0, 0.]))


def _tiny_world(**over):
    args = default_args(num_agents=2, local_ep=1, bs=32, synthetic=True,
                        data='fmnist', **over)
    import rlr_amd.data.datasets as D
    old = D.DEFAULT_SIZES['fmnist']
    D.DEFAULT_SIZES['fmnist'] = (600, 100)
    try:
        train, _ = get_datasets('fmnist', args)
    finally:
        D.DEFAULT_SIZES['fmnist'] = old
    groups = distribute_data(train, args)
    agents = [Agent(i, args, train_dataset=train, data_idxs=groups[i])
              for i in range(2)]
    torch.manual_seed(derive_seed(args.seed, 'init'))
    gm = FlatParamModel(get_model('fmnist'), 'cpu')
    return args, agents, gm


def test_local_train_restores_global_and_returns_fp64():
    args, agents, gm = _tiny_world()
    theta0 = gm.flat_params.clone()
    upd = agents[0].local_train(gm, rnd=1)
    assert upd.dtype == torch.float64
    assert upd.shape == (gm.n_params,)
    assert torch.equal(gm.flat_params, theta0)  # restored
    assert upd.abs().sum() > 0                  # actually trained


def test_local_train_matches_reference_loop():
    """Our fused hot loop == an explicit torch SGD/clip/PGD loop over the
    same model, data order and dropout stream (reference agent.py:40-64)."""
    args, agents, gm = _tiny_world(clip=0.5)
    agent = agents[0]
    agent._materialize()
    theta0 = gm.flat_params.clone()
    upd_ours = agent.local_train(gm, rnd=3)

    # reference-style loop on the same FlatParamModel
    gm.flat_params.copy_(theta0)
    theta0_64 = theta0.double()
    gm.train()
    gm.zero_momentum()
    gm.set_dropout_seed(derive_seed(args.seed, 'dropout', agent.id, 3))
    opt = torch.optim.SGD(gm.model.parameters(), lr=args.client_lr,
                          momentum=args.client_moment)
    n = agent._X.shape[0]
    for ep in range(args.local_ep):
        perm = np_rng(args.seed, 'shuffle', agent.id, 3, ep).permutation(n)
        for lo in range(0, n, args.bs):
            sel = torch.as_tensor(perm[lo:lo + args.bs])
            opt.zero_grad()
            out = gm(agent._X[sel])
            loss = torch.nn.functional.cross_entropy(out, agent._Y[sel])
            loss.backward()
            torch.nn.utils.clip_grad_norm_(gm.model.parameters(), 10)
            opt.step()
            if args.clip > 0:
                with torch.no_grad():
                    update = gm.flat_params - theta0
                    denom = max(1, float(torch.norm(update, p=2)) / args.clip)
                    gm.flat_params.copy_(theta0 + update / denom)
    upd_ref = gm.flat_params.double() - theta0_64
    gm.flat_params.copy_(theta0)
    assert torch.allclose(upd_ours, upd_ref, atol=1e-5), \
        (upd_ours - upd_ref).abs().max()


def test_momentum_reset_between_rounds():
    """A fresh optimizer per local_train (reference agent.py:37): two calls
    from the same theta give identical updates."""
    args, agents, gm = _tiny_world()
    u1 = agents[0].local_train(gm, rnd=5)
    u2 = agents[0].local_train(gm, rnd=5)
    assert torch.equal(u1, u2)


def test_corrupt_agent_poisons_its_data():
    args = default_args(num_agents=2, num_corrupt=1, poison_frac=1.0,
                        synthetic=True, data='fmnist', pattern_type='square',
                        local_ep=1, bs=32)
    import rlr_amd.data.datasets as D
    old = D.DEFAULT_SIZES['fmnist']
    D.DEFAULT_SIZES['fmnist'] = (600, 100)
    try:
        train, _ = get_datasets('fmnist', args)
    finally:
        D.DEFAULT_SIZES['fmnist'] = old
    groups = distribute_data(train, args)
    a0 = Agent(0, args, train_dataset=train, data_idxs=groups[0])
    # all of agent 0's base-class samples are now target_class
    labels0 = train.targets[torch.as_tensor(groups[0])]
    assert (labels0 == args.base_class).sum() == 0
    # agent 1 (honest) untouched
    a1 = Agent(1, args, train_dataset=train, data_idxs=groups[1])
    labels1 = train.targets[torch.as_tensor(groups[1])]
    assert (labels1 == args.base_class).sum() > 0
