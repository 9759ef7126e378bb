"""Manual-tape training step vs the autograd path: bitwise equality.

The manual tape (models.cnn.*.manual_step + *_bwd_into bindings) runs the
same kernel sequence as the autograd path but writes grads directly into
the flat views — assignment must equal accumulate-into-zero bitwise, and
several SGD steps must produce identical parameters."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _run_steps(model_cls, data_shape, use_manual, steps=5, dtype=None):
    from rlr_amd.flatmodel import FlatParamModel
    from rlr_amd.ops import flat as flat_ops
    from rlr_amd.ops import functional as Fo

    torch.manual_seed(7)
    model = model_cls()
    if dtype is not None:
        model.set_compute_dtype(dtype)
    gm = FlatParamModel(model, 'cuda:0')
    gm.train()
    gm.set_dropout_seed(99)
    gm.ensure_grad_views()
    if not use_manual:
        gm.model.manual_step = None  # force the autograd fallback

    g = torch.Generator().manual_seed(3)
    X = torch.randn(steps, 64, *data_shape, generator=g).cuda()
    Y = torch.randint(0, 10, (steps, 64), generator=g).cuda()
    for i in range(steps):
        model = gm.model
        if getattr(model, 'manual_step', None) is not None:
            model.manual_step(X[i], Y[i], gm.dloss_ones())
        else:
            gm.zero_grad()
            out = gm(X[i])
            loss = Fo.cross_entropy(out, Y[i])
            loss.backward()
        flat_ops.clipped_sgd_step_(gm.flat_params, gm.flat_grads,
                                   gm.momentum, 0.1, 0.9, 10.0)
        model.rng.advance_step()
    torch.cuda.synchronize()
    return gm.flat_grads.clone(), gm.flat_params.clone()


@pytest.mark.parametrize('which', ['mnist', 'cifar'])
def test_manual_tape_bitwise(which):
    from rlr_amd.models import CNN_MNIST, CNN_CIFAR
    cls, shape = ((CNN_MNIST, (1, 28, 28)) if which == 'mnist'
                  else (CNN_CIFAR, (3, 32, 32)))
    g_auto, p_auto = _run_steps(cls, shape, use_manual=False)
    g_man, p_man = _run_steps(cls, shape, use_manual=True)
    assert torch.equal(g_auto, g_man), \
        (g_auto - g_man).abs().max().item()
    assert torch.equal(p_auto, p_man)


@pytest.mark.parametrize('dtype', [None, torch.bfloat16])
def test_manual_tape_resnet_bitwise(dtype):
    """ResNet18 tape (residual joins, BN, bf16 path) vs autograd."""
    from rlr_amd.models import ResNet18
    g_auto, p_auto = _run_steps(ResNet18, (3, 32, 32), use_manual=False,
                                steps=3, dtype=dtype)
    g_man, p_man = _run_steps(ResNet18, (3, 32, 32), use_manual=True,
                              steps=3, dtype=dtype)
    assert torch.equal(g_auto, g_man), \
        (g_auto - g_man).abs().max().item()
    assert torch.equal(p_auto, p_man)
