"""FlatParamModel invariants: parameter views alias the flat buffer, the
packing order is the module parameter order (reference
parameters_to_vector semantics, agent.py:35), and snapshot/restore is
exact."""

import torch

from rlr_amd.flatmodel import FlatParamModel
from rlr_amd.models import get_model


def _fm(data='fmnist'):
    torch.manual_seed(0)
    return FlatParamModel(get_model(data), 'cpu')


def test_param_views_alias_flat_buffer():
    fm = _fm()
    total = sum(p.numel() for p in fm.model.parameters())
    assert fm.n_params == total == fm.flat_params.numel()
    for p in fm.model.parameters():
        assert p.data_ptr() >= fm.flat_params.data_ptr()
    # writing through the flat buffer is visible in the module params
    fm.flat_params.zero_()
    for p in fm.model.parameters():
        assert torch.all(p == 0)


def test_packing_order_matches_parameters_to_vector():
    torch.manual_seed(3)
    m = get_model('fmnist')
    ref = torch.nn.utils.parameters_to_vector(m.parameters()).detach()
    fm = FlatParamModel(m, 'cpu')
    assert torch.equal(fm.flat_params, ref)


def test_grads_land_in_flat_buffer():
    fm = _fm()
    fm.train()
    fm.model.p_drop = 0.0
    fm.ensure_grad_views()
    fm.zero_grad()
    x = torch.randn(4, 1, 28, 28)
    y = torch.randint(0, 10, (4,))
    from rlr_amd.ops import functional as Fo
    Fo.cross_entropy(fm(x), y).backward()
    gref = torch.cat([p.grad.reshape(-1) for p in fm.model.parameters()])
    assert torch.equal(fm.flat_grads, gref)
    assert fm.flat_grads.abs().sum() > 0


def test_resnet_buffers_flat():
    torch.manual_seed(1)
    fm = FlatParamModel(get_model('cifar10', 'resnet18'), 'cpu')
    n_buf = sum(b.numel() for b in fm.model.buffers()
                if b.dtype.is_floating_point)
    assert fm.n_buffers == n_buf > 0
    snap = fm.flat_buffers.clone()
    for b in fm.model.buffers():
        if b.dtype.is_floating_point:
            b.add_(1.0)
    assert not torch.equal(fm.flat_buffers, snap)  # views alias
    fm.flat_buffers.copy_(snap)
    off = 0
    for b in fm.model.buffers():
        if b.dtype.is_floating_point:
            assert torch.equal(b.reshape(-1), snap[off:off + b.numel()])
            off += b.numel()


def test_cnn_has_no_buffers():
    assert _fm().n_buffers == 0
