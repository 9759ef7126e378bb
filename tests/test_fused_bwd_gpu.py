"""Direct numerics for the fused-backward kernels the manual tape uses.

Each fusion claims bitwise equivalence to its two-kernel form; the tape
test checks them end-to-end — these pin each one in isolation, including
the pooled-mask maxpool equivalence proof (window max == pooled value for
post-relu inputs)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def test_dropout_relu_bwd_matches_two_kernels():
    from rlr_amd.ops import ext
    E = ext()
    torch.manual_seed(0)
    dy = torch.randn(256, 9216, device='cuda:0')
    y = torch.randn(256, 9216, device='cuda:0')
    mask = (torch.rand(256, 9216, device='cuda:0') > 0.5).to(torch.uint8)
    fused = E.dropout_relu_bwd(dy, mask, y, 0.5)
    two = E.relu_bwd(y, E.dropout_bwd(dy, mask, 0.5))
    assert torch.equal(fused, two)


def test_maxpool_bwd_relu_pooled_mask_equivalence():
    """maxpool2x2_bwd_relu(dy, idx, pooled) must equal maxpool2x2_bwd
    applied after relu_bwd on the PRE-pool tensor — valid because the
    input is post-relu, so the window max equals the pooled value."""
    from rlr_amd.ops import ext
    E = ext()
    torch.manual_seed(1)
    a = torch.relu(torch.randn(64, 64, 24, 24, device='cuda:0'))
    a = _cl(a)
    pl, idx = E.maxpool2x2_fwd(a)
    dy = _cl(torch.randn_like(pl))
    fused = E.maxpool2x2_bwd_relu(dy, idx, pl, list(a.shape))
    two = E.relu_bwd(a, E.maxpool2x2_bwd(dy, idx, list(a.shape)))
    assert torch.equal(fused, two)


def test_add_relu_bwd_masked_join():
    from rlr_amd.ops import ext
    E = ext()
    torch.manual_seed(2)
    for dt in (torch.float32, torch.bfloat16):
        a = torch.randn(1 << 16, device='cuda:0').to(dt)
        b = torch.randn(1 << 16, device='cuda:0').to(dt)
        y = torch.randn(1 << 16, device='cuda:0').to(dt)
        want = torch.where(y.float() > 0, (a.float() + b.float()),
                           torch.zeros(()).cuda())
        if dt is torch.bfloat16:
            want = want.to(torch.bfloat16)
        got = E.add_relu_bwd_(a.clone(), b, y)
        assert torch.equal(got, want.to(dt)), dt


def test_gap_bwd_relu_masked_broadcast():
    from rlr_amd.ops import ext
    E = ext()
    torch.manual_seed(3)
    h = _cl(torch.randn(32, 512, 4, 4, device='cuda:0').to(torch.bfloat16))
    dy = torch.randn(32, 512, device='cuda:0').to(torch.bfloat16)
    got = E.gap_bwd_relu(dy.contiguous(), h, list(h.shape))
    want = E.relu_bwd(h, E.gap_bwd(dy.contiguous(), list(h.shape)))
    assert torch.equal(_cl(got), _cl(want))
