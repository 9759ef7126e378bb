"""bf16 conv kernels vs the fp32 HIP path (which is itself validated
against torch autograd), plus ResNet18 bf16 e2e."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = 'cuda:0'


@pytest.fixture(scope='module')
def ext():
    from rlr_amd.ops import ext as _e
    return _e()


CASES = [
    # (Nb, C, H, W, KO, R, stride, pad) — bf16 fast-path shapes
    (8, 32, 26, 26, 64, 3, 1, 0),
    (8, 64, 32, 32, 64, 3, 1, 1),     # resnet body
    (8, 64, 32, 32, 128, 3, 2, 1),    # downsample
    (8, 64, 32, 32, 128, 1, 2, 0),    # 1x1 shortcut
]


@pytest.mark.parametrize("nb,c,h,w,ko,r,st,pd", CASES)
def test_conv_bf16_fwd_bwd(ext, nb, c, h, w, ko, r, st, pd):
    torch.manual_seed(0)
    x32 = torch.randn(nb, c, h, w, device=DEV)
    w32 = torch.randn(ko, c, r, r, device=DEV) * 0.05
    b32 = torch.randn(ko, device=DEV)
    x16 = x32.to(torch.bfloat16)

    y16 = ext.conv2d_fwd(x16, w32, b32, st, pd, False)
    assert y16.dtype == torch.bfloat16
    ref = torch.nn.functional.conv2d(x16.float(), w32, b32, stride=st,
                                     padding=pd)
    rel = (y16.float() - ref).abs().mean() / (ref.abs().mean() + 1e-3)
    assert rel < 2e-2, float(rel)

    dy16 = torch.randn_like(ref).to(torch.bfloat16)
    dx16, dw, db = ext.conv2d_bwd(x16, w32, dy16, st, pd, True, True)
    assert dx16.dtype == torch.bfloat16 and dw.dtype == torch.float32
    xg = x16.float().requires_grad_(True)
    wg = w32.clone().requires_grad_(True)
    bg = b32.clone().requires_grad_(True)
    torch.nn.functional.conv2d(xg, wg, bg, stride=st,
                               padding=pd).backward(dy16.float())
    rel_dx = (dx16.float() - xg.grad).abs().mean() / \
        (xg.grad.abs().mean() + 1e-3)
    rel_dw = (dw - wg.grad).abs().mean() / (wg.grad.abs().mean() + 1e-3)
    rel_db = (db - bg.grad).abs().mean() / (bg.grad.abs().mean() + 1e-3)
    assert rel_dx < 2e-2, float(rel_dx)
    assert rel_dw < 2e-2, float(rel_dw)
    assert rel_db < 1e-2, float(rel_db)


def test_conv_bf16_first_layer_fallback(ext):
    """C=3 takes the fp32 fallback with casts — must still be correct."""
    x16 = torch.randn(4, 3, 32, 32, device=DEV).to(torch.bfloat16)
    w32 = torch.randn(64, 3, 3, 3, device=DEV) * 0.1
    y = ext.conv2d_fwd(x16, w32, None, 1, 1, True)
    assert y.dtype == torch.bfloat16
    ref = torch.relu(torch.nn.functional.conv2d(x16.float(), w32,
                                                padding=1))
    rel = (y.float() - ref).abs().mean() / (ref.abs().mean() + 1e-3)
    assert rel < 2e-2


def test_resnet18_bf16_fl_round():
    import rlr_amd.data.datasets as D
    old = D.DEFAULT_SIZES['cifar10']
    D.DEFAULT_SIZES['cifar10'] = (2000, 400)
    try:
        from rlr_amd.federated import run
        from rlr_amd.options import default_args
        args = default_args(data='cifar10', model='resnet18', dtype='bf16',
                            num_agents=2, rounds=2, snap=2, local_ep=1,
                            bs=64, synthetic=True, no_tb=True, device=DEV)
        h = run(args)
        assert torch.isfinite(h['final_params']).all()
    finally:
        D.DEFAULT_SIZES['cifar10'] = old


def test_cnn_bf16_learns():
    """FMNIST CNN under --dtype bf16 still learns (first layer falls back,
    conv2/fc run bf16 MFMA)."""
    import rlr_amd.data.datasets as D
    old = D.DEFAULT_SIZES['fmnist']
    D.DEFAULT_SIZES['fmnist'] = (2000, 400)
    try:
        from rlr_amd.federated import run
        from rlr_amd.options import default_args
        args = default_args(data='fmnist', dtype='bf16', num_agents=2,
                            rounds=3, snap=3, local_ep=1, bs=128,
                            synthetic=True, no_tb=True, device=DEV)
        h = run(args)
        assert h['val_acc'][-1] > 0.5, h['val_acc']
    finally:
        D.DEFAULT_SIZES['fmnist'] = old


def test_tap_kernels_run_to_run_deterministic():
    """The tap-resident bf16 conv kernels (s1/s2, all width classes) use
    fixed-order partial combines — two identical calls must be bitwise
    equal (no atomics anywhere on the training path)."""
    import torch
    from rlr_amd.ops import ext
    E = ext()
    torch.manual_seed(11)
    for (c, hw, ko, stride, pad) in [(64, 32, 64, 1, 1),   # tap s1 W=32
                                     (256, 8, 256, 1, 1),  # tap s1 W=8
                                     (512, 4, 512, 1, 1),  # W=4 packed
                                     (64, 32, 128, 2, 1),  # tap s2
                                     (256, 8, 512, 2, 1)]: # s2 OW=4 bwd-d
        x = (torch.randn(32, c, hw, hw, device='cuda:0')
             .to(torch.bfloat16)
             .contiguous(memory_format=torch.channels_last))
        w = torch.randn(ko, c, 3, 3, device='cuda:0') * 0.05
        y1 = E.conv2d_fwd(x, w, None, stride, pad, False)
        y2 = E.conv2d_fwd(x, w, None, stride, pad, False)
        assert torch.equal(y1, y2), (c, hw, ko, stride)
        dy = torch.randn_like(y1)
        dx1, dw1, _ = E.conv2d_bwd(x, w, dy, stride, pad, False, True)
        dx2, dw2, _ = E.conv2d_bwd(x, w, dy, stride, pad, False, True)
        assert torch.equal(dw1, dw2), (c, hw, ko, stride)
        assert torch.equal(dx1, dx2), (c, hw, ko, stride)
