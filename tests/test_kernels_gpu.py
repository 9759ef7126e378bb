"""HIP kernel numerics vs plain PyTorch fp32 references.
Every test needs an MI355X: run with `pytest -m gpu`."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = 'cuda:0'


@pytest.fixture(scope='module')
def ext():
    from rlr_amd.ops import ext as _e
    return _e()


def t_allclose(a, b, rtol=1e-4, atol=1e-5, msg=''):
    assert torch.allclose(a, b, rtol=rtol, atol=atol), \
        f"{msg} max|d|={float((a - b).abs().max())}"


# ---------------------------------------------------------------- elemwise

def test_relu_fwd_bwd(ext):
    x = torch.randn(1000003, device=DEV)
    y = ext.relu_fwd(x)
    assert torch.equal(y, torch.relu(x))
    dy = torch.randn_like(x)
    dx = ext.relu_bwd(y, dy)
    assert torch.equal(dx, dy * (y > 0))


def test_add_relu(ext):
    a, b = torch.randn(5001, device=DEV), torch.randn(5001, device=DEV)
    assert torch.equal(ext.add_relu_fwd(a, b), torch.relu(a + b))


@pytest.mark.parametrize("shape", [(4, 8, 24, 24), (2, 3, 13, 13)])
def test_maxpool2x2(ext, shape):
    x = torch.randn(*shape, device=DEV)
    y, idx = ext.maxpool2x2_fwd(x)
    ref = torch.nn.functional.max_pool2d(x, 2, 2)
    assert torch.equal(y, ref)
    dy = torch.randn_like(y)
    xg = x.clone().requires_grad_(True)
    torch.nn.functional.max_pool2d(xg, 2, 2).backward(dy)
    dx = ext.maxpool2x2_bwd(dy, idx, list(shape))
    # ties may be broken differently; values must agree where unique
    t_allclose(dx, xg.grad, msg='maxpool bwd')


def test_dropout_stats_and_determinism(ext):
    x = torch.ones(1 << 20, device=DEV)
    y1, m1 = ext.dropout_fwd(x, 0.5, 1234, 7)
    y2, m2 = ext.dropout_fwd(x, 0.5, 1234, 7)
    y3, m3 = ext.dropout_fwd(x, 0.5, 1234, 8)
    assert torch.equal(y1, y2) and torch.equal(m1, m2)
    assert not torch.equal(m1, m3)
    keep = m1.float().mean().item()
    assert abs(keep - 0.5) < 0.01
    assert torch.equal(y1, m1.float() * 2.0)
    dy = torch.randn_like(x)
    dx = ext.dropout_bwd(dy, m1, 0.5)
    assert torch.equal(dx, dy * m1.float() * 2.0)


def test_gap(ext):
    x = torch.randn(8, 32, 7, 7, device=DEV)
    t_allclose(ext.gap_fwd(x), x.mean(dim=(2, 3)), msg='gap fwd')
    dy = torch.randn(8, 32, device=DEV)
    xg = x.clone().requires_grad_(True)
    xg.mean(dim=(2, 3)).backward(dy)
    t_allclose(ext.gap_bwd(dy, list(x.shape)), xg.grad, msg='gap bwd')


def test_cross_entropy(ext):
    logits = torch.randn(256, 10, device=DEV)
    labels = torch.randint(0, 10, (256,), device=DEV)
    loss, softmax = ext.cross_entropy_fwd(logits, labels)
    lg = logits.clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lg, labels)
    t_allclose(loss, ref.detach(), rtol=1e-5, msg='ce loss')
    ref.backward()
    dl = ext.cross_entropy_bwd(softmax, labels,
                               torch.ones((), device=DEV))
    t_allclose(dl, lg.grad, rtol=1e-4, atol=1e-7, msg='ce bwd')


def test_eval_update(ext):
    logits = torch.randn(400, 10, device=DEV)
    labels = torch.randint(0, 10, (400,), device=DEV)
    conf = torch.zeros(100, device=DEV)
    loss_sum = torch.zeros(1, dtype=torch.float64, device=DEV)
    ext.eval_update(logits, labels, conf, loss_sum)
    pred = logits.argmax(1)
    ref_conf = torch.bincount(labels * 10 + pred, minlength=100).float()
    assert torch.equal(conf, ref_conf)
    ref_loss = torch.nn.functional.cross_entropy(
        logits, labels, reduction='sum').double()
    t_allclose(loss_sum[0], ref_loss, rtol=1e-6, msg='eval loss')


# -------------------------------------------------------------- flat opt

def test_clipped_sgd_step(ext):
    n = 1_199_882
    p = torch.randn(n, device=DEV)
    g = torch.randn(n, device=DEV) * 5
    v = torch.randn(n, device=DEV).abs()
    p2, g2, v2 = p.clone(), g.clone(), v.clone()
    ext.clipped_sgd_step(p, g, v, 0.1, 0.9, 2.0)
    # torch reference
    total = torch.linalg.vector_norm(g2)
    scale = torch.clamp(2.0 / (total + 1e-6), max=1.0)
    v_ref = 0.9 * v2 + g2 * scale
    p_ref = p2 - 0.1 * v_ref
    t_allclose(v, v_ref, rtol=1e-5, msg='momentum')
    t_allclose(p, p_ref, rtol=1e-5, msg='params')


def test_pgd_project(ext):
    n = 500_000
    t0 = torch.randn(n, device=DEV)
    p = t0 + torch.randn(n, device=DEV)
    p2 = p.clone()
    ext.pgd_project(p, t0, 0.5)
    upd = p2 - t0
    denom = torch.clamp(torch.linalg.vector_norm(upd) / 0.5, min=1.0)
    t_allclose(p, t0 + upd / denom, rtol=1e-5, msg='pgd')
    # inside the ball: no-op
    q = t0 + 1e-6 * torch.randn(n, device=DEV)
    q2 = q.clone()
    ext.pgd_project(q, t0, 10.0)
    assert torch.equal(q, q2)


def test_delta64(ext):
    p = torch.randn(1000, device=DEV)
    t0 = torch.randn(1000, dtype=torch.float64, device=DEV)
    out = ext.delta64(p, t0)
    assert torch.equal(out, p.double() - t0)


# ------------------------------------------------------------ aggregation

def _stacked(k=7, n=100_003):
    torch.manual_seed(3)
    return torch.randn(k, n, dtype=torch.float64, device=DEV)


def test_rlr_vote_gpu(ext):
    U = _stacked()
    lr = ext.rlr_vote(U, 4.0, 1.0)
    sm = torch.sign(U).sum(0).abs()
    ref = torch.where(sm >= 4, 1.0, -1.0).double()
    assert torch.equal(lr, ref)


def test_agg_avg_gpu(ext):
    U = _stacked()
    w = (torch.arange(7, device=DEV) + 1).double()
    out = ext.agg_avg(U, w)
    ref = (U * w[:, None]).sum(0) / w.sum()
    t_allclose(out, ref, rtol=1e-12, atol=1e-12, msg='avg')


@pytest.mark.parametrize("k", [3, 4, 7, 8])
def test_agg_comed_gpu(ext, k):
    U = _stacked(k)
    out = ext.agg_comed(U)
    assert torch.equal(out, torch.median(U, dim=0).values)


def test_agg_sign_gpu(ext):
    U = _stacked()
    assert torch.equal(ext.agg_sign(U), torch.sign(torch.sign(U).sum(0)))


def test_apply_update_gpu(ext):
    p = torch.randn(1001, device=DEV)
    agg = torch.randn(1001, dtype=torch.float64, device=DEV)
    lr = torch.where(torch.rand(1001, device=DEV) > 0.5, 1.0, -1.0).double()
    p2 = p.clone()
    ext.apply_update(p, agg, lr, 1.0)
    ref = (p2.double() + lr * agg).float()
    assert torch.equal(p, ref)


def test_fused_avg_rlr_apply_matches_pieces(ext):
    U = _stacked(5)
    w = (torch.arange(5, device=DEV) + 3).double()
    p = torch.randn(U.shape[1], device=DEV)
    p_fused = p.clone()
    ext.fused_avg_rlr_apply(U, w, p_fused, True, 3.0, 1.0, 0.0, 0, 0, False)
    lr = ext.rlr_vote(U, 3.0, 1.0)
    avg = ext.agg_avg(U, w)
    ext.apply_update(p, avg, lr, 1.0)
    t_allclose(p_fused, p, rtol=1e-6, atol=1e-7, msg='fused vs pieces')


def test_add_noise_deterministic(ext):
    a = torch.zeros(1 << 16, dtype=torch.float64, device=DEV)
    b = torch.zeros(1 << 16, dtype=torch.float64, device=DEV)
    ext.add_noise(a, 0.5, 99, 3)
    ext.add_noise(b, 0.5, 99, 3)
    assert torch.equal(a, b)
    assert abs(a.mean().item()) < 0.02
    assert abs(a.std().item() - 0.5) < 0.02


# ------------------------------------------------------------ gemm/linear

@pytest.mark.parametrize("m,n,k", [(256, 128, 9216), (256, 10, 128),
                                   (128, 9216, 256), (100, 30, 50),
                                   (256, 9216, 128)])
def test_gemm_shapes(ext, m, n, k):
    torch.manual_seed(0)
    A = torch.randn(m, k, device=DEV)
    B = torch.randn(k, n, device=DEV)
    C = ext.gemm(A, B, None, False)
    ref = A @ B
    t_allclose(C, ref, rtol=1e-4, atol=1e-3, msg=f'gemm {m}x{n}x{k}')


def test_gemm_bias_relu(ext):
    A = torch.randn(64, 300, device=DEV)
    B = torch.randn(300, 70, device=DEV)
    bias = torch.randn(70, device=DEV)
    C = ext.gemm(A, B, bias, True)
    t_allclose(C, torch.relu(A @ B + bias), rtol=1e-4, atol=1e-3,
               msg='gemm+bias+relu')


def test_linear_fwd_bwd(ext):
    torch.manual_seed(1)
    x = torch.randn(256, 9216, device=DEV)
    w = torch.randn(128, 9216, device=DEV) * 0.01
    b = torch.randn(128, device=DEV)
    y = ext.linear_fwd(x, w, b, False)
    xg = x.clone().requires_grad_(True)
    wg = w.clone().requires_grad_(True)
    bg = b.clone().requires_grad_(True)
    ref = torch.nn.functional.linear(xg, wg, bg)
    t_allclose(y, ref.detach(), rtol=1e-4, atol=1e-3, msg='linear fwd')
    dy = torch.randn_like(ref)
    ref.backward(dy)
    dx, dw, db = ext.linear_bwd(x, w, dy)
    t_allclose(dx, xg.grad, rtol=1e-4, atol=1e-3, msg='dx')
    t_allclose(dw, wg.grad, rtol=1e-4, atol=1e-2, msg='dw')
    t_allclose(db, bg.grad, rtol=1e-4, atol=1e-2, msg='db')


# ------------------------------------------------------------------ conv

CONV_CASES = [
    # (Nb, C, H, W, Kout, R, stride, pad) — all reference + ResNet shapes
    (8, 1, 28, 28, 32, 3, 1, 0),    # CNN_MNIST conv1
    (8, 32, 26, 26, 64, 3, 1, 0),   # CNN_MNIST conv2
    (8, 3, 32, 32, 64, 3, 1, 0),    # CNN_CIFAR conv1
    (8, 128, 13, 13, 256, 3, 1, 0), # CNN_CIFAR conv3
    (8, 3, 32, 32, 64, 3, 1, 1),    # ResNet stem
    (8, 64, 32, 32, 128, 3, 2, 1),  # ResNet downsample
    (8, 64, 32, 32, 128, 1, 2, 0),  # ResNet 1x1 shortcut
]


@pytest.mark.parametrize("nb,c,h,w,ko,r,st,pd", CONV_CASES)
def test_conv_fwd_bwd(ext, nb, c, h, w, ko, r, st, pd):
    torch.manual_seed(2)
    x = torch.randn(nb, c, h, w, device=DEV)
    wt = torch.randn(ko, c, r, r, device=DEV) * 0.1
    b = torch.randn(ko, device=DEV)
    y = ext.conv2d_fwd(x, wt, b, st, pd, False)
    xg = x.clone().requires_grad_(True)
    wg = wt.clone().requires_grad_(True)
    bg = b.clone().requires_grad_(True)
    ref = torch.nn.functional.conv2d(xg, wg, bg, stride=st, padding=pd)
    t_allclose(y, ref.detach(), rtol=1e-4, atol=1e-3,
               msg=f'conv fwd {c}->{ko} s{st}p{pd}')
    dy = torch.randn_like(ref)
    ref.backward(dy)
    dx, dw, db = ext.conv2d_bwd(x, wt, dy, st, pd, True, True)
    t_allclose(dx, xg.grad, rtol=1e-4, atol=1e-3, msg='conv dx')
    t_allclose(dw, wg.grad, rtol=1e-4, atol=1e-2, msg='conv dw')
    t_allclose(db, bg.grad, rtol=1e-4, atol=1e-2, msg='conv db')


def test_conv_fused_relu(ext):
    x = torch.randn(4, 8, 14, 14, device=DEV)
    w = torch.randn(16, 8, 3, 3, device=DEV) * 0.1
    b = torch.zeros(16, device=DEV)
    y = ext.conv2d_fwd(x, w, b, 1, 0, True)
    ref = torch.relu(torch.nn.functional.conv2d(x, w, b))
    t_allclose(y, ref, rtol=1e-4, atol=1e-3, msg='conv+relu')


# ------------------------------------------------------------- batchnorm

def test_batchnorm_train_fwd_bwd(ext):
    torch.manual_seed(4)
    nb, c, h, w = 16, 32, 8, 8
    x = torch.randn(nb, c, h, w, device=DEV)
    wt = torch.rand(c, device=DEV) + 0.5
    b = torch.randn(c, device=DEV)
    rm = torch.zeros(c, device=DEV)
    rv = torch.ones(c, device=DEV)
    rm2, rv2 = rm.clone(), rv.clone()
    y, sm, sr = ext.batchnorm_fwd(x, wt, b, rm, rv, 0.1, 1e-5, True,
                                  False)

    xg = x.clone().requires_grad_(True)
    wg = wt.clone().requires_grad_(True)
    bg = b.clone().requires_grad_(True)
    ref = torch.nn.functional.batch_norm(xg, rm2, rv2, wg, bg, True, 0.1,
                                         1e-5)
    t_allclose(y, ref.detach(), rtol=1e-4, atol=1e-4, msg='bn fwd')
    t_allclose(rm, rm2, rtol=1e-5, msg='bn running mean')
    t_allclose(rv, rv2, rtol=1e-4, msg='bn running var')
    dy = torch.randn_like(ref)
    ref.backward(dy)
    dx, dw, db = ext.batchnorm_bwd(x, wt, sm, sr, dy)
    t_allclose(dx, xg.grad, rtol=1e-3, atol=1e-4, msg='bn dx')
    t_allclose(dw, wg.grad, rtol=1e-3, atol=1e-3, msg='bn dw')
    t_allclose(db, bg.grad, rtol=1e-3, atol=1e-3, msg='bn db')


def test_batchnorm_eval(ext):
    nb, c = 8, 16
    x = torch.randn(nb, c, 4, 4, device=DEV)
    wt = torch.rand(c, device=DEV) + 0.5
    b = torch.randn(c, device=DEV)
    rm = torch.randn(c, device=DEV)
    rv = torch.rand(c, device=DEV) + 0.5
    y, _, _ = ext.batchnorm_fwd(x, wt, b, rm.clone(), rv.clone(), 0.1, 1e-5,
                                False, False)
    ref = torch.nn.functional.batch_norm(x, rm, rv, wt, b, False, 0.1, 1e-5)
    t_allclose(y, ref, rtol=1e-4, atol=1e-4, msg='bn eval')


# ---------------------------------------------------------------- poison

def test_poison_kernels_match_cpu():
    from rlr_amd.data.poison import (apply_pattern_indexed_, pattern_spec)
    for data, pattern, dtype in [('fmnist', 'square', torch.uint8),
                                 ('fmnist', 'plus', torch.uint8),
                                 ('fmnist', 'copyright', torch.uint8),
                                 ('cifar10', 'plus', torch.uint8),
                                 ('fedemnist', 'square', torch.float32),
                                 ('fedemnist', 'copyright', torch.float32)]:
        for aidx in ([-1, 0, 1, 2, 3] if data == 'cifar10' else [-1]):
            spec = pattern_spec(data, pattern, aidx)
            if data == 'cifar10':
                shape = (20, 32, 32, 3)
            elif dtype == torch.uint8:
                shape = (20, 28, 28)
            else:
                shape = (20, 1, 28, 28)
            if dtype == torch.uint8:
                cpu = torch.randint(0, 256, shape, dtype=torch.uint8)
            else:
                cpu = torch.randn(shape)
            gpu = cpu.to(DEV)
            idx = torch.tensor([0, 3, 7, 19])
            apply_pattern_indexed_(cpu, idx, spec)
            apply_pattern_indexed_(gpu, idx.to(DEV), spec)
            assert torch.equal(gpu.cpu(), cpu), (data, pattern, aidx)


def test_normalize_u8_matches_cpu():
    from rlr_amd.data.datasets import ArrayDataset
    raw = torch.randint(0, 256, (64, 28, 28), dtype=torch.uint8)
    ds_cpu = ArrayDataset(raw, torch.zeros(64, dtype=torch.long), 'fmnist')
    ref = ds_cpu.normalize(raw)
    got = ds_cpu.normalize(raw.to(DEV)).cpu()
    t_allclose(got, ref, rtol=1e-6, atol=1e-6, msg='normalize')
    raw3 = torch.randint(0, 256, (16, 32, 32, 3), dtype=torch.uint8)
    ds3 = ArrayDataset(raw3, torch.zeros(16, dtype=torch.long), 'cifar10')
    t_allclose(ds3.normalize(raw3.to(DEV)).cpu(), ds3.normalize(raw3),
               rtol=1e-6, atol=1e-6, msg='normalize cifar')
