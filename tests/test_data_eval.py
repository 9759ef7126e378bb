"""Dataset generation, evaluation reductions, RNG streams, fedemnist e2e."""

import numpy as np
import torch

from rlr_amd.data import get_datasets, H5Dataset
from rlr_amd.options import default_args
from rlr_amd.utils.evaluation import get_loss_n_accuracy, materialize_eval_set
from rlr_amd.utils.rng import derive_seed, np_rng, sample_agents


def test_synthetic_shapes_and_balance(tiny_sizes):
    args = default_args(synthetic=True)
    train, val = get_datasets('fmnist', args)
    assert train.data.shape == (2000, 28, 28)
    assert train.data.dtype == torch.uint8
    assert (torch.bincount(train.targets, minlength=10) == 200).all()
    c_train, c_val = get_datasets('cifar10', args)
    assert c_train.data.shape == (2000, 32, 32, 3)


def test_synthetic_deterministic(tiny_sizes):
    args = default_args(synthetic=True)
    t1, _ = get_datasets('fmnist', args)
    t2, _ = get_datasets('fmnist', args)
    assert torch.equal(t1.data, t2.data)
    assert torch.equal(t1.targets, t2.targets)


def test_normalize_matches_torchvision_formula(tiny_sizes):
    args = default_args(synthetic=True)
    train, _ = get_datasets('fmnist', args)
    x = train.normalize(train.data[:4])
    ref = (train.data[:4].float() / 255.0 - 0.2860) / 0.3530
    assert torch.allclose(x.squeeze(1), ref, atol=1e-6)
    assert x.shape == (4, 1, 28, 28)


def test_fedemnist_noniid(tiny_sizes):
    args = default_args(synthetic=True, data='fedemnist', num_agents=12)
    users, val = get_datasets('fedemnist', args)
    assert len(users) == 12
    for u in users[:4]:
        classes = torch.unique(u.targets)
        assert 2 <= len(classes) <= 4  # non-IID: few classes per writer
    assert val.inputs.dtype == torch.float32


def test_h5dataset_parity():
    d = {7: {'label': [1, 2], 'pixels': np.zeros((2, 28, 28), np.float32)}}
    h = H5Dataset(d, 7)
    assert len(h) == 2
    assert h.inputs.shape == (2, 1, 28, 28)
    assert set(h.classes().tolist()) == {1, 2}


def test_eval_confusion_matches_manual(tiny_sizes):
    args = default_args(synthetic=True)
    train, val = get_datasets('fmnist', args)
    X, Y = materialize_eval_set(val, device='cpu')

    class Fixed(torch.nn.Module):
        def forward(self, x):
            torch.manual_seed(int(x.sum().abs() * 0) + x.shape[0])
            return torch.randn(x.shape[0], 10)

        def eval(self):
            return self

    model = Fixed()
    loss, (acc, per_class) = get_loss_n_accuracy(model, X, Y, args)
    # manual recompute
    torch.manual_seed(X.shape[0] % 251)
    total, correct = 0.0, 0
    conf = torch.zeros(10, 10)
    for lo in range(0, len(X), args.bs):
        out = model(X[lo:lo + args.bs])
        lab = Y[lo:lo + args.bs]
        total += torch.nn.functional.cross_entropy(
            out, lab, reduction='sum').item()
        pred = out.argmax(1)
        correct += (pred == lab).sum().item()
        for t, p in zip(lab, pred):
            conf[t, p] += 1
    # the model above is batch-size-seeded so both passes see the same
    # logits; compare aggregates
    assert abs(loss - total / len(X)) < 1e-4
    assert acc == correct / len(X)
    assert torch.allclose(per_class, conf.diag() / conf.sum(1), equal_nan=True)


def test_rng_streams_independent_and_stable():
    assert derive_seed(42, 'poison', 1) == derive_seed(42, 'poison', 1)
    assert derive_seed(42, 'poison', 1) != derive_seed(42, 'poison', 2)
    assert derive_seed(42, 'poison', 1) != derive_seed(42, 'shuffle', 1)
    assert derive_seed(43, 'poison', 1) != derive_seed(42, 'poison', 1)


def test_sample_agents_contract():
    s = sample_agents(42, 5, 40, 0.25)
    assert len(s) == 10
    assert len(set(s)) == 10          # no replacement
    assert all(0 <= a < 40 for a in s)
    assert s == sample_agents(42, 5, 40, 0.25)   # deterministic per round
    assert s != sample_agents(42, 6, 40, 0.25)   # varies across rounds


def test_real_data_fallback_warns(tmp_path):
    """Requesting real data with none on disk warns loudly before the
    synthetic substitution (never a silent swap — ADVICE r1 medium)."""
    import warnings
    from rlr_amd.data.datasets import get_datasets
    from rlr_amd.options import default_args
    args = default_args(synthetic=False, data='fmnist')
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter('always')
        get_datasets('fmnist', args, train_n=100, val_n=50,
                     data_dir=str(tmp_path / 'nope'))
    assert any('FALLING BACK TO SYNTHETIC' in str(x.message) for x in w)


def _write_idx(path, arr):
    import numpy as np
    arr = np.asarray(arr, dtype=np.uint8)
    with open(path, 'wb') as f:
        f.write((0x800 | arr.ndim).to_bytes(4, 'big'))
        for d in arr.shape:
            f.write(int(d).to_bytes(4, 'big'))
        f.write(arr.tobytes())


def test_load_real_fmnist_fixture(tmp_path):
    """Drive _load_real end-to-end against a tiny on-disk fixture in the
    torchvision FashionMNIST raw layout (idx-ubyte files)."""
    import numpy as np
    from rlr_amd.data.datasets import get_datasets
    from rlr_amd.options import default_args
    raw = tmp_path / 'FashionMNIST' / 'raw'
    raw.mkdir(parents=True)
    rng = np.random.default_rng(0)
    tr_x = rng.integers(0, 256, (40, 28, 28)).astype('uint8')
    tr_y = (np.arange(40) % 10).astype('uint8')
    va_x = rng.integers(0, 256, (20, 28, 28)).astype('uint8')
    va_y = (np.arange(20) % 10).astype('uint8')
    _write_idx(raw / 'train-images-idx3-ubyte', tr_x)
    _write_idx(raw / 'train-labels-idx1-ubyte', tr_y)
    _write_idx(raw / 't10k-images-idx3-ubyte', va_x)
    _write_idx(raw / 't10k-labels-idx1-ubyte', va_y)

    args = default_args(synthetic=False, data='fmnist')
    train, val = get_datasets('fmnist', args, data_dir=str(tmp_path))
    assert len(train) == 40 and len(val) == 20
    assert (train.data.numpy() == tr_x).all()
    assert train.targets.tolist() == tr_y.tolist()
    x, t = train[3]
    assert x.shape == (1, 28, 28) and t == 3
    # gz-only files load too (torchvision sometimes leaves only .gz)
    import gzip, os
    plain = raw / 't10k-images-idx3-ubyte'
    data = plain.read_bytes()
    os.remove(plain)
    with gzip.open(str(plain) + '.gz', 'wb') as f:
        f.write(data)
    _, val2 = get_datasets('fmnist', args, data_dir=str(tmp_path))
    assert (val2.data.numpy() == va_x).all()


def test_load_real_cifar10_fixture(tmp_path):
    """CIFAR-10 pickled-batch layout fixture through _load_real."""
    import numpy as np, pickle
    from rlr_amd.data.datasets import get_datasets
    from rlr_amd.options import default_args
    base = tmp_path / 'cifar-10-batches-py'
    base.mkdir(parents=True)
    rng = np.random.default_rng(1)

    def write(name, n):
        x = rng.integers(0, 256, (n, 3072)).astype('uint8')
        y = (np.arange(n) % 10).tolist()
        with open(base / name, 'wb') as f:
            pickle.dump({b'data': x, b'labels': y}, f)
        return x, y

    tx1, ty1 = write('data_batch_1', 30)
    tx2, ty2 = write('data_batch_2', 30)
    vx, vy = write('test_batch', 20)

    args = default_args(synthetic=False, data='cifar10')
    train, val = get_datasets('cifar10', args, data_dir=str(tmp_path))
    assert len(train) == 60 and len(val) == 20
    assert train.data.shape == (60, 32, 32, 3)   # HWC raw storage
    want = np.concatenate([tx1, tx2]).reshape(-1, 3, 32, 32).transpose(0, 2, 3, 1)
    assert (train.data.numpy() == want).all()
    assert train.targets.tolist() == ty1 + ty2
    x, t = val[5]
    assert x.shape == (3, 32, 32) and t == vy[5]
