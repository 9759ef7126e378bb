"""Dataset registry (reference: src/utils.py:95-124).

Synthetic-first: the north-star benchmark measures on synthetic data /
random-init weights (BASELINE.json), and this container has no network, so
`get_datasets` defaults to deterministic synthetic tensors *shaped exactly
like* the real datasets (fmnist 60k x 28x28 uint8, cifar10 50k x 32x32x3
uint8 HWC, fed-emnist 3383 non-IID user shards of normalized floats).
Synthetic images are class-prototype + noise so that both the classification
task and the trojan backdoor are actually learnable (the defense-efficacy
test depends on this).  When torchvision data is present on disk the real
datasets load through the same ArrayDataset container.

Raw storage conventions mirror the reference so the poison layer
(utils.py:160-284 semantics) operates pre-normalization on uint8 for
fmnist/cifar10 and on normalized floats for fedemnist.
"""

import os

import numpy as np
import torch

from ..utils.rng import np_rng

NORM_STATS = {
    # mean/std per channel (reference utils.py:100-120)
    'fmnist': ((0.2860,), (0.3530,)),
    'cifar10': ((0.4914, 0.4822, 0.4465), (0.2023, 0.1994, 0.2010)),
}


class ArrayDataset:
    """Torchvision-like container: raw uint8 `.data` (N,H,W) or (N,H,W,C),
    LongTensor `.targets`, normalization applied at read time — but also
    batch/whole-tensor materialization for the GPU-resident path."""

    def __init__(self, data: torch.Tensor, targets: torch.Tensor, kind: str):
        assert kind in NORM_STATS
        self.data = data            # uint8, HWC (cifar) or HW (fmnist)
        self.targets = targets.long()
        self.kind = kind
        mean, std = NORM_STATS[kind]
        self._mean = torch.tensor(mean).view(1, -1, 1, 1)
        self._std = torch.tensor(std).view(1, -1, 1, 1)

    def __len__(self):
        return self.data.shape[0]

    def normalize(self, raw_u8: torch.Tensor) -> torch.Tensor:
        """uint8 (B,H,W) or (B,H,W,C) -> normalized float32 NCHW.
        On CUDA(HIP) this is one fused HIP kernel (poison.hip
        normalize_u8): layout change + /255 + mean/std in a single pass."""
        if raw_u8.is_cuda:
            from ..ops import ext
            mean = self._mean.reshape(-1).to(raw_u8.device)
            std = self._std.reshape(-1).to(raw_u8.device)
            return ext().normalize_u8(raw_u8.contiguous(), mean, std)
        x = raw_u8.float().div_(255.0)
        if x.dim() == 3:            # (B,H,W) -> (B,1,H,W)
            x = x.unsqueeze(1)
        else:                       # (B,H,W,C) -> (B,C,H,W)
            x = x.permute(0, 3, 1, 2).contiguous()
        mean = self._mean.to(x.device)
        std = self._std.to(x.device)
        return x.sub_(mean).div_(std)

    def __getitem__(self, idx):
        x = self.normalize(self.data[idx:idx + 1])[0]
        return x, int(self.targets[idx])


class TensorDataset:
    """Fed-EMNIST-style container: already-normalized float `.inputs`
    (N,1,H,W) + `.targets` (reference stores these in per-user .pt files)."""

    def __init__(self, inputs: torch.Tensor, targets: torch.Tensor):
        self.inputs = inputs
        self.targets = targets.long()

    def __len__(self):
        return self.inputs.shape[0]

    def __getitem__(self, idx):
        return self.inputs[idx], int(self.targets[idx])


class H5Dataset(TensorDataset):
    """API-parity port of the reference's H5Dataset (utils.py:11-36):
    wraps a {client_id: {'label','pixels'}} dict, NCHW-reshaped."""

    def __init__(self, dataset, client_id):
        targets = torch.LongTensor(dataset[client_id]['label'])
        inputs = torch.Tensor(dataset[client_id]['pixels'])
        s = inputs.shape
        super().__init__(inputs.view(s[0], 1, s[1], s[2]), targets)

    def classes(self):
        return torch.unique(self.targets)

    def __add__(self, other):
        self.targets = torch.cat((self.targets, other.targets), 0)
        self.inputs = torch.cat((self.inputs, other.inputs), 0)
        return self

    def to(self, device):
        self.targets = self.targets.to(device)
        self.inputs = self.inputs.to(device)


# ---------------------------------------------------------------- synthetic

def _synth_images(rng, protos, n_per_class, n_classes, shape, noise=60.0):
    """Class prototype + per-sample brightness/contrast jitter + pixel
    noise.  The jitter matters: it gives honest agents genuinely diverse
    samples, so their per-coordinate gradient signs decorrelate the way
    real FMNIST's do — with pure prototype+noise data honest gradients
    align perfectly and the RLR vote semantics invert (the defense-efficacy
    test pins the intended dynamics)."""
    imgs = np.empty((n_per_class * n_classes,) + shape, dtype=np.uint8)
    targets = np.empty(n_per_class * n_classes, dtype=np.int64)
    bshape = (n_per_class,) + (1,) * len(shape)
    for c in range(n_classes):
        lo = c * n_per_class
        brightness = rng.normal(0.0, 40.0, size=bshape)
        contrast = rng.uniform(0.6, 1.4, size=bshape)
        x = ((protos[c][None] - 128.0) * contrast + 128.0 + brightness
             + rng.normal(0.0, noise, size=(n_per_class,) + shape))
        imgs[lo:lo + n_per_class] = np.clip(x, 0, 255).astype(np.uint8)
        targets[lo:lo + n_per_class] = c
    # deterministic interleave so classes are mixed
    perm = rng.permutation(len(targets))
    return imgs[perm], targets[perm]


def _synthetic_pair(kind, seed, train_n, val_n, n_classes=10):
    shape = (28, 28) if kind == 'fmnist' else (32, 32, 3)
    rng = np_rng(seed, 'data', 0 if kind == 'fmnist' else 1)
    # prototypes = shared base image + per-class deltas, shared by train AND
    # val: classes share most structure (like real image sets) but are
    # separably labeled
    base = rng.integers(60, 196, size=shape).astype(np.float64)
    deltas = rng.normal(0.0, 35.0, size=(n_classes,) + shape)
    protos = np.clip(base[None] + deltas, 0, 255)
    tr_img, tr_t = _synth_images(rng, protos, train_n // n_classes, n_classes, shape)
    va_img, va_t = _synth_images(rng, protos, val_n // n_classes, n_classes, shape)
    train = ArrayDataset(torch.from_numpy(tr_img), torch.from_numpy(tr_t), kind)
    val = ArrayDataset(torch.from_numpy(va_img), torch.from_numpy(va_t), kind)
    return train, val


def _synthetic_fedemnist(seed, num_users, samples_per_user, val_n, n_classes=10):
    """Non-IID writers: each user draws from 2-4 classes with its own style
    offset; inputs are already-normalized floats like the reference's
    pre-built .pt shards (utils.py:105-109).  Per-user sample counts are
    lognormal-skewed around `samples_per_user` (the real Fed-EMNIST has
    heavy per-writer size skew, mean ~341), floor 16 — so data-size FedAvg
    weights are genuinely heterogeneous."""
    mean, std = NORM_STATS['fmnist']
    rng = np_rng(seed, 'data', 2)
    protos = rng.normal(0.0, 1.0, size=(n_classes, 28, 28))
    users = []
    for u in range(num_users):
        urng = np_rng(seed, 'data', 3, u)
        # lognormal with sigma=0.5 has mean exp(mu + sigma^2/2); scale so
        # the expected count equals samples_per_user
        n_u = max(16, int(samples_per_user *
                          urng.lognormal(-0.125, 0.5)))
        k = int(urng.integers(2, 5))
        classes = urng.choice(n_classes, k, replace=False)
        t = urng.choice(classes, n_u)
        style = urng.normal(0.0, 0.3, size=(28, 28))
        x = protos[t] + style[None] + urng.normal(0.0, 0.5, size=(len(t), 28, 28))
        users.append(TensorDataset(
            torch.from_numpy(x.astype(np.float32)).view(-1, 1, 28, 28),
            torch.from_numpy(t.astype(np.int64))))
    vrng = np_rng(seed, 'data', 4)
    vt = vrng.integers(0, n_classes, size=val_n)
    vx = protos[vt] + vrng.normal(0.0, 0.5, size=(val_n, 28, 28))
    val = TensorDataset(torch.from_numpy(vx.astype(np.float32)).view(-1, 1, 28, 28),
                        torch.from_numpy(vt.astype(np.int64)))
    return users, val


# ----------------------------------------------------------------- registry

DEFAULT_SIZES = {
    'fmnist': (60000, 10000),
    'cifar10': (50000, 10000),
    'fedemnist': (3383, 341),  # users, mean samples/user (reference scale)
}


def get_datasets(data, args=None, train_n=None, val_n=None, data_dir='../data'):
    """Returns (train, val) — for fedemnist, (list-of-user TensorDatasets, val).

    Synthetic unless real torchvision data is on disk and args.synthetic is
    False (reference downloads at utils.py:100-122; this environment has no
    network, so synthetic is the operational default)."""
    seed = getattr(args, 'seed', 42) if args is not None else 42
    synthetic = True if args is None else bool(getattr(args, 'synthetic', True))

    if data in ('fmnist', 'cifar10'):
        if not synthetic:
            if _try_real_available(data, data_dir):
                return _load_real(data, data_dir)
            # never silently substitute synthetic for requested real data:
            # the metrics would be misread as real-dataset results
            import warnings
            warnings.warn(
                f"real {data} requested (--synthetic not set) but no dataset "
                f"found under {data_dir!r}; FALLING BACK TO SYNTHETIC DATA — "
                f"metrics are NOT real-{data} results. Pass --synthetic to "
                f"silence this, or place the dataset on disk.",
                stacklevel=2)
        tn, vn = DEFAULT_SIZES[data]
        return _synthetic_pair(data, seed, train_n or tn, val_n or vn)
    elif data == 'fedemnist':
        users, spu = DEFAULT_SIZES['fedemnist']
        if args is not None:
            users = getattr(args, 'num_agents', users) or users
        return _synthetic_fedemnist(seed, train_n or users, spu, val_n or 10000)
    raise ValueError(f"unknown dataset {data}")


def _try_real_available(data, data_dir):
    sub = {'fmnist': 'FashionMNIST', 'cifar10': 'cifar-10-batches-py'}[data]
    return os.path.isdir(os.path.join(data_dir, sub))


# Native on-disk readers (torchvision layouts, no torchvision dependency —
# it is not installed in the target image).  FashionMNIST ships as
# idx-ubyte files (possibly still .gz); CIFAR-10 as python-pickled batches.

def _read_idx(path):
    """Parse an idx1/idx3 ubyte file (the (Fashion)MNIST on-disk format)."""
    import gzip
    opener = gzip.open if not os.path.exists(path) and \
        os.path.exists(path + '.gz') else open
    if opener is gzip.open:
        path = path + '.gz'
    with opener(path, 'rb') as f:
        buf = f.read()
    magic = int.from_bytes(buf[0:4], 'big')
    ndim = magic & 0xFF
    dims = [int.from_bytes(buf[4 + 4 * i: 8 + 4 * i], 'big')
            for i in range(ndim)]
    data = np.frombuffer(buf, dtype=np.uint8, offset=4 + 4 * ndim)
    return torch.from_numpy(data.reshape(dims).copy())


def _load_fmnist_raw(data_dir):
    raw = os.path.join(data_dir, 'FashionMNIST', 'raw')
    out = []
    for split in ('train', 't10k'):
        imgs = _read_idx(os.path.join(raw, f'{split}-images-idx3-ubyte'))
        lbls = _read_idx(os.path.join(raw, f'{split}-labels-idx1-ubyte'))
        out.append(ArrayDataset(imgs, lbls.long(), 'fmnist'))
    return tuple(out)


def _load_cifar10_raw(data_dir):
    import pickle
    base = os.path.join(data_dir, 'cifar-10-batches-py')

    def batches(names):
        xs, ys = [], []
        for name in names:
            with open(os.path.join(base, name), 'rb') as f:
                d = pickle.load(f, encoding='bytes')
            xs.append(np.asarray(d[b'data'], dtype=np.uint8))
            ys.extend(d[b'labels'])
        x = np.concatenate(xs).reshape(-1, 3, 32, 32).transpose(0, 2, 3, 1)
        return (ArrayDataset(torch.from_numpy(np.ascontiguousarray(x)),
                             torch.tensor(ys), 'cifar10'))

    train = batches([f'data_batch_{i}' for i in range(1, 6)
                     if os.path.exists(os.path.join(base, f'data_batch_{i}'))])
    val = batches(['test_batch'])
    return train, val


def _load_real(data, data_dir):
    if data == 'fmnist':
        return _load_fmnist_raw(data_dir)
    return _load_cifar10_raw(data_dir)
