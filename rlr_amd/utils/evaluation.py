"""Evaluation (reference utils.py:128-157).

Tensor-resident: the val set is one normalized device tensor; batches are
slices (no DataLoader, no H2D per batch).  The confusion matrix is built
with a device scatter-add (bincount of t*C+p) instead of the reference's
per-sample python loop (utils.py:151-152 — SURVEY.md §7 hard part 5);
on the GPU path the argmax/eq/confusion/loss reductions are one fused HIP
kernel per batch (ops/csrc: eval_update)."""

import torch

from ..ops import ext, force_eager


@torch.no_grad()
def get_loss_n_accuracy(model, X, Y, args, num_classes=10, bs=None,
                        shard=None):
    """Returns (avg_loss, (accuracy, per_class_accuracy)) — reference
    signature parity, tensor inputs.

    shard=(rank, world_size): batches are strided across ranks and the
    (loss_sum, confusion) accumulators all-reduced, so eval wall time is
    ~1/world_size (SURVEY.md §2c call-site 4).  Counts are exact integer
    sums — accuracy/confusion are bitwise world-size-invariant; loss_sum
    fp64 summation order differs across world sizes by at most an ulp."""
    model.eval()
    bs = bs or args.bs
    n = X.shape[0]
    device = X.device
    use_hip = X.is_cuda and not force_eager()

    starts = list(range(0, n, bs))
    if shard is not None and shard[1] > 1:
        starts = starts[shard[0]::shard[1]]

    conf = torch.zeros(num_classes * num_classes, dtype=torch.float32,
                       device=device)
    loss_sum = torch.zeros(1, dtype=torch.float64, device=device)
    for lo in starts:
        inputs, labels = X[lo:lo + bs], Y[lo:lo + bs]
        outputs = model(inputs)
        if outputs.dtype != torch.float32:
            outputs = outputs.float()
        if use_hip:
            ext().eval_update(outputs.contiguous(), labels, conf, loss_sum)
        else:
            loss_sum += torch.nn.functional.cross_entropy(
                outputs, labels, reduction='sum').double()
            pred = outputs.argmax(dim=1)
            conf += torch.bincount(labels * num_classes + pred,
                                   minlength=num_classes * num_classes
                                   ).float()
    if shard is not None and shard[1] > 1:
        from ..parallel import dist as pdist
        pdist.all_reduce_(conf)
        pdist.all_reduce_(loss_sum)
    conf = conf.view(num_classes, num_classes).cpu()
    avg_loss = (loss_sum.item() / n) if n else 0.0
    correct = conf.diag().sum().item()
    accuracy = correct / n if n else 0.0
    per_class = conf.diag() / conf.sum(1)
    return avg_loss, (accuracy, per_class)


def materialize_eval_set(dataset, idxs=None, device='cpu'):
    """Normalize a dataset (or an index subset) into resident device
    tensors (X, Y)."""
    if hasattr(dataset, 'inputs'):  # TensorDataset
        X, Y = dataset.inputs, dataset.targets
        if idxs is not None:
            sel = torch.as_tensor(list(idxs), device=X.device)
            X, Y = X[sel], Y[sel]
    else:
        if idxs is None:
            raw, Y = dataset.data, dataset.targets
        else:
            sel = torch.as_tensor(list(idxs), device=dataset.data.device)
            raw, Y = dataset.data[sel], dataset.targets[sel]
        X = dataset.normalize(raw)
    return X.to(device), Y.to(device)
