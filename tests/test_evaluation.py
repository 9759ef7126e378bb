"""Evaluation units: CPU confusion-matrix path vs a hand-rolled loop
(reference utils.py:134-156 semantics) and the checkpoint version guard."""

import pytest
import torch

from rlr_amd.flatmodel import FlatParamModel
from rlr_amd.models import get_model
from rlr_amd.options import default_args
from rlr_amd.utils.checkpoint import load_checkpoint, save_checkpoint
from rlr_amd.utils.evaluation import get_loss_n_accuracy


def test_confusion_matrix_matches_manual_loop():
    torch.manual_seed(0)
    fm = FlatParamModel(get_model('fmnist'), 'cpu')
    X = torch.randn(70, 1, 28, 28)
    Y = torch.randint(0, 10, (70,))
    args = default_args(bs=32)
    loss, (acc, per_class) = get_loss_n_accuracy(fm, X, Y, args)

    # hand-rolled reference (per-sample loop, utils.py:151-152)
    fm.eval()
    with torch.no_grad():
        out = fm(X)
    pred = out.argmax(1)
    conf = torch.zeros(10, 10)
    for t, p in zip(Y.tolist(), pred.tolist()):
        conf[t, p] += 1
    ref_loss = torch.nn.functional.cross_entropy(
        out, Y, reduction='sum').item() / 70
    assert acc == pytest.approx(conf.diag().sum().item() / 70)
    assert loss == pytest.approx(ref_loss, rel=1e-5)
    want = conf.diag() / conf.sum(1)
    got = per_class
    mask = ~torch.isnan(want)
    assert torch.allclose(got[mask], want[mask])


def test_eval_batch_split_invariant():
    """Metrics must not depend on the eval batch size."""
    torch.manual_seed(1)
    fm = FlatParamModel(get_model('fmnist'), 'cpu')
    X = torch.randn(50, 1, 28, 28)
    Y = torch.randint(0, 10, (50,))
    args = default_args()
    l1, (a1, _) = get_loss_n_accuracy(fm, X, Y, args, bs=7)
    l2, (a2, _) = get_loss_n_accuracy(fm, X, Y, args, bs=50)
    assert a1 == a2
    assert l1 == pytest.approx(l2, rel=1e-6)


def test_checkpoint_version_guard(tmp_path):
    fm = FlatParamModel(get_model('fmnist'), 'cpu')
    args = default_args()
    path = str(tmp_path / 'c.pt')
    save_checkpoint(path, fm, 3, args, 0.5)
    state = torch.load(path, weights_only=False)
    state['version'] = -1
    torch.save(state, path)
    with pytest.raises(AssertionError):
        load_checkpoint(path, fm)
