"""Partitioner shard math (reference utils.py:58-92): FMNIST/10 agents ->
6000 samples/agent (600/class); CIFAR/40 agents -> 1250/agent (125/class)."""

import torch

from rlr_amd.data import get_datasets, distribute_data, DatasetSplit
from rlr_amd.options import default_args


def test_fmnist_10_agents_full_size():
    args = default_args(num_agents=10, synthetic=True, data='fmnist')
    train, _ = get_datasets('fmnist', args)
    groups = distribute_data(train, args)
    assert len(groups) == 10
    for uid, idxs in groups.items():
        assert len(idxs) == 6000
        labels = train.targets[torch.as_tensor(idxs)]
        counts = torch.bincount(labels, minlength=10)
        assert (counts == 600).all()
    # no index appears twice
    all_idxs = [i for idxs in groups.values() for i in idxs]
    assert len(all_idxs) == len(set(all_idxs)) == 60000


def test_cifar_40_agents_full_size():
    args = default_args(num_agents=40, synthetic=True, data='cifar10')
    train, _ = get_datasets('cifar10', args)
    groups = distribute_data(train, args)
    assert len(groups) == 40
    for uid, idxs in groups.items():
        assert len(idxs) == 1250
        labels = train.targets[torch.as_tensor(idxs)]
        assert (torch.bincount(labels, minlength=10) == 125).all()


def test_single_agent_gets_everything(tiny_sizes):
    args = default_args(num_agents=1, synthetic=True)
    train, _ = get_datasets('fmnist', args)
    groups = distribute_data(train, args)
    assert list(groups[0]) == list(range(len(train)))


def test_dataset_split_view(tiny_sizes):
    args = default_args(synthetic=True)
    train, _ = get_datasets('fmnist', args)
    idxs = [3, 7, 11]
    view = DatasetSplit(train, idxs)
    assert len(view) == 3
    assert view.targets.tolist() == [int(train.targets[i]) for i in idxs]
    x, t = view[1]
    assert t == int(train.targets[7])
    assert x.shape == (1, 28, 28)
