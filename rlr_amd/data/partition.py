"""IID data partitioner + index-view dataset (reference utils.py:39-92
semantics, rebuilt).

Partition model: the dataset is cut into `num_agents * class_per_agent`
equal shards per the reference's math — ``shard_size =
len(dataset) // (num_agents * class_per_agent)`` — built by slicing each
class's index pool into strided sub-sequences, then dealing one shard of
each class to every agent in class order.  FMNIST/10 agents -> 6000
samples/agent (600/class); CIFAR/40 agents -> 1250/agent (125/class) —
asserted in tests/test_partition.py.

Device invariance: class pools are built with `nonzero` on a CPU copy of
the labels (ascending index order, deterministic on every backend) rather
than sorting — CUDA sort tie-ordering differs from CPU's, which would make
partitions device-dependent.
"""

import torch


class DatasetSplit:
    """Index-list view over a parent dataset (reference utils.py:39-54).
    Materializes `.targets` so poison/eval code can see labels through the
    view."""

    def __init__(self, dataset, idxs):
        self.dataset = dataset
        self.idxs = list(idxs)
        sel = torch.as_tensor(self.idxs, dtype=torch.long,
                              device=dataset.targets.device)
        self.targets = dataset.targets[sel]

    def classes(self):
        return torch.unique(self.targets)

    def __len__(self):
        return len(self.idxs)

    def __getitem__(self, item):
        return self.dataset[self.idxs[item]]


def distribute_data(dataset, args, n_classes=10, class_per_agent=10):
    """Deal class-balanced shards to agents.  Returns {agent: [indices]}."""
    n_agents = args.num_agents
    if n_agents == 1:
        return {0: list(range(len(dataset)))}

    shard_size = len(dataset) // (n_agents * class_per_agent)
    shards_per_class = (len(dataset) // n_classes) // shard_size

    labels = dataset.targets.detach().cpu()
    # per-class shard queues: class pool (ascending indices) strided into
    # `shards_per_class` interleaved slices
    queues = []
    for c in range(n_classes):
        pool = (labels == c).nonzero(as_tuple=True)[0].tolist()
        queues.append([pool[s::shards_per_class]
                       for s in range(shards_per_class)])

    assignment = {}
    for agent in range(n_agents):
        mine, taken = [], 0
        for c in range(n_classes):
            if taken == class_per_agent:
                break
            if queues[c]:
                mine.extend(queues[c].pop(0))
                taken += 1
        assignment[agent] = mine
    return assignment
