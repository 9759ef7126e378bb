"""IID data partitioner + index-view dataset (reference utils.py:39-92).

`distribute_data` reproduces the reference's shard math exactly: sort by
label, bucket per class, strided-chunk each class list into slice_size
shards of shard_size = len/(num_agents*class_per_agent), deal one shard per
class to each agent (utils.py:58-92).  FMNIST/10 agents -> 6000/agent
(600/class); CIFAR/40 agents -> 1250/agent (125/class) — asserted in tests.
"""

from collections import defaultdict

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
    if args.num_agents == 1:
        return {0: list(range(len(dataset)))}

    def chunker_list(seq, size):
        return [seq[i::size] for i in range(size)]

    labels_sorted = dataset.targets.sort()
    class_by_labels = list(zip(labels_sorted.values.tolist(),
                               labels_sorted.indices.tolist()))
    labels_dict = defaultdict(list)
    for label, idx in class_by_labels:
        labels_dict[label].append(idx)

    shard_size = len(dataset) // (args.num_agents * class_per_agent)
    slice_size = (len(dataset) // n_classes) // shard_size
    for k, v in labels_dict.items():
        labels_dict[k] = chunker_list(v, slice_size)

    dict_users = defaultdict(list)
    for user_idx in range(args.num_agents):
        class_ctr = 0
        for j in range(n_classes):
            if class_ctr == class_per_agent:
                break
            elif len(labels_dict[j]) > 0:
                dict_users[user_idx] += labels_dict[j][0]
                del labels_dict[j % n_classes][0]
                class_ctr += 1
    return dict(dict_users)
