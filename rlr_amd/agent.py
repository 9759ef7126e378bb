"""Client runtime (reference src/agent.py:10-64).

Behavior-compatible semantics, MI355X-native execution:
  * the agent's full (possibly trojaned) local dataset is normalized once
    and kept resident in HBM as one float tensor — no DataLoader, no
    per-batch H2D copies on the hot path (288 GB per GPU holds every
    agent's shard; reference copies each batch host->device, agent.py:43).
  * local_train mirrors agent.py:33-64: fresh momentum each round
    (agent.py:37), per-batch grad-norm clip 10 + SGD step (fused kernel),
    per-BATCH PGD projection when clip>0 (agent.py:53-60 — inside the batch
    loop, not per-epoch), fp64 flat update return (agent.py:62-64).
  * all RNG (poison-index choice, epoch shuffling, dropout) comes from
    per-(agent, round, ...) derived streams, so results are independent of
    which rank trains the agent (world-size invariance)."""

import numpy as np
import torch

from .data import DatasetSplit, poison_dataset
from .ops import flat as flat_ops
from .ops import functional as Fo
from .utils.rng import derive_seed, np_rng


class Agent:
    def __init__(self, id, args, train_dataset=None, data_idxs=None,
                 user_dataset=None):
        self.id = id
        self.args = args
        self.device = torch.device(args.device)
        self._X = None  # lazily materialized device tensor (n,C,H,W) float
        self._Y = None

        if train_dataset is None:
            # fedemnist-style: agent owns a pre-built normalized shard
            # (reference agent.py:16-20 loads user_<id>_trainset.pt)
            assert user_dataset is not None
            self.train_dataset = user_dataset
            if self.id < args.num_corrupt:
                poison_dataset(self.train_dataset, args, data_idxs,
                               agent_idx=self.id)
        else:
            self.train_dataset = DatasetSplit(train_dataset, data_idxs)
            # corrupt agents poison their slice of the PARENT dataset's raw
            # storage (reference agent.py:22-25)
            if self.id < args.num_corrupt:
                poison_dataset(train_dataset, args, data_idxs,
                               agent_idx=self.id)
                # refresh the view's materialized labels
                self.train_dataset = DatasetSplit(train_dataset, data_idxs)

        self.n_data = len(self.train_dataset)

    # ---------------------------------------------------------- data path

    def _materialize(self):
        """Normalize the agent's shard once into a resident device tensor."""
        if self._X is not None:
            return
        ds = self.train_dataset
        if isinstance(ds, DatasetSplit):
            idx = torch.as_tensor(ds.idxs, device=ds.dataset.data.device)
            raw = ds.dataset.data[idx]
            self._X = ds.dataset.normalize(raw).to(self.device, non_blocking=True)
            self._Y = ds.dataset.targets[idx].to(self.device, non_blocking=True)
        else:  # TensorDataset (fedemnist): already normalized floats
            self._X = ds.inputs.to(self.device, non_blocking=True)
            self._Y = ds.targets.to(self.device, non_blocking=True)

    # ---------------------------------------------------------- hot loop

    def local_train(self, global_model, criterion=None, rnd=0):
        """Run local_ep epochs of clipped SGD from the current global
        parameters; return the fp64 flat update and RESTORE the global
        model (the reference driver restores it instead, federated.py:72).
        `criterion` is accepted for reference API parity; the loss is
        always mean-reduced cross-entropy (federated.py:61)."""
        args = self.args
        self._materialize()
        gm = global_model

        theta0 = gm.flat_params.clone()
        theta0_64 = theta0.double()
        buf0 = gm.flat_buffers.clone() if gm.n_buffers else None

        gm.train()
        gm.zero_momentum()  # fresh optimizer per round (reference agent.py:37)
        gm.set_dropout_seed(derive_seed(args.seed, 'dropout', self.id, rnd))

        n = self._X.shape[0]
        bs = args.bs
        use_graphs = (self.device.type == 'cuda'
                      and getattr(args, 'hip_graphs', True)
                      and not Fo.force_eager())
        engine = gm.get_engine(args) if use_graphs else None
        if engine is not None:
            engine.begin_round(theta0)
        else:
            gm.ensure_grad_views()
        for ep in range(args.local_ep):
            perm = np_rng(args.seed, 'shuffle', self.id, rnd, ep).permutation(n)
            perm_t = torch.as_tensor(perm, device=self.device)
            for lo in range(0, n, bs):
                sel = perm_t[lo:lo + bs]
                if engine is not None:
                    engine.step(self._X, self._Y, sel)
                    continue
                inputs, labels = self._X[sel], self._Y[sel]
                model = gm.model
                if (self.device.type == 'cuda' and not Fo.force_eager()
                        and getattr(model, 'manual_step', None) is not None
                        and (model.compute_dtype is None
                             or getattr(model, 'manual_bf16_ok', False))):
                    model.manual_step(inputs, labels, gm.dloss_ones())
                else:
                    gm.zero_grad()
                    outputs = gm(inputs)
                    loss = Fo.cross_entropy(outputs, labels)
                    loss.backward()
                flat_ops.clipped_sgd_step_(gm.flat_params, gm.flat_grads,
                                           gm.momentum, args.client_lr,
                                           args.client_moment, 10.0)
                if args.clip > 0:
                    flat_ops.pgd_project_(gm.flat_params, theta0, args.clip)
                gm.model.rng.advance_step()

        update = flat_ops.delta64(gm.flat_params, theta0_64)
        # restore the global replica for the next agent on this rank
        gm.flat_params.copy_(theta0)
        if buf0 is not None:
            self.buffer_delta = gm.flat_buffers - buf0
            gm.flat_buffers.copy_(buf0)
        return update
