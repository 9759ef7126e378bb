"""Aggregation server (reference src/aggregation.py).

The server consumes a stacked (S, n_params) fp64 update matrix in sampled
order — on multi-GPU runs this is the RCCL all-gather result, identical on
every rank, and every rank applies the aggregate redundantly (deterministic,
zero extra traffic; SURVEY.md §2c call-site 3).  The dict-based reference
API (aggregation.py:19) is accepted too.

Rules (each a HIP kernel on the GPU path, ops/csrc/aggregation.hip):
  * RLR sign-vote (aggregation.py:48-54): sm = |sum_k sign(U_k)| per coord;
    lr = +server_lr where sm >= theta else -server_lr.  (The reference's
    in-place two-pass masking is only correct because -server_lr < theta
    for positive server_lr; this implementation is the explicit select and
    asserts that precondition — SURVEY.md §7 quirks.)
  * FedAvg (aggregation.py:57-64): sum_k n_k U_k / sum_k n_k, fp64, summed
    in sampled order for world-size-invariant bitwise results.
  * coordinate median (aggregation.py:66-69): torch.median column semantics
    (lower of the two middle values for even K).
  * sign (aggregation.py:71-75): sign(sum_k sign(U_k)).
  * optional N(0, noise*clip) gaussian noise (aggregation.py:34-35) from a
    per-round derived stream.  NOTE: the fused GPU avg path draws noise
    from the on-device philox Box-Muller stream, while the CPU and
    comed/sign paths draw from a torch CPU generator — the same
    seed+config yields a DIFFERENT (but equally deterministic) noise
    realization across the two backends.  Each backend is reproducible
    and world-size-invariant on its own; see PARITY.md.
  * fused apply (aggregation.py:38-40): theta <- float32(theta + lr * agg).
"""

import torch
from torch.nn import functional as F

from .ops import ext, force_eager
from .utils.rng import torch_gen


def _gpu(t):
    return t.is_cuda and not force_eager()


class Aggregation:
    def __init__(self, agent_data_sizes, n_params, poisoned_val_loader,
                 args, writer=None):
        self.agent_data_sizes = agent_data_sizes
        self.args = args
        self.writer = writer
        self.server_lr = args.server_lr
        self.n_params = n_params
        self.poisoned_val_loader = poisoned_val_loader
        self.cum_net_mov = 0.0

    # ------------------------------------------------------------- entry

    def aggregate_updates(self, global_model, agent_updates, cur_round,
                          agent_ids=None):
        """agent_updates: dict {agent_id: fp64 vec} (reference API) or a
        stacked (S, n) fp64 tensor with agent_ids giving the sampled order."""
        if isinstance(agent_updates, dict):
            agent_ids = list(agent_updates.keys())
            stacked = torch.stack([agent_updates[i] for i in agent_ids])
        else:
            stacked = agent_updates
            assert agent_ids is not None

        if _gpu(stacked) and self.args.aggr == 'avg':
            # headline path: ONE fused kernel does sign-vote + weighted avg
            # + noise + fp32 apply in a single pass over the K x n matrix
            from .utils.rng import derive_seed
            w = self._weights(agent_ids, stacked.device)
            noise_std = (self.args.noise * self.args.clip
                         if self.args.noise > 0 else 0.0)
            seed = derive_seed(self.args.seed, 'noise', cur_round)
            ext().fused_avg_rlr_apply(
                stacked, w, global_model.flat_params,
                self.args.robustLR_threshold > 0,
                float(self.args.robustLR_threshold), float(self.server_lr),
                float(noise_std), seed, 0, False)
            return

        lr_vector = None
        if self.args.robustLR_threshold > 0:
            lr_vector = self.compute_robustLR(stacked)

        if self.args.aggr == 'avg':
            agg = self.agg_avg(stacked, agent_ids)
        elif self.args.aggr == 'comed':
            agg = self.agg_comed(stacked)
        elif self.args.aggr == 'sign':
            agg = self.agg_sign(stacked)
        else:
            raise ValueError(self.args.aggr)

        if self.args.noise > 0:
            agg = agg + self._noise(cur_round, agg.device, agg.dtype)

        self._apply(global_model, lr_vector, agg)

    # ------------------------------------------------------------- rules

    def compute_robustLR(self, stacked: torch.Tensor) -> torch.Tensor:
        assert self.server_lr > 0 and self.args.robustLR_threshold > 0
        if _gpu(stacked):
            return ext().rlr_vote(stacked, float(self.args.robustLR_threshold),
                                  float(self.server_lr))
        sm = torch.abs(torch.sign(stacked).sum(dim=0))
        return torch.where(sm >= self.args.robustLR_threshold,
                           torch.full_like(sm, self.server_lr),
                           torch.full_like(sm, -self.server_lr))

    def _weights(self, agent_ids, device):
        w = torch.tensor([float(self.agent_data_sizes[i]) for i in agent_ids],
                         dtype=torch.float64, device=device)
        return w

    def agg_avg(self, stacked, agent_ids):
        w = self._weights(agent_ids, stacked.device)
        if _gpu(stacked):
            return ext().agg_avg(stacked, w)
        # sequential accumulation in sampled order (ws-invariant)
        out = torch.zeros(stacked.shape[1], dtype=torch.float64,
                          device=stacked.device)
        for k in range(stacked.shape[0]):
            out += w[k] * stacked[k]
        return out / w.sum()

    def agg_comed(self, stacked):
        # the register-select kernel holds one column of K values per lane;
        # beyond K=64 fall back to torch.median on device (same semantics)
        if _gpu(stacked) and stacked.shape[0] <= 64:
            return ext().agg_comed(stacked)
        return torch.median(stacked, dim=0).values

    def agg_sign(self, stacked):
        if _gpu(stacked):
            return ext().agg_sign(stacked)
        return torch.sign(torch.sign(stacked).sum(dim=0))

    def _noise(self, cur_round, device, dtype):
        g = torch_gen(self.args.seed, 'noise', cur_round)
        n = torch.normal(mean=0.0, std=self.args.noise * self.args.clip,
                         size=(self.n_params,), generator=g)
        return n.to(device=device, dtype=dtype)

    def _apply(self, global_model, lr_vector, agg):
        """theta <- float32(theta + lr * agg) (reference aggregation.py:38-40)."""
        p = global_model.flat_params
        if _gpu(p):
            ext().apply_update(p, agg,
                               lr_vector if lr_vector is not None
                               else torch.empty(0, device=p.device),
                               float(self.server_lr))
            return
        if lr_vector is None:
            new = p.double() + self.server_lr * agg
        else:
            new = p.double() + lr_vector.double() * agg
        p.copy_(new.float())

    def aggregate_buffers(self, global_model, buffer_deltas, agent_ids):
        """FedAvg-BN: data-weighted mean of BatchNorm running-stat deltas
        (build extension — the reference has no BN)."""
        if global_model.n_buffers == 0 or buffer_deltas is None:
            return
        if isinstance(buffer_deltas, (list, tuple)) and not buffer_deltas:
            return
        w = self._weights(agent_ids, buffer_deltas.device
                          if isinstance(buffer_deltas, torch.Tensor)
                          else buffer_deltas[0].device)
        if isinstance(buffer_deltas, list):
            buffer_deltas = torch.stack(buffer_deltas)
        if _gpu(buffer_deltas):
            # same weighted-mean HIP kernel as the parameter path (the
            # buffer matrix is tiny — K x ~10k — so the fp64 round-trip
            # is free and keeps one aggregation code path)
            mean = ext().agg_avg(buffer_deltas.double().contiguous(), w)
        else:
            mean = (buffer_deltas.double()
                    * (w / w.sum()).unsqueeze(1)).sum(dim=0)
        global_model.flat_buffers.add_(mean.to(global_model.flat_buffers.dtype))

    # ------------------------------------------------- server-side extras

    def clip_updates(self, agent_updates_dict):
        """Server-side L2 projection (reference aggregation.py:77-81;
        disconnected from the main path there and here — the client does it
        per batch, agent.py:54-60)."""
        for update in agent_updates_dict.values():
            l2 = torch.norm(update, p=2)
            update.div_(max(1, l2 / self.args.clip))

    def plot_norms(self, agent_updates_dict, cur_round, norm=2):
        """Honest vs corrupt update-norm scalars (reference
        aggregation.py:83-100)."""
        honest, corrupt = [], []
        for key, upd in agent_updates_dict.items():
            (corrupt if key < self.args.num_corrupt else honest).append(upd)
        if honest and self.writer:
            avg = sum(torch.norm(u, p=norm) for u in honest) / len(honest)
            self.writer.add_scalar(f'Norms/Avg_Honest_L{norm}', avg, cur_round)
        if corrupt and self.writer:
            avg = sum(torch.norm(u, p=norm) for u in corrupt) / len(corrupt)
            self.writer.add_scalar(f'Norms/Avg_Corrupt_L{norm}', avg, cur_round)

    # -------------------------------------------- Fisher diagnostics

    def comp_diag_fisher(self, model_params, poisoned_eval, adv=True):
        """Diagonal Fisher over the poisoned validation tensors — the
        per-parameter squared-gradient of the summed target logit
        (reference aggregation.py:102-129 semantics, tensor-resident and
        on-device; the reference's probe model never leaves the CPU,
        SURVEY.md §7 quirks).  adv=True scores against the TRUE labels,
        adv=False against the attacker's base class.  The reference
        gathers raw LOGITS (its log_softmax result is discarded,
        aggregation.py:123) — behavior kept."""
        from . import models as M
        from .flatmodel import FlatParamModel
        X, Y = poisoned_eval
        probe = FlatParamModel(
            M.get_model(self.args.data, getattr(self.args, 'model', None)),
            X.device)
        probe.load_vector(model_params.float())
        probe.eval()
        fisher = torch.zeros_like(probe.flat_params)
        n = X.shape[0]
        for lo in range(0, n, self.args.bs):
            xb = X[lo:lo + self.args.bs]
            yb = (Y[lo:lo + self.args.bs] if adv
                  else torch.full_like(Y[lo:lo + self.args.bs],
                                       self.args.base_class))
            probe.zero_grad()
            logits = probe(xb)
            F.log_softmax(logits, dim=1)  # parity: computed, unused
            logits.gather(1, yb.view(-1, 1)).sum().backward()
            fisher += probe.flat_grads.square() / n
        return fisher.detach()

    def plot_sign_agreement(self, robustLR, cur_global_params,
                            new_global_params, cur_round):
        """Sign-agreement diagnostics (reference aggregation.py:132-191
        scalar names/semantics): how much of this round's movement landed
        on coordinates the attacker's Fisher ranks as important but the
        honest Fisher does not — split by whether the RLR vote pushed the
        coordinate forward (+server_lr) or flipped it (-server_lr)."""
        update = new_global_params - cur_global_params
        n_par = update.numel()

        def top_mask(fisher):
            m = torch.zeros(n_par, dtype=torch.bool, device=update.device)
            m[fisher.topk(self.args.top_frac).indices] = True
            return m

        adv_top = top_mask(self.comp_diag_fisher(cur_global_params,
                                                 self.poisoned_val_loader))
        hon_top = top_mask(self.comp_diag_fisher(cur_global_params,
                                                 self.poisoned_val_loader,
                                                 adv=False))
        fwd = robustLR.view(-1) == self.server_lr    # vote kept direction
        flipped = robustLR.view(-1) == -self.server_lr

        def l2_on(mask):
            return update[mask].norm().item()

        scalars = {
            'Sign/Adv_Maxim_L2': l2_on(fwd & adv_top & ~hon_top),
            'Sign/Hon_Maxim_L2': l2_on(fwd & hon_top & ~adv_top),
            'Sign/Adv_Minim_L2': l2_on(flipped & adv_top & ~hon_top),
            'Sign/Hon_Minim_L2': l2_on(flipped & hon_top & ~adv_top),
        }
        scalars['Sign/Adv_Net_L2'] = (scalars['Sign/Adv_Maxim_L2']
                                      - scalars['Sign/Adv_Minim_L2'])
        scalars['Sign/Hon_Net_L2'] = (scalars['Sign/Hon_Maxim_L2']
                                      - scalars['Sign/Hon_Minim_L2'])
        self.cum_net_mov += (scalars['Sign/Hon_Net_L2']
                             - scalars['Sign/Adv_Net_L2'])
        scalars['Sign/Model_Net_L2_Cumulative'] = self.cum_net_mov
        if self.writer:
            for name, v in scalars.items():
                self.writer.add_scalar(name, v, cur_round)
        return scalars
