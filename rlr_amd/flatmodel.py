"""FlatParamModel — a model whose parameters, grads and momentum live in
flat device buffers; module params/grads are views into them.

This is the MI355X-native replacement for the reference's
parameters_to_vector / vector_to_parameters dance (agent.py:35,56,60,63;
federated.py:59,66,72; aggregation.py:38-40): the flat buffer IS the
parameter vector, so snapshot/broadcast/restore are single contiguous
copies and the fused clip+SGD / PGD kernels (ops/flat.py) operate on one
tensor.

Floating-point buffers (BatchNorm running stats — only present in the
build's ResNet18 extension; the reference has no BN) are flattened into a
separate `flat_buffers` tensor so the FL round protocol can snapshot /
restore / average them alongside the parameters (FedAvg-BN convention).
Integer buffers (num_batches_tracked) stay module-local."""

import torch


class FlatParamModel:
    def __init__(self, model: torch.nn.Module, device):
        self.model = model.to(device)
        self.device = torch.device(device)

        params = [p for p in model.parameters() if p.requires_grad]
        self.n_params = sum(p.numel() for p in params)
        self.flat_params = torch.zeros(self.n_params, device=device)
        self.flat_grads = torch.zeros(self.n_params, device=device)
        self.momentum = torch.zeros(self.n_params, device=device)

        self.params = params
        self.param_offsets = []
        offset = 0
        for p in params:
            n = p.numel()
            self.param_offsets.append(offset)
            self.flat_params[offset:offset + n].copy_(p.data.reshape(-1))
            p.data = self.flat_params[offset:offset + n].view(p.shape)
            p.grad = self.flat_grads[offset:offset + n].view(p.shape)
            offset += n

        # float buffers (BN running stats)
        fbufs = [b for b in model.buffers() if b.dtype.is_floating_point]
        self.n_buffers = sum(b.numel() for b in fbufs)
        self.flat_buffers = torch.zeros(self.n_buffers, device=device)
        offset = 0
        for b in fbufs:
            n = b.numel()
            self.flat_buffers[offset:offset + n].copy_(b.data.reshape(-1))
            b.data = self.flat_buffers[offset:offset + n].view(b.shape)
            offset += n

    # -- vector API (parameters_to_vector parity) --
    def get_vector(self) -> torch.Tensor:
        """The live flat parameter vector (zero-copy)."""
        return self.flat_params

    def load_vector(self, vec: torch.Tensor):
        self.flat_params.copy_(vec)

    def snapshot(self):
        s = {'params': self.flat_params.clone()}
        if self.n_buffers:
            s['buffers'] = self.flat_buffers.clone()
        return s

    def restore(self, snap):
        self.flat_params.copy_(snap['params'])
        if self.n_buffers:
            self.flat_buffers.copy_(snap['buffers'])

    def zero_grad(self):
        self.flat_grads.zero_()

    def ensure_grad_views(self):
        """Re-point every p.grad at its flat_grads view (the graphed step
        detaches them so autograd ADOPTS fresh grad tensors instead of
        accumulating — engine.py gathers those into flat_grads in one
        kernel)."""
        for p, off in zip(self.params, self.param_offsets):
            p.grad = self.flat_grads[off:off + p.numel()].view(p.shape)

    def clear_grads(self):
        for p in self.params:
            p.grad = None

    def zero_momentum(self):
        self.momentum.zero_()

    # convenience passthroughs
    def train(self):
        self.model.train()

    def eval(self):
        self.model.eval()

    def __call__(self, x):
        return self.model(x)

    def set_dropout_seed(self, seed):
        self.model.set_dropout_seed(seed)

    def dloss_ones(self):
        """Static d(loss) seed tensor for the manual-tape backward."""
        if not hasattr(self, '_dloss'):
            self._dloss = torch.ones(1, device=self.device)
        return self._dloss

    def get_engine(self, args):
        """Cached hipGraph TrainEngine (engine.py) for this model."""
        if not hasattr(self, '_engine') or self._engine is None:
            from .engine import TrainEngine
            self._engine = TrainEngine(self, args)
        return self._engine
