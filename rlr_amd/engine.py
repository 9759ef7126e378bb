"""hipGraph-captured training step.

One FL step of the tiny reference CNNs is ~35-40 short kernels; eagerly
launched that is host-launch-bound (~3.5 us per launch — MI355X_MICROARCH
'graph-replay-floor').  The TrainEngine captures the whole step — the
model's hand-rolled manual tape (fwd + fused backward with grads written
straight into the flat views; models/*.manual_step) or the autograd
fallback, then the fused clip+SGD and optional PGD projection — into one
hipGraph per batch size and replays it (~10-16 us host cost).  Per step
only three cheap ops run outside the graph: two index_select gathers into
the static input buffers and the dropout-state advance (the philox
seed/offset pair lives in DEVICE memory precisely so replays draw fresh
masks — ops/functional.DropoutCtx).

The captured kernel stream is identical to the eager GPU path, so results
are bitwise equal with graphs on or off (asserted in test_e2e_gpu), and
the manual tape is bitwise-equal to autograd (test_manual_tape_gpu)."""

import torch

from .ops import flat as flat_ops
from .ops import functional as Fo


class TrainEngine:
    def __init__(self, gm, args):
        self.gm = gm
        self.args = args
        self.graphs = {}
        self.theta0_static = (torch.zeros_like(gm.flat_params)
                              if args.clip > 0 else None)
        self.pool = None

    # ------------------------------------------------------------ round

    def begin_round(self, theta0):
        if self.theta0_static is not None:
            self.theta0_static.copy_(theta0)

    # ------------------------------------------------------------- step

    def _step_body(self, static_x, static_y):
        # Preferred path: the model's hand-rolled fwd+bwd (manual_step)
        # writes each weight/bias grad DIRECTLY into its flat_grads view —
        # no zero-grad fill and no autograd accumulate-add kernels, with a
        # bitwise-identical kernel stream otherwise (assignment ==
        # accumulate-into-zero).  The autograd fallback below remains for
        # models without a manual tape (bf16 ResNet).
        gm, args = self.gm, self.args
        model = gm.model
        if (getattr(model, 'manual_step', None) is not None
                and (model.compute_dtype is None
                     or getattr(model, 'manual_bf16_ok', False))):
            model.manual_step(static_x, static_y, gm.dloss_ones())
        else:
            gm.flat_grads.zero_()
            out = gm(static_x)
            loss = Fo.cross_entropy(out, static_y)
            loss.backward()
        flat_ops.clipped_sgd_step_(gm.flat_params, gm.flat_grads,
                                   gm.momentum, args.client_lr,
                                   args.client_moment, 10.0)
        if args.clip > 0:
            flat_ops.pgd_project_(gm.flat_params, self.theta0_static,
                                  args.clip)

    def _build(self, bs, x_shape, device):
        gm = self.gm
        # graph capture must not race other streams' submissions (several
        # replicas train concurrently on their own streams): drain the
        # device first; capture itself is thread_local-error-mode
        torch.cuda.synchronize(device)
        static_x = torch.zeros((bs,) + tuple(x_shape), device=device)
        static_y = torch.zeros(bs, dtype=torch.long, device=device)

        # warmup + capture corrupt params/momentum/dropout state; snapshot
        # and restore so graph building is invisible to training
        snap_p = gm.flat_params.clone()
        snap_m = gm.momentum.clone()
        snap_b = gm.flat_buffers.clone() if gm.n_buffers else None
        rng = gm.model.rng
        snap_state = rng.gpu_state(device).clone()
        snap_site = rng.site

        gm.ensure_grad_views()
        side = torch.cuda.Stream(device=device)
        side.wait_stream(torch.cuda.current_stream(device))
        with torch.cuda.stream(side):
            for _ in range(2):
                self._step_body(static_x, static_y)
                rng.site = snap_site
        torch.cuda.current_stream(device).wait_stream(side)
        torch.cuda.synchronize(device)

        g = torch.cuda.CUDAGraph()
        # a GC cycle during capture frees pre-capture tensors -> hipFree
        # inside an active capture -> abort; collect first, then hold GC off
        import gc
        gc.collect()
        gc.disable()
        try:
            with torch.cuda.graph(g, pool=self.pool,
                                  capture_error_mode="thread_local"):
                self._step_body(static_x, static_y)
        finally:
            gc.enable()
        if self.pool is None:
            self.pool = g.pool()
        rng.site = snap_site

        gm.flat_params.copy_(snap_p)
        gm.momentum.copy_(snap_m)
        if snap_b is not None:
            gm.flat_buffers.copy_(snap_b)
        rng.gpu_state(device).copy_(snap_state)
        torch.cuda.synchronize(device)

        entry = (g, static_x, static_y)
        self.graphs[(bs,) + tuple(x_shape)] = entry
        return entry

    def step(self, X, Y, sel):
        bs = sel.numel()
        key = (bs,) + tuple(X.shape[1:])
        entry = self.graphs.get(key)
        if entry is None:
            entry = self._build(bs, X.shape[1:], X.device)
        g, static_x, static_y = entry
        torch.index_select(X, 0, sel, out=static_x)
        torch.index_select(Y, 0, sel, out=static_y)
        g.replay()
        self.gm.model.rng.advance_step()
