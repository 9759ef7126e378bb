"""Autograd-visible ops.

Every op has two implementations:
  * CUDA(HIP) tensors -> a hand-written CDNA4 kernel from ops/csrc via a
    torch.autograd.Function (required; loud failure if the extension is
    missing — see ops.__init__).
  * CPU tensors -> the plain torch.nn.functional op (GPU-less CI only).

The HIP forward kernels fuse the activation where the reference composes
relu(conv(...)) / relu(linear(...)) (reference models.py:22-28,47-56): the
backward recovers the relu mask from the *post*-activation output (y>0).

Dropout parity note: the reference uses nn.Dropout2d(p=0.5) on flattened
2-D activations (models.py:17-19,40-44), where each "channel" is a single
scalar — i.e. plain elementwise Bernoulli dropout.  This build implements
elementwise dropout with a counter-based philox stream so masks are
reproducible for any world size (SURVEY.md §7 hard part 3).
"""

import torch
import torch.nn.functional as F

from . import ext, force_eager


def _gpu(x: torch.Tensor) -> bool:
    return x.is_cuda and not force_eager()


class DropoutCtx:
    """Deterministic dropout stream.

    CPU path: (seed64, monotonically increasing call counter).
    GPU path: a 2 x u64 DEVICE buffer [seed, base_offset] + a per-forward
    site index; the kernel draws philox(seed, base+site, idx).  Keeping the
    state in device memory lets a hipGraph-captured training step advance
    its masks between replays (engine.py bumps base_offset by SITE_STRIDE
    after every step) without re-capture."""

    SITE_STRIDE = 16  # max dropout call sites per forward

    def __init__(self):
        self.seed = 0
        self.counter = 0        # CPU stream
        self.site = 0           # GPU per-forward site index
        self._state = None      # device u64[2]

    def reset(self, seed: int):
        self.seed = int(seed)
        self.counter = 0
        self.site = 0
        if self._state is not None:
            self._state.copy_(torch.tensor([self.seed, 0],
                                           dtype=torch.int64))

    def next_offset(self) -> int:
        c = self.counter
        self.counter += 1
        return c

    # ---- GPU/graph path ----
    def gpu_state(self, device) -> torch.Tensor:
        if self._state is None or self._state.device != torch.device(device):
            self._state = torch.tensor([self.seed, 0], dtype=torch.int64,
                                       device=device)
        return self._state

    def next_site(self) -> int:
        s = self.site
        self.site += 1
        assert self.site <= self.SITE_STRIDE
        return s

    def advance_step(self):
        """Call once after each training step (outside any graph)."""
        self.site = 0
        if self._state is not None:
            self._state[1] += self.SITE_STRIDE


# ----------------------------------------------------------------- conv2d

class _ConvReLU(torch.autograd.Function):
    """y = conv2d(x, w, stride, pad) + b, optionally relu-fused.
    The reference only uses no-pad stride-1 3x3 (models.py:14-15,36-38);
    stride/pad generality serves the build's ResNet18 extension."""

    @staticmethod
    def forward(ctx, x, w, b, stride, pad, relu):
        y = ext().conv2d_fwd(x, w, b, stride, pad, relu)
        ctx.save_for_backward(x, w, y)
        ctx.cfg = (stride, pad, relu, b is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        stride, pad, relu, has_b = ctx.cfg
        if relu:
            dy = ext().relu_bwd(y, dy)
        dx, dw, db = ext().conv2d_bwd(x, w, dy, stride, pad, has_b,
                                      ctx.needs_input_grad[0])
        return dx, dw, db, None, None, None


def conv2d(x, w, b, stride=1, pad=0, relu=False):
    if _gpu(x):
        return _ConvReLU.apply(x, w, b, stride, pad, relu)
    y = F.conv2d(x, w, b, stride=stride, padding=pad)
    return F.relu(y) if relu else y


# ----------------------------------------------------------------- linear

class _LinearReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, relu):
        y = ext().linear_fwd(x, w, b, relu)
        ctx.save_for_backward(x, w, y)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        if ctx.relu:
            dy = ext().relu_bwd(y, dy)
        dx, dw, db = ext().linear_bwd(x, w, dy)
        return dx, dw, db, None


def linear(x, w, b, relu=False):
    if _gpu(x):
        return _LinearReLU.apply(x, w, b, relu)
    y = F.linear(x, w, b)
    return F.relu(y) if relu else y


# ------------------------------------------------------------------- relu

class _ReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y = ext().relu_fwd(x)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        return ext().relu_bwd(y, dy)


def relu(x):
    if _gpu(x):
        return _ReLU.apply(x)
    return F.relu(x)


# ------------------------------------------------------------- max_pool2d

class _MaxPool2x2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, idx = ext().maxpool2x2_fwd(x)
        ctx.save_for_backward(idx)
        ctx.in_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        # no .contiguous() here: plain contiguous() would force NCHW and
        # the binding's cl() would copy straight back to channels_last
        return ext().maxpool2x2_bwd(dy, idx, list(ctx.in_shape))


def max_pool2d_2x2(x):
    """kernel 2, stride 2 — the reference's only pooling
    (models.py:16,39)."""
    if _gpu(x):
        return _MaxPool2x2.apply(x)
    return F.max_pool2d(x, 2, 2)


# ---------------------------------------------------------------- dropout

class _Dropout(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, state, site):
        y, mask = ext().dropout_fwd_dev(x, p, state, site)
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        return ext().dropout_bwd(dy.contiguous(), mask, ctx.p), None, None, None


def dropout(x, p, training, rng: DropoutCtx):
    if not training or p == 0:
        return x
    if _gpu(x):
        return _Dropout.apply(x, p, rng.gpu_state(x.device), rng.next_site())
    from ..utils.rng import derive_seed
    g = torch.Generator(device='cpu')
    g.manual_seed(derive_seed(rng.seed, 'dropout', rng.next_offset()))
    mask = (torch.rand(x.shape, generator=g) >= p).to(x.dtype).to(x.device)
    return x * mask / (1.0 - p)


# ---------------------------------------------------------------- flatten

class _Flatten(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.chw = (x.shape[1], x.shape[2], x.shape[3])
        return ext().nhwc_flatten(x)

    @staticmethod
    def backward(ctx, dy):
        c, h, w = ctx.chw
        return ext().nhwc_unflatten(dy.contiguous(), c, h, w)


def flatten2d(x):
    """(B,C,H,W) -> (B, C*H*W) flattened in the reference's CHW order
    (models.py:21,50); on the GPU path this is one permute kernel off the
    channels_last storage."""
    if _gpu(x) and x.dim() == 4:
        return _Flatten.apply(x)
    return x.reshape(x.shape[0], -1)


# ---------------------------------------------------------- cross entropy

class _CrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        loss, softmax = ext().cross_entropy_fwd(logits, labels)
        ctx.save_for_backward(softmax, labels)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        softmax, labels = ctx.saved_tensors
        return ext().cross_entropy_bwd(softmax, labels, dloss), None


def cross_entropy(logits, labels):
    """Mean-reduced CE (the reference's criterion, federated.py:61).
    bf16 logits are cast up (autograd casts the gradient back down)."""
    if _gpu(logits):
        if logits.dtype == torch.bfloat16:
            logits = logits.float()
        return _CrossEntropy.apply(logits, labels)
    return F.cross_entropy(logits, labels)


# ------------------------------------------------------------- batch norm

class _BatchNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, running_mean, running_var, momentum, eps,
                training):
        y, save_mean, save_rstd = ext().batchnorm_fwd(
            x, w, b, running_mean, running_var, momentum, eps, training,
            False)
        ctx.save_for_backward(x, w, save_mean, save_rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, save_mean, save_rstd = ctx.saved_tensors
        # dy is normalized to channels_last inside the binding; a plain
        # .contiguous() here would bounce it through NCHW (2 extra copies
        # per BN — rocprofv3 evidence in profiles/)
        dx, dw, db = ext().batchnorm_bwd(x, w, save_mean, save_rstd, dy)
        return dx, dw, db, None, None, None, None, None


def batch_norm(x, w, b, running_mean, running_var, momentum, eps, training):
    if _gpu(x):
        return _BatchNorm.apply(x, w, b, running_mean, running_var, momentum,
                                eps, training)
    return F.batch_norm(x, running_mean, running_var, w, b, training,
                        momentum, eps)


# -------------------------------------------------------- global avg pool

class _GlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.in_shape = x.shape
        return ext().gap_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        return ext().gap_bwd(dy.contiguous(), list(ctx.in_shape))


def global_avg_pool(x):
    """(N,C,H,W) -> (N,C) mean over HxW (build's ResNet18 head)."""
    if _gpu(x):
        return _GlobalAvgPool.apply(x)
    return x.mean(dim=(2, 3))


# -------------------------------------------------------------- add+relu

class _AddReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        y = ext().add_relu_fwd(a, b)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        d = ext().relu_bwd(y, dy)
        return d, d


def add_relu(a, b):
    """relu(a + b) — the residual-join fusion for ResNet blocks."""
    if _gpu(a):
        return _AddReLU.apply(a, b)
    return F.relu(a + b)
