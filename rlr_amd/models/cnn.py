"""The reference model zoo (reference src/models.py:11-58), rebuilt on the
rlr_amd op layer.  Parameter shapes, registration order and default init are
identical to the reference (nn.Conv2d / nn.Linear containers), so the flat
parameter vector is layout-compatible: CNN_MNIST = 1,199,882 params,
CNN_CIFAR = 537,610 params (asserted in tests/test_models.py).

forward() composes the fused conv+relu / linear+relu HIP ops; dropout uses
the deterministic philox stream (ops.functional.DropoutCtx)."""

import torch.nn as nn

from ..ops import functional as Fo


class _OpsModel(nn.Module):
    def __init__(self):
        super().__init__()
        self.rng = Fo.DropoutCtx()
        self.compute_dtype = None  # None = fp32; torch.bfloat16 for the
                                   # bf16 MFMA path (--dtype bf16, GPU only)

    def set_dropout_seed(self, seed: int):
        self.rng.reset(seed)

    def set_compute_dtype(self, dtype):
        self.compute_dtype = dtype

    def _cast_in(self, x):
        if self.compute_dtype is not None and x.dtype != self.compute_dtype:
            x = x.to(self.compute_dtype)
        return x


class CNN_MNIST(_OpsModel):
    """28x28x1: conv(1->32,3x3)+relu, conv(32->64,3x3)+relu, maxpool2,
    flatten 9216, dropout .5, fc 9216->128+relu, dropout, fc 128->10
    (reference models.py:11-31)."""

    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 32, kernel_size=(3, 3))
        self.conv2 = nn.Conv2d(32, 64, kernel_size=(3, 3))
        self.fc1 = nn.Linear(9216, 128)
        self.fc2 = nn.Linear(128, 10)
        self.p_drop = 0.5

    def forward(self, x):
        x = self._cast_in(x)
        x = Fo.conv2d(x, self.conv1.weight, self.conv1.bias, relu=True)
        x = Fo.conv2d(x, self.conv2.weight, self.conv2.bias, relu=True)
        x = Fo.max_pool2d_2x2(x)
        x = Fo.flatten2d(x)
        x = Fo.dropout(x, self.p_drop, self.training, self.rng)
        x = Fo.linear(x, self.fc1.weight, self.fc1.bias, relu=True)
        x = Fo.dropout(x, self.p_drop, self.training, self.rng)
        x = Fo.linear(x, self.fc2.weight, self.fc2.bias)
        return x

    def manual_step(self, x, labels, dloss):
        """Hand-rolled fwd+bwd (fp32 GPU): identical kernel sequence to
        the autograd path, but weight/bias grads are written DIRECTLY into
        the preset p.grad flat-views (assignment == accumulate-into-zero,
        bitwise) — no zero-grad fill, no autograd accumulate-adds."""
        from ..ops import ext
        E = ext()
        p = self.p_drop
        st = self.rng.gpu_state(x.device)
        a1 = E.conv2d_fwd(x, self.conv1.weight, self.conv1.bias, 1, 0, True)
        a2 = E.conv2d_fwd(a1, self.conv2.weight, self.conv2.bias, 1, 0, True)
        pl, idx = E.maxpool2x2_fwd(a2)
        fl = E.nhwc_flatten(pl)
        d1, m1 = E.dropout_fwd_dev(fl, p, st, 0)
        h1 = E.linear_fwd(d1, self.fc1.weight, self.fc1.bias, True)
        d2, m2 = E.dropout_fwd_dev(h1, p, st, 1)
        out = E.linear_fwd(d2, self.fc2.weight, self.fc2.bias, False)
        loss, softmax = E.cross_entropy_fwd(out, labels)

        # Backward with the relu masks FUSED away (each bitwise-identical
        # to the separate relu_bwd: see maxpool2x2_bwd_gather4_k /
        # dropout_relu_bwd_k / conv_bwd_data relu_y notes in ops/csrc):
        #   * fc1's relu folds into the dropout-2 backward;
        #   * conv2's relu folds into the maxpool gather via the pooled
        #     values (window max == pooled value for post-relu inputs);
        #   * conv1's relu folds into conv2's bwd-data epilogue.
        g = E.cross_entropy_bwd(softmax, labels, dloss)
        g = E.linear_bwd_into(d2, self.fc2.weight, g,
                              self.fc2.weight.grad, self.fc2.bias.grad, True)
        g = E.dropout_relu_bwd(g, m2, h1, p)
        g = E.linear_bwd_into(d1, self.fc1.weight, g,
                              self.fc1.weight.grad, self.fc1.bias.grad, True)
        g = E.dropout_bwd(g, m1, p)
        g = E.nhwc_unflatten(g, pl.shape[1], pl.shape[2], pl.shape[3])
        g = E.maxpool2x2_bwd_relu(g, idx, pl, list(a2.shape))
        g = E.conv2d_bwd_into(a1, self.conv2.weight, g, 1, 0, True,
                              self.conv2.weight.grad, self.conv2.bias.grad,
                              a1)
        E.conv2d_bwd_into(x, self.conv1.weight, g, 1, 0, False,
                          self.conv1.weight.grad, self.conv1.bias.grad,
                          None)
        return loss


class CNN_CIFAR(_OpsModel):
    """32x32x3: 3 x [conv3x3 (3->64->128->256)+relu+maxpool2], flatten 1024,
    dropout-interleaved FCs 1024->128->256->10 (reference models.py:33-58;
    the reference's `64*4*4` flatten literal equals 256*2*2 = 1024)."""

    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 3)
        self.conv2 = nn.Conv2d(64, 128, 3)
        self.conv3 = nn.Conv2d(128, 256, 3)
        self.fc1 = nn.Linear(1024, 128)
        self.fc2 = nn.Linear(128, 256)
        self.fc3 = nn.Linear(256, 10)
        self.p_drop = 0.5

    def forward(self, x):
        x = self._cast_in(x)
        x = Fo.max_pool2d_2x2(Fo.conv2d(x, self.conv1.weight, self.conv1.bias, relu=True))
        x = Fo.max_pool2d_2x2(Fo.conv2d(x, self.conv2.weight, self.conv2.bias, relu=True))
        x = Fo.max_pool2d_2x2(Fo.conv2d(x, self.conv3.weight, self.conv3.bias, relu=True))
        x = Fo.flatten2d(x)
        x = Fo.dropout(x, self.p_drop, self.training, self.rng)
        x = Fo.linear(x, self.fc1.weight, self.fc1.bias, relu=True)
        x = Fo.dropout(x, self.p_drop, self.training, self.rng)
        x = Fo.linear(x, self.fc2.weight, self.fc2.bias, relu=True)
        x = Fo.dropout(x, self.p_drop, self.training, self.rng)
        x = Fo.linear(x, self.fc3.weight, self.fc3.bias)
        return x

    def manual_step(self, x, labels, dloss):
        """Hand-rolled fwd+bwd — see CNN_MNIST.manual_step."""
        from ..ops import ext
        E = ext()
        p = self.p_drop
        st = self.rng.gpu_state(x.device)
        a1 = E.conv2d_fwd(x, self.conv1.weight, self.conv1.bias, 1, 0, True)
        p1, i1 = E.maxpool2x2_fwd(a1)
        a2 = E.conv2d_fwd(p1, self.conv2.weight, self.conv2.bias, 1, 0, True)
        p2, i2 = E.maxpool2x2_fwd(a2)
        a3 = E.conv2d_fwd(p2, self.conv3.weight, self.conv3.bias, 1, 0, True)
        p3, i3 = E.maxpool2x2_fwd(a3)
        fl = E.nhwc_flatten(p3)
        d1, m1 = E.dropout_fwd_dev(fl, p, st, 0)
        h1 = E.linear_fwd(d1, self.fc1.weight, self.fc1.bias, True)
        d2, m2 = E.dropout_fwd_dev(h1, p, st, 1)
        h2 = E.linear_fwd(d2, self.fc2.weight, self.fc2.bias, True)
        d3, m3 = E.dropout_fwd_dev(h2, p, st, 2)
        out = E.linear_fwd(d3, self.fc3.weight, self.fc3.bias, False)
        loss, softmax = E.cross_entropy_fwd(out, labels)

        # relu masks fused (see CNN_MNIST.manual_step): fc relus fold
        # into the dropout backwards; every conv relu folds into the
        # following maxpool gather via the pooled values.
        g = E.cross_entropy_bwd(softmax, labels, dloss)
        g = E.linear_bwd_into(d3, self.fc3.weight, g,
                              self.fc3.weight.grad, self.fc3.bias.grad, True)
        g = E.dropout_relu_bwd(g, m3, h2, p)
        g = E.linear_bwd_into(d2, self.fc2.weight, g,
                              self.fc2.weight.grad, self.fc2.bias.grad, True)
        g = E.dropout_relu_bwd(g, m2, h1, p)
        g = E.linear_bwd_into(d1, self.fc1.weight, g,
                              self.fc1.weight.grad, self.fc1.bias.grad, True)
        g = E.dropout_bwd(g, m1, p)
        g = E.nhwc_unflatten(g, p3.shape[1], p3.shape[2], p3.shape[3])
        g = E.maxpool2x2_bwd_relu(g, i3, p3, list(a3.shape))
        g = E.conv2d_bwd_into(p2, self.conv3.weight, g, 1, 0, True,
                              self.conv3.weight.grad, self.conv3.bias.grad,
                              None)
        g = E.maxpool2x2_bwd_relu(g, i2, p2, list(a2.shape))
        g = E.conv2d_bwd_into(p1, self.conv2.weight, g, 1, 0, True,
                              self.conv2.weight.grad, self.conv2.bias.grad,
                              None)
        g = E.maxpool2x2_bwd_relu(g, i1, p1, list(a1.shape))
        E.conv2d_bwd_into(x, self.conv1.weight, g, 1, 0, False,
                          self.conv1.weight.grad, self.conv1.bias.grad,
                          None)
        return loss
