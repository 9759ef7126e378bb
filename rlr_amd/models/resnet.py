"""ResNet18 with BatchNorm — the build's deliberate capability extension
beyond the reference's CNN zoo (required by BASELINE.json configs 3-4:
"CIFAR10 ResNet18", Conv2d/BatchNorm/ReLU/Linear hot path; absent from the
reference).

CIFAR-style stem (3x3 stride 1, no maxpool) for 32x32 inputs; 4 stages of
2 BasicBlocks at widths 64/128/256/512; global average pool; fc 512->10.
~11.17 M parameters.  All compute goes through the rlr_amd op layer
(fused conv+relu, batch_norm, add_relu residual joins)."""

import torch.nn as nn

from ..ops import functional as Fo
from .cnn import _OpsModel


class _BN(nn.BatchNorm2d):
    """nn.BatchNorm2d container (params + running stats), rlr_amd compute."""

    def forward(self, x):
        return Fo.batch_norm(x, self.weight, self.bias, self.running_mean,
                             self.running_var, self.momentum, self.eps,
                             self.training)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch, out_ch, stride=1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, stride=stride, padding=1,
                               bias=False)
        self.bn1 = _BN(out_ch)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, stride=1, padding=1,
                               bias=False)
        self.bn2 = _BN(out_ch)
        self.down_conv = None
        if stride != 1 or in_ch != out_ch:
            self.down_conv = nn.Conv2d(in_ch, out_ch, 1, stride=stride,
                                       bias=False)
            self.down_bn = _BN(out_ch)

    def forward(self, x):
        out = Fo.conv2d(x, self.conv1.weight, None, self.conv1.stride[0],
                        self.conv1.padding[0])
        out = Fo.relu(self.bn1(out))
        out = Fo.conv2d(out, self.conv2.weight, None, 1, 1)
        out = self.bn2(out)
        sc = x
        if self.down_conv is not None:
            sc = Fo.conv2d(x, self.down_conv.weight, None,
                           self.down_conv.stride[0], 0)
            sc = self.down_bn(sc)
        return Fo.add_relu(out, sc)

    # ---- manual tape (ResNet18.manual_step) ----

    def _bn_fwd(self, E, bn, x, relu=False):
        return E.batchnorm_fwd(x, bn.weight, bn.bias, bn.running_mean,
                               bn.running_var, bn.momentum, bn.eps, True,
                               relu)

    def _tape_fwd(self, E, x):
        s = self.conv1.stride[0]
        o1 = E.conv2d_fwd(x, self.conv1.weight, None, s, 1, False)
        a1, sm1, sr1 = self._bn_fwd(E, self.bn1, o1, relu=True)  # bn+relu
        o2 = E.conv2d_fwd(a1, self.conv2.weight, None, 1, 1, False)
        b2, sm2, sr2 = self._bn_fwd(E, self.bn2, o2)
        if self.down_conv is not None:
            sc = E.conv2d_fwd(x, self.down_conv.weight, None, s, 0, False)
            sb, smd, srd = self._bn_fwd(E, self.down_bn, sc)
        else:
            sc = smd = srd = None
            sb = x
        y = E.add_relu_fwd(b2, sb)
        return y, (x, o1, sm1, sr1, a1, o2, sm2, sr2, sc, smd, srd)

    def _tape_bwd(self, E, dy, saved):
        """dy arrives PRE-MASKED by this block's output relu (the
        downstream consumer fuses the mask — add_relu_bwd_ or
        gap_bwd_relu); the returned dx is masked by this block's INPUT
        relu, so the producer upstream never runs a separate relu_bwd."""
        (x, o1, sm1, sr1, a1, o2, sm2, sr2, sc, smd, srd) = saved
        s = self.conv1.stride[0]
        d = dy
        g = E.batchnorm_bwd_into(o2, self.bn2.weight, sm2, sr2, d,
                                 self.bn2.weight.grad, self.bn2.bias.grad)
        g = E.conv2d_bwd_wdx_into(a1, self.conv2.weight, g, 1, 1, True,
                                  self.conv2.weight.grad, a1)  # relu fused
        g = E.batchnorm_bwd_into(o1, self.bn1.weight, sm1, sr1, g,
                                 self.bn1.weight.grad, self.bn1.bias.grad)
        gmain = E.conv2d_bwd_wdx_into(x, self.conv1.weight, g, s, 1,
                                      True, self.conv1.weight.grad, None)
        if self.down_conv is not None:
            gs = E.batchnorm_bwd_into(sc, self.down_bn.weight, smd, srd, d,
                                      self.down_bn.weight.grad,
                                      self.down_bn.bias.grad)
            gs = E.conv2d_bwd_wdx_into(x, self.down_conv.weight, gs, s, 0,
                                       True, self.down_conv.weight.grad,
                                       None)
            return E.add_relu_bwd_(gmain, gs, x)  # join + input-relu mask
        return E.add_relu_bwd_(gmain, d, x)


class ResNet18(_OpsModel):
    def __init__(self, num_classes=10):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 3, stride=1, padding=1, bias=False)
        self.bn1 = _BN(64)
        widths = [64, 128, 256, 512]
        strides = [1, 2, 2, 2]
        layers = []
        in_ch = 64
        for w, s in zip(widths, strides):
            layers.append(BasicBlock(in_ch, w, s))
            layers.append(BasicBlock(w, w, 1))
            in_ch = w
        self.layers = nn.Sequential(*layers)
        self.fc = nn.Linear(512, num_classes)

    manual_bf16_ok = True  # the tape below handles the bf16 compute path

    def forward(self, x):
        x = self._cast_in(x)
        x = Fo.conv2d(x, self.conv1.weight, None, 1, 1)
        x = Fo.relu(self.bn1(x))
        x = self.layers(x)
        x = Fo.global_avg_pool(x)
        return Fo.linear(x, self.fc.weight, self.fc.bias)

    def manual_step(self, x, labels, dloss):
        """Hand-rolled fwd+bwd (see CNN_MNIST.manual_step): conv/BN grads
        land directly in their flat_grads views — no autograd accumulate
        kernels (~62 per step for this model).  The residual joins sum the
        two incoming gradients with one fused add (commutative: bitwise-
        equal to autograd's accumulation)."""
        import torch
        from ..ops import ext
        E = ext()
        if self.compute_dtype is not None and x.dtype != self.compute_dtype:
            x = x.to(self.compute_dtype)
        a0 = E.conv2d_fwd(x, self.conv1.weight, None, 1, 1, False)
        r0, sm0, sr0 = E.batchnorm_fwd(
            a0, self.bn1.weight, self.bn1.bias, self.bn1.running_mean,
            self.bn1.running_var, self.bn1.momentum, self.bn1.eps, True,
            True)  # bn+relu fused
        h = r0
        saves = []
        for blk in self.layers:
            h, sv = blk._tape_fwd(E, h)
            saves.append(sv)
        gp = E.gap_fwd(h)
        out = E.linear_fwd(gp, self.fc.weight, self.fc.bias, False)
        logits = out.float() if out.dtype != torch.float32 else out
        loss, softmax = E.cross_entropy_fwd(logits.contiguous(), labels)

        g = E.cross_entropy_bwd(softmax, labels, dloss)
        if out.dtype != torch.float32:
            g = g.to(out.dtype)
        dx_fc, dw_fc, db_fc = E.linear_bwd(gp, self.fc.weight, g)
        # the fc is 512x10: a copy into the views is cheaper than a
        # dedicated bf16 out-variant of the gemm
        self.fc.weight.grad.copy_(dw_fc.view_as(self.fc.weight))
        self.fc.bias.grad.copy_(db_fc)
        # last block's output-relu mask fused into the gap gradient
        g = E.gap_bwd_relu(dx_fc.contiguous(), h, list(h.shape))
        for blk, sv in zip(reversed(self.layers), reversed(saves)):
            g = blk._tape_bwd(E, g, sv)
        # g is already masked by r0 (block 0's fused join)
        g = E.batchnorm_bwd_into(a0, self.bn1.weight, sm0, sr0, g,
                                 self.bn1.weight.grad, self.bn1.bias.grad)
        E.conv2d_bwd_wdx_into(x, self.conv1.weight, g, 1, 1, False,
                              self.conv1.weight.grad, None)
        return loss
