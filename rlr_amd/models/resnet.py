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

    def forward(self, x):
        x = self._cast_in(x)
        x = Fo.conv2d(x, self.conv1.weight, None, 1, 1)
        x = Fo.relu(self.bn1(x))
        x = self.layers(x)
        x = Fo.global_avg_pool(x)
        return Fo.linear(x, self.fc.weight, self.fc.bias)
