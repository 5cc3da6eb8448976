"""ResNet-50 for the flagship images/sec benchmark (BASELINE.json
configs 2-3), written from scratch.

MI355X mapping: the bottleneck 1x1 convolutions run on the in-house
256x256 MFMA GEMM over NHWC rows (sparkdl.ops.Conv1x1 — forward and
dgrad in hand-written kernels); 3x3/7x7 conv cores go through MIOpen
(channels_last + bf16 autocast picks the implicit-GEMM MFMA paths); every
BatchNorm/ReLU/residual-add runs through sparkdl.ops' fused bf16-NHWC
BatchNormAct2d kernels (SURVEY.md §2.2 N4/N5 — the conv-block epilogue
fusion); the optimizer step is sparkdl.ops.FusedSGD (one launch for all
161 tensors); gradient all-reduce is the bucketed DistributedOptimizer.
"""

import torch
import torch.nn as nn

from sparkdl.ops import BatchNormAct2d, Conv1x1


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, width, stride=1):
        super().__init__()
        cout = width * self.expansion
        self.conv1 = Conv1x1(cin, width)
        self.bn1 = BatchNormAct2d(width, relu=True)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = BatchNormAct2d(width, relu=True)
        self.conv3 = Conv1x1(width, cout)
        # bn3 fuses the residual add + final ReLU of the block
        self.bn3 = BatchNormAct2d(cout, relu=True)
        if stride != 1 or cin != cout:
            self.down_conv = Conv1x1(cin, cout, stride=stride)
            self.down_bn = BatchNormAct2d(cout, relu=False)
        else:
            self.down_conv = None

    def forward(self, x):
        if self.down_conv is not None:
            idn = self.down_bn(self.down_conv(x))
        else:
            idn = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=idn)


class ResNet50(nn.Module):
    LAYERS = (3, 4, 6, 3)

    def __init__(self, num_classes=1000):
        super().__init__()
        self.stem_conv = nn.Conv2d(3, 64, 7, stride=2, padding=3,
                                   bias=False)
        self.stem_bn = BatchNormAct2d(64, relu=True)
        self.stem_pool = nn.MaxPool2d(3, stride=2, padding=1)
        cin = 64
        stages = []
        for i, blocks in enumerate(self.LAYERS):
            width = 64 * (2 ** i)
            stride = 1 if i == 0 else 2
            layer = []
            for b in range(blocks):
                layer.append(Bottleneck(cin, width, stride if b == 0 else 1))
                cin = width * Bottleneck.expansion
            stages.append(nn.Sequential(*layer))
        self.stages = nn.Sequential(*stages)
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(2048, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
        # zero-init the last BN in each block (standard ResNet recipe)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)

    def forward(self, x):
        x = self.stem_pool(self.stem_bn(self.stem_conv(x)))
        x = self.stages(x)
        x = self.pool(x).flatten(1)
        return self.fc(x)
