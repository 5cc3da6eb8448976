"""ResNet-50 for the flagship images/sec benchmark (BASELINE.json
configs 2-3), written from scratch.

MI355X mapping: convolution cores go through MIOpen (plain library convs,
channels_last + bf16 autocast picks the implicit-GEMM MFMA paths);
the optimizer step is sparkdl.ops.FusedSGD (one kernel launch for all
161 tensors); gradient all-reduce is the bucketed DistributedOptimizer.
"""

import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, width, stride=1):
        super().__init__()
        cout = width * self.expansion
        self.conv1 = nn.Conv2d(cin, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, cout, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout)
        self.relu = nn.ReLU(inplace=True)
        if stride != 1 or cin != cout:
            self.down = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                nn.BatchNorm2d(cout))
        else:
            self.down = None

    def forward(self, x):
        idn = x if self.down is None else self.down(x)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.relu(out + idn)


class ResNet50(nn.Module):
    LAYERS = (3, 4, 6, 3)

    def __init__(self, num_classes=1000):
        super().__init__()
        self.stem = nn.Sequential(
            nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False),
            nn.BatchNorm2d(64),
            nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1))
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
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        # zero-init the last BN in each block (standard ResNet recipe)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)

    def forward(self, x):
        x = self.stem(x)
        x = self.stages(x)
        x = self.pool(x).flatten(1)
        return self.fc(x)
