"""Convolutional classifiers for the image case studies.

Architectures mirror the reference case studies' keras models so tap-layer
indices line up (reference: case_study_mnist.py:50-69,
case_study_cifar10.py:33-57); ResNet-20 is the flagship benchmark model
named by BASELINE.json ("CIFAR-10 ResNet-20 LSA ... sharded DP").
"""

import torch
import torch.nn as nn

from .base import TapModel


class MnistCNN(TapModel):
    """Conv32-Pool-Conv64-Pool-Flatten-Dropout(.5)-Dense10.

    Layer indices (NC [0,1,2,3], SA [3] — reference case_study_mnist.py:25-27):
    0 conv1+relu, 1 pool1, 2 conv2+relu, 3 pool2 (5x5x64 = 1600 ATs),
    4 flatten, 5 dropout, 6 dense.
    Used for both MNIST and Fashion-MNIST (identical in the reference).
    """

    num_classes = 10
    input_shape = (1, 28, 28)

    def __init__(self):
        super().__init__()
        self.layers = nn.ModuleList(
            [
                nn.Sequential(nn.Conv2d(1, 32, 3), nn.ReLU()),
                nn.MaxPool2d(2),
                nn.Sequential(nn.Conv2d(32, 64, 3), nn.ReLU()),
                nn.MaxPool2d(2),
                nn.Flatten(),
                nn.Dropout(0.5),
                nn.Linear(5 * 5 * 64, 10),
            ]
        )


class Cifar10CNN(TapModel):
    """TF-tutorial CNN: Conv32-Pool-Conv64-Pool-Conv64-Flatten-Dense64-Dense10.

    Layer indices (NC [0,1,2,3], SA [3] = pool2, 6x6x64 = 2304 ATs —
    reference case_study_cifar10.py:24-26,33-57). No dropout layer, so no
    MC-dropout VR for this study (reference eval_apfd_table.py:97).
    """

    num_classes = 10
    input_shape = (3, 32, 32)

    def __init__(self):
        super().__init__()
        self.layers = nn.ModuleList(
            [
                nn.Sequential(nn.Conv2d(3, 32, 3), nn.ReLU()),
                nn.MaxPool2d(2),
                nn.Sequential(nn.Conv2d(32, 64, 3), nn.ReLU()),
                nn.MaxPool2d(2),
                nn.Sequential(nn.Conv2d(64, 64, 3), nn.ReLU()),
                nn.Flatten(),
                nn.Sequential(nn.Linear(4 * 4 * 64, 64), nn.ReLU()),
                nn.Linear(64, 10),
            ]
        )


class _BasicBlock(nn.Module):
    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(cout)
        self.conv2 = nn.Conv2d(cout, cout, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(cout)
        self.relu = nn.ReLU(inplace=True)
        if stride != 1 or cin != cout:
            self.shortcut = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                nn.BatchNorm2d(cout),
            )
        else:
            self.shortcut = nn.Identity()

    def forward(self, x):
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.relu(out + self.shortcut(x))


class ResNet20(TapModel):
    """CIFAR ResNet-20 (16/32/64 widths, 3 stages x 3 basic blocks).

    The benchmark flagship (BASELINE.json config 3). Layer indices:
    0 stem (conv+bn+relu), 1-3 stage1, 4-6 stage2, 7-9 stage3,
    10 global-avg-pool+flatten (64 ATs), 11 linear head.
    Default SA tap: layer 9 (the 8x8x64 = 4096-wide pre-pool feature map,
    the workload-representative AT width); NC taps [0, 3, 6, 9].
    """

    num_classes = 10
    input_shape = (3, 32, 32)
    sa_layers = [9]
    nc_layers = [0, 3, 6, 9]

    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.num_classes = num_classes
        stem = nn.Sequential(
            nn.Conv2d(3, 16, 3, padding=1, bias=False),
            nn.BatchNorm2d(16),
            nn.ReLU(inplace=True),
        )
        blocks = []
        cin = 16
        for stage, cout in enumerate([16, 32, 64]):
            for b in range(3):
                stride = 2 if (stage > 0 and b == 0) else 1
                blocks.append(_BasicBlock(cin, cout, stride))
                cin = cout
        pool = nn.Sequential(nn.AdaptiveAvgPool2d(1), nn.Flatten())
        head = nn.Linear(64, num_classes)
        self.layers = nn.ModuleList([stem, *blocks, pool, head])
