"""CIFAR ResNet-18/34/50/101/152 (capability parity with
/root/reference/src/model_ops/resnet.py:14-113 — itself vendored kuangliu/pytorch-cifar
boilerplate) plus an ImageNet-geometry ResNet-50 for the 224x224 synthetic config
(BASELINE config 5).

Written fresh in idiomatic PyTorch; runs on PyTorch-ROCm (MIOpen/hipBLASLt kernels).
The architecture spec (block structure, stage widths, stride schedule, projection
shortcut rule) is the public ResNet design and is the capability being reproduced.
"""
import torch.nn as nn
import torch.nn.functional as F


def _conv_bn(width_in: int, width_out: int, k: int, stride: int = 1) -> nn.Sequential:
    return nn.Sequential(
        nn.Conv2d(width_in, width_out, k, stride=stride, padding=k // 2, bias=False),
        nn.BatchNorm2d(width_out),
    )


def _skip_path(width_in: int, width_out: int, stride: int) -> nn.Module:
    """Identity when shapes line up, else the standard 1x1 projection."""
    if stride == 1 and width_in == width_out:
        return nn.Identity()
    return _conv_bn(width_in, width_out, 1, stride)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, width_in: int, width: int, stride: int = 1):
        super().__init__()
        self.body1 = _conv_bn(width_in, width, 3, stride)
        self.body2 = _conv_bn(width, width, 3)
        self.skip = _skip_path(width_in, width * self.expansion, stride)

    def forward(self, x):
        out = self.body2(F.relu(self.body1(x)))
        return F.relu(out + self.skip(x))


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, width_in: int, width: int, stride: int = 1):
        super().__init__()
        self.body1 = _conv_bn(width_in, width, 1)
        self.body2 = _conv_bn(width, width, 3, stride)
        self.body3 = _conv_bn(width, width * self.expansion, 1)
        self.skip = _skip_path(width_in, width * self.expansion, stride)

    def forward(self, x):
        out = F.relu(self.body1(x))
        out = F.relu(self.body2(out))
        out = self.body3(out)
        return F.relu(out + self.skip(x))


_STAGE_WIDTHS = (64, 128, 256, 512)


def _stages(block, depths):
    """The four ResNet stages as (width, depth, stride) with stride 1 first."""
    return [(w, d, 1 if i == 0 else 2)
            for i, (w, d) in enumerate(zip(_STAGE_WIDTHS, depths))]


class ResNet(nn.Module):
    """CIFAR geometry: 3x3 stem, 4 stages at 32x32 input."""

    def __init__(self, block, depths, num_classes=10, in_channels=3):
        super().__init__()
        self.stem = _conv_bn(in_channels, 64, 3)
        width_in = 64
        stages = []
        for width, depth, stride in _stages(block, depths):
            blocks = []
            for s in [stride] + [1] * (depth - 1):
                blocks.append(block(width_in, width, s))
                width_in = width * block.expansion
            stages.append(nn.Sequential(*blocks))
        self.stages = nn.Sequential(*stages)
        self.fc = nn.Linear(width_in, num_classes)

    def forward(self, x):
        out = F.relu(self.stem(x))
        out = self.stages(out)
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


class ResNetImageNet(ResNet):
    """ImageNet geometry (7x7 stride-2 stem + maxpool) for 224x224 synthetic runs."""

    def __init__(self, block, depths, num_classes=1000, in_channels=3):
        super().__init__(block, depths, num_classes, in_channels)
        self.stem = _conv_bn(in_channels, 64, 7, stride=2)
        self.pool = nn.MaxPool2d(3, stride=2, padding=1)

    def forward(self, x):
        out = self.pool(F.relu(self.stem(x)))
        out = self.stages(out)
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


def ResNet18(num_classes=10, in_channels=3):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, in_channels)


def ResNet34(num_classes=10, in_channels=3):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, in_channels)


def ResNet50(num_classes=10, in_channels=3):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, in_channels)


def ResNet101(num_classes=10, in_channels=3):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, in_channels)


def ResNet152(num_classes=10, in_channels=3):
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes, in_channels)


def ResNet50ImageNet(num_classes=1000, in_channels=3):
    return ResNetImageNet(Bottleneck, [3, 4, 6, 3], num_classes, in_channels)
