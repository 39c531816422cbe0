"""CIFAR ResNets (He et al. 2015, option-A/B shortcuts).

Native implementations of the reference model zoo entries resnet20/32/44/
56/110 (reference models/resnet.py:109-147) and the resnet_mod* variants
(reference models/resnet_mod.py:129-167). ``resnetN`` uses projection
(option-B) shortcuts; ``resnet_modN`` uses parameter-free zero-padded
(option-A) shortcuts — the two families the reference exposes.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from .common import BNReLU, BNAddReLU


def _conv3x3(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, kernel_size=3, stride=stride, padding=1,
                     bias=False)


class PadShortcut(nn.Module):
    """Option-A shortcut: stride-2 subsample + zero-pad channels."""

    def __init__(self, cin, cout, stride):
        super().__init__()
        self.stride = stride
        self.pad = cout - cin

    def forward(self, x):
        out = x[:, :, ::self.stride, ::self.stride]
        return F.pad(out, (0, 0, 0, 0, 0, self.pad))


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, cout, stride=1, option='B'):
        super().__init__()
        self.conv1 = _conv3x3(cin, cout, stride)
        self.bn1 = BNReLU(cout)
        self.conv2 = _conv3x3(cout, cout)
        self.bn2 = BNAddReLU(cout)   # bn + residual add + relu, fused
        self.shortcut = nn.Sequential()
        if stride != 1 or cin != cout:
            if option == 'A':
                self.shortcut = PadShortcut(cin, cout, stride)
            else:
                self.shortcut = nn.Sequential(
                    nn.Conv2d(cin, cout, kernel_size=1, stride=stride,
                              bias=False),
                    nn.BatchNorm2d(cout))

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), self.shortcut(x))


class CifarResNet(nn.Module):
    def __init__(self, depth, num_classes=10, option='B'):
        super().__init__()
        assert (depth - 2) % 6 == 0, 'depth must be 6n+2'
        n = (depth - 2) // 6
        self.conv1 = _conv3x3(3, 16)
        self.bn1 = BNReLU(16)
        self.layer1 = self._make_layer(16, 16, n, 1, option)
        self.layer2 = self._make_layer(16, 32, n, 2, option)
        self.layer3 = self._make_layer(32, 64, n, 2, option)
        self.fc = nn.Linear(64, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode='fan_out',
                                        nonlinearity='relu')
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def _make_layer(self, cin, cout, blocks, stride, option):
        layers = [BasicBlock(cin, cout, stride, option)]
        for _ in range(1, blocks):
            layers.append(BasicBlock(cout, cout, 1, option))
        return nn.Sequential(*layers)

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        out = self.layer3(self.layer2(self.layer1(out)))
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


def resnet20(num_classes=10):
    return CifarResNet(20, num_classes)


def resnet32(num_classes=10):
    return CifarResNet(32, num_classes)


def resnet44(num_classes=10):
    return CifarResNet(44, num_classes)


def resnet56(num_classes=10):
    return CifarResNet(56, num_classes)


def resnet110(num_classes=10):
    return CifarResNet(110, num_classes)


def resnet_mod20(num_classes=10):
    return CifarResNet(20, num_classes, option='A')


def resnet_mod32(num_classes=10):
    return CifarResNet(32, num_classes, option='A')


def resnet_mod44(num_classes=10):
    return CifarResNet(44, num_classes, option='A')


def resnet_mod56(num_classes=10):
    return CifarResNet(56, num_classes, option='A')


def resnet_mod110(num_classes=10):
    return CifarResNet(110, num_classes, option='A')
