"""Pre-activation CIFAR ResNets (reference models/preresnet.py:113-151)."""
import torch.nn as nn
import torch.nn.functional as F


class PreActBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(cin)
        self.conv1 = nn.Conv2d(cin, cout, 3, stride, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(cout)
        self.conv2 = nn.Conv2d(cout, cout, 3, 1, 1, bias=False)
        self.shortcut = None
        if stride != 1 or cin != cout:
            self.shortcut = nn.Conv2d(cin, cout, 1, stride, bias=False)

    def forward(self, x):
        out = F.relu(self.bn1(x), inplace=True)
        identity = x if self.shortcut is None else self.shortcut(out)
        out = self.conv1(out)
        out = self.conv2(F.relu(self.bn2(out), inplace=True))
        return out + identity


class PreResNet(nn.Module):
    def __init__(self, depth, num_classes=10):
        super().__init__()
        assert (depth - 2) % 6 == 0
        n = (depth - 2) // 6
        self.conv1 = nn.Conv2d(3, 16, 3, padding=1, bias=False)
        self.layer1 = self._make_layer(16, 16, n, 1)
        self.layer2 = self._make_layer(16, 32, n, 2)
        self.layer3 = self._make_layer(32, 64, n, 2)
        self.bn_final = nn.BatchNorm2d(64)
        self.fc = nn.Linear(64, num_classes)

    def _make_layer(self, cin, cout, blocks, stride):
        layers = [PreActBlock(cin, cout, stride)]
        for _ in range(1, blocks):
            layers.append(PreActBlock(cout, cout))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.conv1(x)
        x = self.layer3(self.layer2(self.layer1(x)))
        x = F.relu(self.bn_final(x), inplace=True)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def preresnet20(num_classes=10):
    return PreResNet(20, num_classes)


def preresnet32(num_classes=10):
    return PreResNet(32, num_classes)


def preresnet44(num_classes=10):
    return PreResNet(44, num_classes)


def preresnet56(num_classes=10):
    return PreResNet(56, num_classes)


def preresnet110(num_classes=10):
    return PreResNet(110, num_classes)
