"""CIFAR ResNeXt-29 (reference models/resnext.py:110-126)."""
import torch.nn as nn
import torch.nn.functional as F


class ResNeXtBottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, width, cardinality, base_width, stride=1):
        super().__init__()
        D = cardinality * (width * base_width // 64)
        cout = width * self.expansion
        self.conv1 = nn.Conv2d(cin, D, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(D)
        self.conv2 = nn.Conv2d(D, D, 3, stride, 1, groups=cardinality,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(D)
        self.conv3 = nn.Conv2d(D, cout, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout)
        self.shortcut = nn.Sequential()
        if stride != 1 or cin != cout:
            self.shortcut = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride, bias=False),
                nn.BatchNorm2d(cout))

    def forward(self, x):
        out = F.relu(self.bn1(self.conv1(x)), inplace=True)
        out = F.relu(self.bn2(self.conv2(out)), inplace=True)
        out = self.bn3(self.conv3(out))
        return F.relu(out + self.shortcut(x), inplace=True)


class CifarResNeXt(nn.Module):
    def __init__(self, cardinality=8, depth=29, base_width=64,
                 num_classes=10):
        super().__init__()
        assert (depth - 2) % 9 == 0
        n = (depth - 2) // 9
        self.conv1 = nn.Conv2d(3, 64, 3, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.inplanes = 64
        self.layer1 = self._make_layer(64, n, cardinality, base_width, 1)
        self.layer2 = self._make_layer(128, n, cardinality, base_width, 2)
        self.layer3 = self._make_layer(256, n, cardinality, base_width, 2)
        self.fc = nn.Linear(256 * ResNeXtBottleneck.expansion, num_classes)

    def _make_layer(self, width, blocks, cardinality, base_width, stride):
        layers = [ResNeXtBottleneck(self.inplanes, width, cardinality,
                                    base_width, stride)]
        self.inplanes = width * ResNeXtBottleneck.expansion
        for _ in range(1, blocks):
            layers.append(ResNeXtBottleneck(self.inplanes, width,
                                            cardinality, base_width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = F.relu(self.bn1(self.conv1(x)), inplace=True)
        x = self.layer3(self.layer2(self.layer1(x)))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def resnext29_8_64(num_classes=10):
    return CifarResNeXt(cardinality=8, depth=29, base_width=64,
                        num_classes=num_classes)


def resnext29_16_64(num_classes=10):
    return CifarResNeXt(cardinality=16, depth=29, base_width=64,
                        num_classes=num_classes)
