"""DenseNets: CIFAR DenseNet-BC (densenet100_12, reference
models/densenet.py:99) and ImageNet densenet121/161/201 (replacing the
reference's torchvision path, dl_trainer.py:97-102)."""
import torch
import torch.nn as nn
import torch.nn.functional as F


class _DenseLayer(nn.Module):
    def __init__(self, cin, growth, bn_size=4):
        super().__init__()
        self.norm1 = nn.BatchNorm2d(cin)
        self.conv1 = nn.Conv2d(cin, bn_size * growth, 1, bias=False)
        self.norm2 = nn.BatchNorm2d(bn_size * growth)
        self.conv2 = nn.Conv2d(bn_size * growth, growth, 3, padding=1,
                               bias=False)

    def forward(self, x):
        out = self.conv1(F.relu(self.norm1(x), inplace=True))
        out = self.conv2(F.relu(self.norm2(out), inplace=True))
        return torch.cat([x, out], 1)


class _Transition(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.norm = nn.BatchNorm2d(cin)
        self.conv = nn.Conv2d(cin, cout, 1, bias=False)

    def forward(self, x):
        x = self.conv(F.relu(self.norm(x), inplace=True))
        return F.avg_pool2d(x, 2)


class DenseNet(nn.Module):
    def __init__(self, growth=32, block_config=(6, 12, 24, 16),
                 num_init_features=64, num_classes=1000, cifar=False):
        super().__init__()
        if cifar:
            self.stem = nn.Conv2d(3, num_init_features, 3, padding=1,
                                  bias=False)
        else:
            self.stem = nn.Sequential(
                nn.Conv2d(3, num_init_features, 7, 2, 3, bias=False),
                nn.BatchNorm2d(num_init_features),
                nn.ReLU(inplace=True),
                nn.MaxPool2d(3, 2, 1))
        blocks = []
        ch = num_init_features
        for i, n in enumerate(block_config):
            for _ in range(n):
                blocks.append(_DenseLayer(ch, growth))
                ch += growth
            if i != len(block_config) - 1:
                blocks.append(_Transition(ch, ch // 2))
                ch = ch // 2
        self.features = nn.Sequential(*blocks)
        self.norm_final = nn.BatchNorm2d(ch)
        self.classifier = nn.Linear(ch, num_classes)

    def forward(self, x):
        x = self.features(self.stem(x))
        x = F.relu(self.norm_final(x), inplace=True)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.classifier(x)


def densenet100_12(num_classes=10):
    """DenseNet-BC L=100 k=12 for CIFAR (reference densenet.py:99)."""
    return DenseNet(growth=12, block_config=(16, 16, 16),
                    num_init_features=24, num_classes=num_classes,
                    cifar=True)


def densenet121(num_classes=1000):
    return DenseNet(32, (6, 12, 24, 16), 64, num_classes)


def densenet161(num_classes=1000):
    return DenseNet(48, (6, 12, 36, 24), 96, num_classes)


def densenet201(num_classes=1000):
    return DenseNet(32, (6, 12, 48, 32), 64, num_classes)
