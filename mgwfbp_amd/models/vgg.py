"""Config-driven VGG nets.

``VGG('VGG16')`` etc. for CIFAR (reference models/vgg.py:6-46) and
``vgg16i`` for ImageNet (replacing the reference's torchvision.vgg16,
reference dl_trainer.py:107-108).
"""
import torch.nn as nn

cfg = {
    'VGG11': [64, 'M', 128, 'M', 256, 256, 'M', 512, 512, 'M', 512, 512,
              'M'],
    'VGG13': [64, 64, 'M', 128, 128, 'M', 256, 256, 'M', 512, 512, 'M',
              512, 512, 'M'],
    'VGG16': [64, 64, 'M', 128, 128, 'M', 256, 256, 256, 'M', 512, 512,
              512, 'M', 512, 512, 512, 'M'],
    'VGG19': [64, 64, 'M', 128, 128, 'M', 256, 256, 256, 256, 'M', 512,
              512, 512, 512, 'M', 512, 512, 512, 512, 'M'],
}


def _make_layers(config, batch_norm=True, in_channels=3):
    layers = []
    for v in config:
        if v == 'M':
            layers.append(nn.MaxPool2d(kernel_size=2, stride=2))
        else:
            layers.append(nn.Conv2d(in_channels, v, kernel_size=3,
                                    padding=1))
            if batch_norm:
                layers.append(nn.BatchNorm2d(v))
            layers.append(nn.ReLU(inplace=True))
            in_channels = v
    return nn.Sequential(*layers)


class VGG(nn.Module):
    """CIFAR VGG: conv stack + single linear head (32x32 input)."""

    def __init__(self, vgg_name='VGG16', num_classes=10):
        super().__init__()
        self.features = _make_layers(cfg[vgg_name])
        self.classifier = nn.Linear(512, num_classes)

    def forward(self, x):
        out = self.features(x)
        out = out.flatten(1)
        return self.classifier(out)


class VGGImageNet(nn.Module):
    """ImageNet VGG (224x224): conv stack + 3-layer FC head."""

    def __init__(self, vgg_name='VGG16', num_classes=1000):
        super().__init__()
        self.features = _make_layers(cfg[vgg_name], batch_norm=False)
        self.avgpool = nn.AdaptiveAvgPool2d((7, 7))
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(inplace=True),
            nn.Dropout(),
            nn.Linear(4096, 4096), nn.ReLU(inplace=True), nn.Dropout(),
            nn.Linear(4096, num_classes))

    def forward(self, x):
        x = self.avgpool(self.features(x))
        return self.classifier(x.flatten(1))


def vgg16(num_classes=10):
    return VGG('VGG16', num_classes)


def vgg19(num_classes=10):
    return VGG('VGG19', num_classes)


def vgg16i(num_classes=1000):
    return VGGImageNet('VGG16', num_classes)
