"""Inception-v4 (reference models/inceptionv4.py:314) and Inception-v3
(replacing the reference's torchvision.inception_v3 path,
dl_trainer.py:105-106; aux classifier omitted — the trainer never uses
it)."""
import torch
import torch.nn as nn
import torch.nn.functional as F


class Conv2dBN(nn.Module):
    def __init__(self, cin, cout, **kwargs):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, bias=False, **kwargs)
        self.bn = nn.BatchNorm2d(cout, eps=0.001)

    def forward(self, x):
        return F.relu(self.bn(self.conv(x)), inplace=True)


# ------------------------------ Inception-v4 ------------------------------

class Stem(nn.Module):
    def __init__(self):
        super().__init__()
        self.step1 = nn.Sequential(
            Conv2dBN(3, 32, kernel_size=3, stride=2),
            Conv2dBN(32, 32, kernel_size=3),
            Conv2dBN(32, 64, kernel_size=3, padding=1))
        self.branch_pool = nn.MaxPool2d(3, 2)
        self.branch_conv = Conv2dBN(64, 96, kernel_size=3, stride=2)
        self.mixed_a = nn.Sequential(
            Conv2dBN(160, 64, kernel_size=1),
            Conv2dBN(64, 96, kernel_size=3))
        self.mixed_b = nn.Sequential(
            Conv2dBN(160, 64, kernel_size=1),
            Conv2dBN(64, 64, kernel_size=(7, 1), padding=(3, 0)),
            Conv2dBN(64, 64, kernel_size=(1, 7), padding=(0, 3)),
            Conv2dBN(64, 96, kernel_size=3))
        self.out_conv = Conv2dBN(192, 192, kernel_size=3, stride=2)
        self.out_pool = nn.MaxPool2d(3, 2)

    def forward(self, x):
        x = self.step1(x)
        x = torch.cat([self.branch_pool(x), self.branch_conv(x)], 1)
        x = torch.cat([self.mixed_a(x), self.mixed_b(x)], 1)
        return torch.cat([self.out_conv(x), self.out_pool(x)], 1)


class InceptionA4(nn.Module):
    def __init__(self, cin=384):
        super().__init__()
        self.b1 = Conv2dBN(cin, 96, kernel_size=1)
        self.b2 = nn.Sequential(Conv2dBN(cin, 64, kernel_size=1),
                                Conv2dBN(64, 96, kernel_size=3, padding=1))
        self.b3 = nn.Sequential(Conv2dBN(cin, 64, kernel_size=1),
                                Conv2dBN(64, 96, kernel_size=3, padding=1),
                                Conv2dBN(96, 96, kernel_size=3, padding=1))
        self.b4 = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                Conv2dBN(cin, 96, kernel_size=1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], 1)


class ReductionA4(nn.Module):
    def __init__(self, cin=384):
        super().__init__()
        self.b1 = Conv2dBN(cin, 384, kernel_size=3, stride=2)
        self.b2 = nn.Sequential(Conv2dBN(cin, 192, kernel_size=1),
                                Conv2dBN(192, 224, kernel_size=3, padding=1),
                                Conv2dBN(224, 256, kernel_size=3, stride=2))
        self.b3 = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x)], 1)


class InceptionB4(nn.Module):
    def __init__(self, cin=1024):
        super().__init__()
        self.b1 = Conv2dBN(cin, 384, kernel_size=1)
        self.b2 = nn.Sequential(
            Conv2dBN(cin, 192, kernel_size=1),
            Conv2dBN(192, 224, kernel_size=(1, 7), padding=(0, 3)),
            Conv2dBN(224, 256, kernel_size=(7, 1), padding=(3, 0)))
        self.b3 = nn.Sequential(
            Conv2dBN(cin, 192, kernel_size=1),
            Conv2dBN(192, 192, kernel_size=(7, 1), padding=(3, 0)),
            Conv2dBN(192, 224, kernel_size=(1, 7), padding=(0, 3)),
            Conv2dBN(224, 224, kernel_size=(7, 1), padding=(3, 0)),
            Conv2dBN(224, 256, kernel_size=(1, 7), padding=(0, 3)))
        self.b4 = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                Conv2dBN(cin, 128, kernel_size=1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], 1)


class ReductionB4(nn.Module):
    def __init__(self, cin=1024):
        super().__init__()
        self.b1 = nn.Sequential(Conv2dBN(cin, 192, kernel_size=1),
                                Conv2dBN(192, 192, kernel_size=3, stride=2))
        self.b2 = nn.Sequential(
            Conv2dBN(cin, 256, kernel_size=1),
            Conv2dBN(256, 256, kernel_size=(1, 7), padding=(0, 3)),
            Conv2dBN(256, 320, kernel_size=(7, 1), padding=(3, 0)),
            Conv2dBN(320, 320, kernel_size=3, stride=2))
        self.b3 = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x)], 1)


class InceptionC4(nn.Module):
    def __init__(self, cin=1536):
        super().__init__()
        self.b1 = Conv2dBN(cin, 256, kernel_size=1)
        self.b2_stem = Conv2dBN(cin, 384, kernel_size=1)
        self.b2_a = Conv2dBN(384, 256, kernel_size=(1, 3), padding=(0, 1))
        self.b2_b = Conv2dBN(384, 256, kernel_size=(3, 1), padding=(1, 0))
        self.b3_stem = nn.Sequential(
            Conv2dBN(cin, 384, kernel_size=1),
            Conv2dBN(384, 448, kernel_size=(3, 1), padding=(1, 0)),
            Conv2dBN(448, 512, kernel_size=(1, 3), padding=(0, 1)))
        self.b3_a = Conv2dBN(512, 256, kernel_size=(1, 3), padding=(0, 1))
        self.b3_b = Conv2dBN(512, 256, kernel_size=(3, 1), padding=(1, 0))
        self.b4 = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                Conv2dBN(cin, 256, kernel_size=1))

    def forward(self, x):
        b2 = self.b2_stem(x)
        b3 = self.b3_stem(x)
        return torch.cat([self.b1(x), self.b2_a(b2), self.b2_b(b2),
                          self.b3_a(b3), self.b3_b(b3), self.b4(x)], 1)


class InceptionV4(nn.Module):
    def __init__(self, num_classes=1000):
        super().__init__()
        blocks = [Stem()]
        blocks += [InceptionA4() for _ in range(4)]
        blocks.append(ReductionA4())
        blocks += [InceptionB4() for _ in range(7)]
        blocks.append(ReductionB4())
        blocks += [InceptionC4() for _ in range(3)]
        self.features = nn.Sequential(*blocks)
        self.dropout = nn.Dropout(0.2)
        self.fc = nn.Linear(1536, num_classes)

    def forward(self, x):
        x = self.features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(self.dropout(x))


def inceptionv4(num_classes=1000):
    return InceptionV4(num_classes)


# ------------------------------ Inception-v3 ------------------------------

class InceptionA3(nn.Module):
    def __init__(self, cin, pool_features):
        super().__init__()
        self.b1 = Conv2dBN(cin, 64, kernel_size=1)
        self.b2 = nn.Sequential(Conv2dBN(cin, 48, kernel_size=1),
                                Conv2dBN(48, 64, kernel_size=5, padding=2))
        self.b3 = nn.Sequential(Conv2dBN(cin, 64, kernel_size=1),
                                Conv2dBN(64, 96, kernel_size=3, padding=1),
                                Conv2dBN(96, 96, kernel_size=3, padding=1))
        self.b4 = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                Conv2dBN(cin, pool_features, kernel_size=1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], 1)


class ReductionA3(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b1 = Conv2dBN(cin, 384, kernel_size=3, stride=2)
        self.b2 = nn.Sequential(Conv2dBN(cin, 64, kernel_size=1),
                                Conv2dBN(64, 96, kernel_size=3, padding=1),
                                Conv2dBN(96, 96, kernel_size=3, stride=2))
        self.b3 = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x)], 1)


class InceptionB3(nn.Module):
    def __init__(self, cin, c7):
        super().__init__()
        self.b1 = Conv2dBN(cin, 192, kernel_size=1)
        self.b2 = nn.Sequential(
            Conv2dBN(cin, c7, kernel_size=1),
            Conv2dBN(c7, c7, kernel_size=(1, 7), padding=(0, 3)),
            Conv2dBN(c7, 192, kernel_size=(7, 1), padding=(3, 0)))
        self.b3 = nn.Sequential(
            Conv2dBN(cin, c7, kernel_size=1),
            Conv2dBN(c7, c7, kernel_size=(7, 1), padding=(3, 0)),
            Conv2dBN(c7, c7, kernel_size=(1, 7), padding=(0, 3)),
            Conv2dBN(c7, c7, kernel_size=(7, 1), padding=(3, 0)),
            Conv2dBN(c7, 192, kernel_size=(1, 7), padding=(0, 3)))
        self.b4 = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                Conv2dBN(cin, 192, kernel_size=1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], 1)


class ReductionB3(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b1 = nn.Sequential(Conv2dBN(cin, 192, kernel_size=1),
                                Conv2dBN(192, 320, kernel_size=3, stride=2))
        self.b2 = nn.Sequential(
            Conv2dBN(cin, 192, kernel_size=1),
            Conv2dBN(192, 192, kernel_size=(1, 7), padding=(0, 3)),
            Conv2dBN(192, 192, kernel_size=(7, 1), padding=(3, 0)),
            Conv2dBN(192, 192, kernel_size=3, stride=2))
        self.b3 = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x)], 1)


class InceptionC3(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b1 = Conv2dBN(cin, 320, kernel_size=1)
        self.b2_stem = Conv2dBN(cin, 384, kernel_size=1)
        self.b2_a = Conv2dBN(384, 384, kernel_size=(1, 3), padding=(0, 1))
        self.b2_b = Conv2dBN(384, 384, kernel_size=(3, 1), padding=(1, 0))
        self.b3_stem = nn.Sequential(
            Conv2dBN(cin, 448, kernel_size=1),
            Conv2dBN(448, 384, kernel_size=3, padding=1))
        self.b3_a = Conv2dBN(384, 384, kernel_size=(1, 3), padding=(0, 1))
        self.b3_b = Conv2dBN(384, 384, kernel_size=(3, 1), padding=(1, 0))
        self.b4 = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                Conv2dBN(cin, 192, kernel_size=1))

    def forward(self, x):
        b2 = self.b2_stem(x)
        b3 = self.b3_stem(x)
        return torch.cat([self.b1(x), self.b2_a(b2), self.b2_b(b2),
                          self.b3_a(b3), self.b3_b(b3), self.b4(x)], 1)


class InceptionV3(nn.Module):
    def __init__(self, num_classes=1000):
        super().__init__()
        self.stem = nn.Sequential(
            Conv2dBN(3, 32, kernel_size=3, stride=2),
            Conv2dBN(32, 32, kernel_size=3),
            Conv2dBN(32, 64, kernel_size=3, padding=1),
            nn.MaxPool2d(3, 2),
            Conv2dBN(64, 80, kernel_size=1),
            Conv2dBN(80, 192, kernel_size=3),
            nn.MaxPool2d(3, 2))
        self.features = nn.Sequential(
            InceptionA3(192, 32), InceptionA3(256, 64), InceptionA3(288, 64),
            ReductionA3(288),
            InceptionB3(768, 128), InceptionB3(768, 160),
            InceptionB3(768, 160), InceptionB3(768, 192),
            ReductionB3(768),
            InceptionC3(1280), InceptionC3(2048))
        self.dropout = nn.Dropout()
        self.fc = nn.Linear(2048, num_classes)

    def forward(self, x):
        x = self.features(self.stem(x))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(self.dropout(x))


def inceptionv3(num_classes=1000):
    return InceptionV3(num_classes)
