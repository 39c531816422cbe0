"""GoogLeNet / Inception-v1 (reference models/googlenet.py:18-107;
aux_logits disabled by default as in the reference factory :55)."""
import torch
import torch.nn as nn
import torch.nn.functional as F


class BasicConv2d(nn.Module):
    def __init__(self, cin, cout, **kwargs):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, bias=False, **kwargs)
        self.bn = nn.BatchNorm2d(cout)

    def forward(self, x):
        return F.relu(self.bn(self.conv(x)), inplace=True)


class Inception(nn.Module):
    def __init__(self, cin, c1, c3r, c3, c5r, c5, pool_proj):
        super().__init__()
        self.branch1 = BasicConv2d(cin, c1, kernel_size=1)
        self.branch2 = nn.Sequential(
            BasicConv2d(cin, c3r, kernel_size=1),
            BasicConv2d(c3r, c3, kernel_size=3, padding=1))
        self.branch3 = nn.Sequential(
            BasicConv2d(cin, c5r, kernel_size=1),
            BasicConv2d(c5r, c5, kernel_size=3, padding=1))
        self.branch4 = nn.Sequential(
            nn.MaxPool2d(3, stride=1, padding=1),
            BasicConv2d(cin, pool_proj, kernel_size=1))

    def forward(self, x):
        return torch.cat([self.branch1(x), self.branch2(x),
                          self.branch3(x), self.branch4(x)], 1)


class GoogLeNet(nn.Module):
    def __init__(self, num_classes=1000, aux_logits=False):
        super().__init__()
        assert not aux_logits, 'aux_logits path not used by the trainer'
        self.pre = nn.Sequential(
            BasicConv2d(3, 64, kernel_size=7, stride=2, padding=3),
            nn.MaxPool2d(3, stride=2, ceil_mode=True),
            BasicConv2d(64, 64, kernel_size=1),
            BasicConv2d(64, 192, kernel_size=3, padding=1),
            nn.MaxPool2d(3, stride=2, ceil_mode=True))
        self.a3 = Inception(192, 64, 96, 128, 16, 32, 32)
        self.b3 = Inception(256, 128, 128, 192, 32, 96, 64)
        self.maxpool = nn.MaxPool2d(3, stride=2, ceil_mode=True)
        self.a4 = Inception(480, 192, 96, 208, 16, 48, 64)
        self.b4 = Inception(512, 160, 112, 224, 24, 64, 64)
        self.c4 = Inception(512, 128, 128, 256, 24, 64, 64)
        self.d4 = Inception(512, 112, 144, 288, 32, 64, 64)
        self.e4 = Inception(528, 256, 160, 320, 32, 128, 128)
        self.a5 = Inception(832, 256, 160, 320, 32, 128, 128)
        self.b5 = Inception(832, 384, 192, 384, 48, 128, 128)
        self.dropout = nn.Dropout(0.4)
        self.fc = nn.Linear(1024, num_classes)

    def forward(self, x):
        x = self.pre(x)
        x = self.maxpool(self.b3(self.a3(x)))
        x = self.maxpool(self.e4(self.d4(self.c4(self.b4(self.a4(x))))))
        x = self.b5(self.a5(x))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(self.dropout(x))


def googlenet(num_classes=1000):
    return GoogLeNet(num_classes=num_classes)
