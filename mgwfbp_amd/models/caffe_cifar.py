"""Caffe cifar10_quick-style net (reference models/caffe_cifar.py:57)."""
import torch.nn as nn
import torch.nn.functional as F


class CaffeCifar(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 32, 5, padding=2)
        self.conv2 = nn.Conv2d(32, 32, 5, padding=2)
        self.conv3 = nn.Conv2d(32, 64, 5, padding=2)
        self.fc1 = nn.Linear(64 * 4 * 4, 64)
        self.fc2 = nn.Linear(64, num_classes)

    def forward(self, x):
        x = F.max_pool2d(F.relu(self.conv1(x)), 2)
        x = F.avg_pool2d(F.relu(self.conv2(x)), 2)
        x = F.avg_pool2d(F.relu(self.conv3(x)), 2)
        x = F.relu(self.fc1(x.flatten(1)))
        return self.fc2(x)


def caffe_cifar(num_classes=10):
    return CaffeCifar(num_classes)
