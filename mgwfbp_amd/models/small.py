"""Small models: LeNet (CIFAR), MnistNet, FCN5Net, LinearRegression.

Reference equivalents: models/lenet.py:5-24, dl_trainer.py:65-82
(MnistNet), models/fcn.py:9-35.
"""
import torch.nn as nn
import torch.nn.functional as F


class LeNet(nn.Module):
    """Classic LeNet-5 on 3x32x32 CIFAR input (reference lenet.py:5-24)."""

    def __init__(self, num_classes=10):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 6, 5)
        self.conv2 = nn.Conv2d(6, 16, 5)
        self.fc1 = nn.Linear(16 * 5 * 5, 120)
        self.fc2 = nn.Linear(120, 84)
        self.fc3 = nn.Linear(84, num_classes)

    def forward(self, x):
        out = F.max_pool2d(F.relu(self.conv1(x)), 2)
        out = F.max_pool2d(F.relu(self.conv2(out)), 2)
        out = out.flatten(1)
        out = F.relu(self.fc1(out))
        out = F.relu(self.fc2(out))
        return self.fc3(out)


class MnistNet(nn.Module):
    """Two-conv MNIST net (reference dl_trainer.py:65-82)."""

    def __init__(self, num_classes=10):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 10, kernel_size=5)
        self.conv2 = nn.Conv2d(10, 20, kernel_size=5)
        self.fc1 = nn.Linear(320, 50)
        self.fc2 = nn.Linear(50, num_classes)

    def forward(self, x):
        x = F.relu(F.max_pool2d(self.conv1(x), 2))
        x = F.relu(F.max_pool2d(self.conv2(x), 2))
        x = x.flatten(1)
        x = F.relu(self.fc1(x))
        return self.fc2(x)


class FCN5Net(nn.Module):
    """5-layer fully-connected MNIST net (reference fcn.py:9-25)."""

    def __init__(self, num_classes=10):
        super().__init__()
        self.fc1 = nn.Linear(28 * 28, 2048)
        self.fc2 = nn.Linear(2048, 4096)
        self.fc3 = nn.Linear(4096, 1024)
        self.fc4 = nn.Linear(1024, num_classes)

    def forward(self, x):
        x = x.flatten(1)
        x = F.relu(self.fc1(x))
        x = F.relu(self.fc2(x))
        x = F.relu(self.fc3(x))
        return self.fc4(x)


class LinearRegression(nn.Module):
    """Linear model (reference fcn.py:28-35, dnn='lr'; the reference
    config pairs it with MNIST, so inputs are flattened)."""

    def __init__(self, in_features=784, out_features=10):
        super().__init__()
        self.linear = nn.Linear(in_features, out_features)

    def forward(self, x):
        return self.linear(x.flatten(1))
