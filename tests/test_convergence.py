"""End-to-end learning correctness: the full stack (bucket views, hooks
disabled at world=1, SGD) memorizes a small fixed dataset."""
import torch
import torch.nn as nn

from mgwfbp_amd import models
from mgwfbp_amd.distributed_optimizer import DistributedOptimizer


def test_lenet_memorizes_fixed_batch():
    torch.manual_seed(0)
    net = models.LeNet()
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.01, momentum=0.9),
        named_parameters=list(net.named_parameters()), threshold=0)
    crit = nn.CrossEntropyLoss()
    g = torch.Generator().manual_seed(1)
    x = torch.randn(32, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (32,), generator=g)
    first = None
    for step in range(400):
        opt.zero_grad()
        loss = crit(net(x), y)
        loss.backward()
        opt.step()
        if first is None:
            first = loss.item()
    final = loss.item()
    assert final < 0.1 * first, (first, final)
    acc = (net(x).argmax(1) == y).float().mean().item()
    assert acc > 0.9, acc
