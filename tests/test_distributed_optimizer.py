"""DistributedOptimizer unit tests (CPU, world=1)."""
import copy

import pytest
import torch
import torch.nn as nn

from mgwfbp_amd import models
from mgwfbp_amd.distributed_optimizer import DistributedOptimizer


def _make_lenet(seed=0):
    torch.manual_seed(seed)
    return models.LeNet()


def _data(seed=0, n=8):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (n,), generator=g)
    return x, y


class TestSingleProcess:
    def test_step_matches_plain_sgd(self):
        # world=1: the wrapped optimizer must produce the same update as
        # plain torch SGD on the same model/data
        net_a = _make_lenet(0)
        net_b = copy.deepcopy(net_a)
        opt_a = torch.optim.SGD(net_a.parameters(), lr=0.1, momentum=0.9)
        opt_b = torch.optim.SGD(net_b.parameters(), lr=0.1, momentum=0.9)
        opt_b = DistributedOptimizer(
            opt_b, named_parameters=list(net_b.named_parameters()),
            threshold=0)
        crit = nn.CrossEntropyLoss()
        for step in range(3):
            x, y = _data(step)
            opt_a.zero_grad()
            crit(net_a(x), y).backward()
            opt_a.step()
            opt_b.zero_grad()
            crit(net_b(x), y).backward()
            opt_b.step()
        for pa, pb in zip(net_a.parameters(), net_b.parameters()):
            assert torch.allclose(pa, pb, atol=1e-6), 'params diverged'

    def test_grads_are_views_into_group_buffers(self):
        net = _make_lenet(0)
        opt = DistributedOptimizer(
            torch.optim.SGD(net.parameters(), lr=0.1),
            named_parameters=list(net.named_parameters()), threshold=1 << 30)
        # single group: every grad's storage is the flat buffer's storage
        assert len(opt._merged_parameters) == 1
        flat = next(iter(opt._merged_parameters.values()))
        for p in net.parameters():
            assert p.grad is not None
            assert p.grad.untyped_storage().data_ptr() == \
                flat.untyped_storage().data_ptr()

    def test_group_key_format(self):
        net = _make_lenet(0)
        names = [k for k, _ in net.named_parameters()]
        opt = DistributedOptimizer(
            torch.optim.SGD(net.parameters(), lr=0.1),
            named_parameters=list(net.named_parameters()), threshold=1 << 30)
        key = next(iter(opt._merged_parameters))
        # reference ':'.join(names) in backward order (reference
        # distributed_optimizer.py:285-288)
        assert key == ':'.join(reversed(names))

    def test_zero_grad_zeroes_buffers(self):
        net = _make_lenet(0)
        opt = DistributedOptimizer(
            torch.optim.SGD(net.parameters(), lr=0.1),
            named_parameters=list(net.named_parameters()), threshold=0)
        x, y = _data(0)
        nn.CrossEntropyLoss()(net(x), y).backward()
        assert any(p.grad.abs().sum() > 0 for p in net.parameters())
        opt.zero_grad()
        for p in net.parameters():
            assert p.grad.abs().sum() == 0
            # views must survive zero_grad
            assert p.grad.numel() == p.numel()

    def test_mgwfbp_solver_path(self):
        net = _make_lenet(0)
        names = [k for k, _ in net.named_parameters()]
        times = [1e-4] * len(names)
        opt = DistributedOptimizer(
            torch.optim.SGD(net.parameters(), lr=0.1),
            named_parameters=list(net.named_parameters()),
            seq_layernames=names, layerwise_times=times)
        assert len(opt._groups) >= 1
        flat_names = [k for g in opt._groups for k in g]
        assert sorted(flat_names) == sorted(names)

    def test_gradient_accumulation_local_flag(self):
        net = _make_lenet(0)
        opt = DistributedOptimizer(
            torch.optim.SGD(net.parameters(), lr=0.1),
            named_parameters=list(net.named_parameters()), threshold=0)
        crit = nn.CrossEntropyLoss()
        opt.zero_grad()
        opt.local = True
        x, y = _data(0)
        crit(net(x), y).backward()
        g1 = {id(p): p.grad.clone() for p in net.parameters()}
        opt.local = False
        x, y = _data(1)
        crit(net(x), y).backward()
        for p in net.parameters():
            # second backward accumulated on top of the first
            assert not torch.allclose(p.grad, g1[id(p)])
