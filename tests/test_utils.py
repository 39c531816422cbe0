"""Dense-alias helpers + channels_last bucket views (CPU)."""
import copy

import torch
import torch.nn as nn

from mgwfbp_amd import utils
from mgwfbp_amd import models
from mgwfbp_amd.distributed_optimizer import DistributedOptimizer


class TestDenseAlias:
    def test_contiguous_alias(self):
        t = torch.arange(24., requires_grad=False).view(2, 3, 4)
        a = utils.dense_flat_alias(t)
        assert torch.equal(a, torch.arange(24.))

    def test_channels_last_alias_is_storage_order(self):
        t = torch.randn(2, 3, 4, 5).to(memory_format=torch.channels_last)
        a = utils.dense_flat_alias(t)
        # storage order = NHWC
        assert torch.equal(a.view(2, 4, 5, 3), t.permute(0, 2, 3, 1))

    def test_non_dense_rejected(self):
        t = torch.randn(4, 4)[:, :2]   # strided slice, not dense
        try:
            utils.dense_flat_alias(t)
            assert False, 'expected ValueError'
        except ValueError:
            pass

    def test_grad_view_like_channels_last(self):
        p = torch.randn(2, 3, 4, 5).to(memory_format=torch.channels_last)
        flat = torch.zeros(p.numel())
        v = utils.grad_view_like(flat, p)
        assert v.shape == p.shape
        assert v.stride() == p.stride()
        v.copy_(p)
        # flat buffer holds NHWC order
        assert torch.equal(flat.view(2, 4, 5, 3), p.permute(0, 2, 3, 1))


class TestChannelsLastOptimizer:
    def test_step_matches_plain_sgd_channels_last(self):
        torch.manual_seed(0)
        net_a = models.resnet20()
        net_b = copy.deepcopy(net_a)
        net_b = net_b.to(memory_format=torch.channels_last)
        opt_a = torch.optim.SGD(net_a.parameters(), lr=0.1, momentum=0.9,
                                weight_decay=1e-4)
        opt_b = DistributedOptimizer(
            torch.optim.SGD(net_b.parameters(), lr=0.1, momentum=0.9,
                            weight_decay=1e-4),
            named_parameters=list(net_b.named_parameters()), threshold=0)
        crit = nn.CrossEntropyLoss()
        for step in range(2):
            g = torch.Generator().manual_seed(step)
            x = torch.randn(4, 3, 32, 32, generator=g)
            y = torch.randint(0, 10, (4,), generator=g)
            opt_a.zero_grad()
            crit(net_a(x), y).backward()
            opt_a.step()
            opt_b.zero_grad()
            crit(net_b(x.to(memory_format=torch.channels_last)),
                 y).backward()
            opt_b.step()
        for pa, pb in zip(net_a.parameters(), net_b.parameters()):
            assert torch.allclose(pa, pb, atol=1e-5), \
                (pa - pb).abs().max().item()
