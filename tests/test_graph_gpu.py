"""hipGraph training-step capture/replay numerics (GPU)."""
import copy

import pytest
import torch

pytestmark = pytest.mark.gpu


def _make(dnn='resnet20', bs=8):
    from mgwfbp_amd.dl_trainer import DLTrainer
    from mgwfbp_amd.distributed_optimizer import DistributedOptimizer
    t = DLTrainer(0, 1, dist=False, batch_size=bs, ngpus=1, data_dir='',
                  dataset='cifar10', dnn=dnn, lr=0.05, nworkers=1,
                  prefix='g', synthetic=True, dtype='fp32')
    opt = DistributedOptimizer(
        t.optimizer, named_parameters=list(t.net.named_parameters()),
        threshold=0)
    t.update_optimizer(opt)
    return t, opt


def test_replay_matches_eager_sequence():
    """Graphed replays over the same data sequence must track the eager
    step sequence (params, momentum, BN running stats all updated
    in-graph)."""
    from mgwfbp_amd.graph_step import GraphedTrainStep
    torch.manual_seed(0)
    te, oe = _make()
    torch.manual_seed(0)
    tg, og = _make()
    for pa, pb in zip(te.net.parameters(), tg.net.parameters()):
        assert torch.equal(pa, pb)

    gstep = GraphedTrainStep(tg, og, warmup=2)
    # capture warmup mutated tg's state; bring the EAGER side to tg's
    # state by copying IN-PLACE (the captured graph bakes tg's device
    # pointers, so tg's tensors must never be replaced)
    te.net.load_state_dict(tg.net.state_dict())
    for pa, pb in zip((p for g in oe.param_groups for p in g['params']),
                      (p for g in og.param_groups for p in g['params'])):
        buf_b = og.state.get(pb, {}).get('momentum_buffer')
        if buf_b is not None:
            oe.state.setdefault(pa, {})
            if 'momentum_buffer' not in oe.state[pa]:
                oe.state[pa]['momentum_buffer'] = buf_b.clone()
            else:
                oe.state[pa]['momentum_buffer'].copy_(buf_b)
    # fixed batch sequence
    batches = [tg.fetch_data() for _ in range(3)]

    for x, y in batches:
        oe.zero_grad()
        with te.autocast():
            loss = te.criterion(te.net(x), y)
        loss.backward()
        te.update_model()
    for x, y in batches:
        gstep.static_x.copy_(x)
        gstep.static_y.copy_(y)
        gstep.graph.replay()
    torch.cuda.synchronize()

    # tolerance covers MIOpen conv-backward run-to-run nondeterminism
    # (atomics in wgrad), not the graph mechanism itself
    for (na, pa), (nb, pb) in zip(te.net.named_parameters(),
                                  tg.net.named_parameters()):
        assert torch.allclose(pa, pb, atol=5e-3, rtol=5e-3), \
            (na, (pa - pb).abs().max().item())
    # BN running stats advanced identically
    sda, sdb = te.net.state_dict(), tg.net.state_dict()
    for k in sda:
        if 'running_' in k:
            assert torch.allclose(sda[k], sdb[k], atol=1e-3, rtol=1e-3), k


def test_replay_loss_finite_and_params_move():
    from mgwfbp_amd.graph_step import GraphedTrainStep
    t, o = _make('lenet')
    gstep = GraphedTrainStep(t, o, warmup=2)
    before = torch.cat([p.detach().reshape(-1).clone()
                        for p in t.net.parameters()])
    for _ in range(5):
        gstep.step()
    torch.cuda.synchronize()
    after = torch.cat([p.detach().reshape(-1)
                       for p in t.net.parameters()])
    assert torch.isfinite(torch.tensor(gstep.loss()))
    assert not torch.allclose(before, after)


def test_lr_buffer_schedule_across_replays():
    """The LR schedule must take effect across graph replays through
    the fused SGD's device LR buffer (no re-capture)."""
    from mgwfbp_amd.graph_step import GraphedTrainStep
    t, o = _make('lenet')
    gstep = GraphedTrainStep(t, o, warmup=2)
    assert o._fused_sgd is not None

    p0 = torch.cat([p.detach().reshape(-1).clone()
                    for p in t.net.parameters()])
    gstep.set_lr(0.0)          # zero LR -> replay must not move params
    gstep.step()
    torch.cuda.synchronize()
    p1 = torch.cat([p.detach().reshape(-1)
                    for p in t.net.parameters()])
    assert torch.equal(p0, p1), 'params moved despite lr=0'
    gstep.set_lr(0.05)
    gstep.step()
    torch.cuda.synchronize()
    p2 = torch.cat([p.detach().reshape(-1)
                    for p in t.net.parameters()])
    assert not torch.allclose(p0, p2), 'params frozen with lr>0'
