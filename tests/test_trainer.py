"""DLTrainer CPU tests: training step, LR schedules, checkpoint
save/resume, evaluate parsing, synthetic datasets for every family."""
import math
import os

import pytest
import torch

from mgwfbp_amd.dl_trainer import DLTrainer, create_net, train_with_single
from mgwfbp_amd import evaluate as ev


def _trainer(**kw):
    args = dict(rank=0, size=1, dist=False, batch_size=4, ngpus=0,
                data_dir='', dataset='cifar10', dnn='lenet', lr=0.1,
                nworkers=1, prefix='t', synthetic=True)
    args.update(kw)
    return DLTrainer(**args)


class TestTrainStep:
    def test_loss_decreases_lenet(self):
        torch.manual_seed(0)
        t = _trainer()
        first = None
        for i in range(8):
            t.zero_grad()
            loss, _ = t.train(1)
            t.update_model()
            if first is None:
                first = loss
        assert loss < first * 1.5  # moves; not diverging

    def test_ptb_lstm_step(self):
        t = _trainer(dataset='ptb', dnn='lstm', lr=1.0, batch_size=4)
        t.net.batch_size = 4
        hidden = t.net.init_hidden()
        t.zero_grad()
        loss, hidden = t.train(1, hidden=hidden)
        t.update_model()
        assert math.isfinite(loss)

    def test_an4_ctc_step(self):
        t = _trainer(dataset='an4', dnn='lstman4', lr=1e-4, batch_size=2)
        t.zero_grad()
        loss, _ = t.train(1)
        t.update_model()
        assert math.isfinite(loss)

    def test_mnist_fcn5(self):
        t = _trainer(dataset='mnist', dnn='fcn5net')
        t.zero_grad()
        loss, _ = t.train(1)
        t.update_model()
        assert math.isfinite(loss)

    def test_test_method_accuracy(self):
        t = _trainer()
        acc = t.test(0)
        assert 0.0 <= acc <= 1.0


class TestLRSchedules:
    def test_general_warmup_then_steps(self):
        t = _trainer(dnn='resnet20')
        t.train_iter = 0
        lr0 = t.adjust_learning_rate(0, t.optimizer)
        assert lr0 < t.base_lr           # warmup ramp
        import mgwfbp_amd.settings as s
        old = s.WARMUP
        s.WARMUP = False
        try:
            assert t.adjust_learning_rate(0, t.optimizer) == t.base_lr
            assert t.adjust_learning_rate(100, t.optimizer) == \
                pytest.approx(t.base_lr * 0.1)
            assert t.adjust_learning_rate(130, t.optimizer) == \
                pytest.approx(t.base_lr * 0.01)
            assert t.adjust_learning_rate(170, t.optimizer) == \
                pytest.approx(t.base_lr * 0.001)
        finally:
            s.WARMUP = old

    def test_lstman4_anneal(self):
        t = _trainer(dataset='an4', dnn='lstman4', lr=2e-4, batch_size=2)
        base = t.lr
        t.train_iter = t.num_batches_per_epoch  # epoch 1
        t.adjust_learning_rate(1, t.optimizer)
        assert t.lr == pytest.approx(base / 1.01)

    def test_vgg_halving(self):
        t = _trainer(dnn='vgg16', dataset='cifar10')
        t._adjust_learning_rate_vgg16(25, t.optimizer)
        assert t.lr == pytest.approx(t.base_lr / 2)
        t._adjust_learning_rate_vgg16(50, t.optimizer)
        assert t.lr == pytest.approx(t.base_lr / 4)

    def test_cosine(self):
        import mgwfbp_amd.settings as s
        old = s.WARMUP
        s.WARMUP = False
        try:
            t = _trainer()
            t._adjust_learning_rate_cosine(14, t.optimizer)
            assert t.lr == pytest.approx(t.base_lr)
        finally:
            s.WARMUP = old


class TestCheckpoint:
    def test_save_load_roundtrip(self, tmp_path):
        t = _trainer()
        t.zero_grad()
        t.train(1)
        t.update_model()
        t.train_epoch = 3
        t.train_iter = 42
        fn = t.save_checkpoint(
            filename=str(tmp_path / 'lenet-rank0-epoch3.pth'))
        t2 = _trainer()
        t2.load_model_from_file(fn)
        assert t2.train_epoch == 3
        assert t2.train_iter == 42
        for pa, pb in zip(t.net.parameters(), t2.net.parameters()):
            assert torch.equal(pa, pb)

    def test_checkpoint_path_contract(self, tmp_path):
        # reference filename contract: <dnn>-rank<r>-epoch<e>.pth under
        # <dnn>-n<P>-bs<B>-lr<LR> (reference dl_trainer.py:769-777)
        t = _trainer()
        t.train_epoch = 5
        path = t.get_checkpoint_path(weights_dir=str(tmp_path))
        assert path.endswith('lenet-rank0-epoch5.pth')
        assert 'lenet-n1-bs4-lr' in path

    def test_evaluate_parse_rundir(self):
        dnn, p, bs, lr = ev.parse_rundir('/x/resnet20-n8-bs32-lr0.1000')
        assert (dnn, p, bs) == ('resnet20', 8, 32)
        assert lr == pytest.approx(0.1)


class TestCreateNet:
    def test_unknown_raises(self):
        with pytest.raises(ValueError):
            create_net(10, 'nosuchnet')

    def test_lstman4_returns_ext(self):
        net, ext = create_net(29, 'lstman4', datapath='')
        assert ext is not None and 'labels' in ext


class TestElastic:
    def test_update_nworker_rebuilds_sampler(self):
        t = _trainer()
        t.update_nworker(4, new_rank=2)
        assert t.rank == 2
        assert t.nworkers == 4
        assert t.train_sampler is not None
        assert t.train_sampler.num_replicas == 4
