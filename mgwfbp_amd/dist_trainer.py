"""Distributed training entry — ``mgwfbp()`` (reference
dist_trainer.py:29-143), relaunched for MI355X.

Launch: ``torchrun --nproc-per-node N -m mgwfbp_amd.dist_trainer --dnn
resnet50 ...`` (env-var rendezvous; replaces mpirun + hvd.init, reference
dist_mpi.sh:12 / dist_trainer.py:133). One process per GPU; collectives
go over RCCL/xGMI through the comm core.
"""
from __future__ import annotations

import argparse
import os
import time

import torch

from . import comm
from . import settings
from .settings import logger
from .compression import compressors
from .dl_trainer import DLTrainer
from .distributed_optimizer import (DistributedOptimizer,
                                    broadcast_parameters)
from .profiling import benchmark


def mgwfbp(dnn, dataset, data_dir, nworkers, lr, batch_size, nsteps_update,
           max_epochs, nwpernode=8, pretrain=None, num_steps=1,
           compressor='none', density=1.0, threshold=0, writer=None,
           save_epochs=0, max_iters=None):
    rank = comm.rank()
    if torch.cuda.is_available():
        torch.cuda.set_device(comm.local_rank() % torch.cuda.device_count())
    if rank != 0:
        pretrain = None
    trainer = DLTrainer(rank, nworkers, dist=False, batch_size=batch_size,
                        is_weak_scaling=True, ngpus=1, data_dir=data_dir,
                        dataset=dataset, dnn=dnn, lr=lr, nworkers=nworkers,
                        prefix=settings.PREFIX, pretrain=pretrain,
                        num_steps=num_steps, tb_writer=writer)
    init_epoch = torch.tensor([trainer.train_epoch], dtype=torch.int64)
    init_iter = torch.tensor([trainer.train_iter], dtype=torch.int64)
    if comm.size() > 1:
        dev = trainer.device if comm.backend_name() == 'rccl' else 'cpu'
        init_epoch = init_epoch.to(dev)
        init_iter = init_iter.to(dev)
        comm.broadcast(init_epoch, root_rank=0)
        comm.broadcast(init_iter, root_rank=0)
    trainer.train_epoch = int(init_epoch[0])
    trainer.train_iter = int(init_iter[0])

    # layer-wise backward profile feeding the merge solver (reference
    # dist_trainer.py:45-46); rank 0's times are authoritative
    seq_layernames, layerwise_times = None, None
    if settings.ADAPTIVE_MERGE:
        seq_layernames, layerwise_times, _ = benchmark(trainer)
        if comm.size() > 1:
            t = torch.tensor(layerwise_times, dtype=torch.float64)
            if comm.backend_name() == 'rccl':
                t = t.to(trainer.device)
            comm.broadcast(t, root_rank=0)
            layerwise_times = [float(x) for x in t.cpu()]
        if rank == 0:
            logger.info('profiled %d layers; total backward %.6fs',
                        len(seq_layernames), sum(layerwise_times))

    is_sparse = density < 1
    compression = compressors.get(compressor, compressors['none'])
    norm_clip = None
    if dnn in ('lstm', 'lstman4'):
        norm_clip = 0.25 if dnn == 'lstm' else 400

    optimizer = DistributedOptimizer(
        trainer.optimizer,
        named_parameters=list(trainer.net.named_parameters()),
        compression=compression, is_sparse=is_sparse, density=density,
        seq_layernames=seq_layernames, layerwise_times=layerwise_times,
        norm_clip=norm_clip, threshold=threshold, writer=writer)
    trainer.update_optimizer(optimizer)

    broadcast_parameters(trainer.net.state_dict(), root_rank=0)

    iters_per_epoch = trainer.num_batches_per_epoch
    times = []
    display = 40 if iters_per_epoch > 40 else iters_per_epoch - 1
    display = max(display, 1)
    hidden = trainer.net.init_hidden() if dnn == 'lstm' else None
    nupdates = (max_iters if max_iters is not None
                else max_epochs * iters_per_epoch)
    # MGX_HIP_GRAPH=1: capture the step as a hipGraph after a few eager
    # iterations and replay it (LR schedule flows through the fused
    # SGD's device LR buffer). RNN workloads and gradient accumulation
    # stay eager.
    want_graph = (os.environ.get('MGX_HIP_GRAPH', '0') == '1'
                  and torch.cuda.is_available()
                  and dnn not in ('lstm', 'lstman4')
                  and nsteps_update == 1)
    gstep = None
    start_iter = trainer.train_iter
    for i in range(start_iter, nupdates):
        s = time.time()
        if want_graph and gstep is None and i >= start_iter + 3:
            try:
                from .graph_step import GraphedTrainStep
                gstep = GraphedTrainStep(trainer, optimizer)
                trainer.train_iter += 2   # capture warmup ran 2 steps
            except Exception as e:
                logger.warning('hipGraph capture failed (%s); staying '
                               'eager', e)
                want_graph = False
        if gstep is not None:
            lr = trainer.adjust_learning_rate(trainer.train_epoch,
                                              optimizer)
            gstep.set_lr(lr)
            gstep.step()
            trainer.train_iter += 1
            if trainer.train_iter % iters_per_epoch == 0:
                trainer.train_epoch += 1
            optimizer.train_epoch = trainer.train_epoch
            times.append(time.time() - s)
        else:
            optimizer.zero_grad()
            for j in range(nsteps_update):
                optimizer.local = (j < nsteps_update - 1)
                _, hidden = trainer.train(1, hidden=hidden)
            if dnn in ('lstm', 'lstman4') and comm.size() <= 1:
                # single-process: no hooks fire, so the optimizer's
                # internal device-side per-group clip never runs — clip
                # externally (reference dist_trainer.py:89-94). At
                # world>1 norm_clip is honored inside synchronize()
                # (device-side, no .item(); the reference factory
                # dropped the arg — SURVEY.md §7.5).
                torch.nn.utils.clip_grad_norm_(trainer.net.parameters(),
                                               norm_clip)
            trainer.update_model()
            optimizer.train_epoch = trainer.train_epoch
            times.append(time.time() - s)
        if i % display == 0 and i > 0 and rank == 0:
            avg = sum(times[-display:]) / min(len(times), display)
            logger.info('Time per iteration including communication: %f, '
                        'Speed: %f images/s', avg,
                        batch_size * nsteps_update * nworkers / avg)
        if save_epochs and rank == 0 and i > 0 \
                and i % (iters_per_epoch * save_epochs) == 0:
            trainer.save_checkpoint()
    return trainer


def main():
    parser = argparse.ArgumentParser(
        description='MI355X merged-gradient WFBP distributed trainer')
    parser.add_argument('--batch-size', type=int, default=32)
    parser.add_argument('--nsteps-update', type=int, default=1)
    parser.add_argument('--nworkers', type=int, default=None,
                        help='defaults to WORLD_SIZE')
    parser.add_argument('--nwpernode', type=int, default=8)
    parser.add_argument('--dataset', type=str, default='cifar10',
                        choices=['imagenet', 'cifar10', 'mnist', 'ptb',
                                 'an4'])
    parser.add_argument('--dnn', type=str, default='resnet20')
    parser.add_argument('--data-dir', type=str, default='')
    parser.add_argument('--lr', type=float, default=0.1)
    parser.add_argument('--num-steps', type=int, default=35)
    parser.add_argument('--max-epochs', type=int,
                        default=settings.MAX_EPOCHS)
    parser.add_argument('--max-iters', type=int, default=None)
    parser.add_argument('--pretrain', type=str, default=None)
    parser.add_argument('--compressor', type=str, default='none')
    parser.add_argument('--density', type=float, default=1.0)
    parser.add_argument('--threshold', type=int, default=0,
                        help='merge threshold in elements when '
                             'ADAPTIVE_MERGE is off (0 = WFBP)')
    parser.add_argument('--save-epochs', type=int, default=0)
    args = parser.parse_args()

    comm.init()
    nworkers = args.nworkers or comm.size()
    logdir = 'logs/%s-n%d-bs%d-lr%.4f-ns%d-ds%.5f' % (
        args.dnn, nworkers, args.batch_size, args.lr, args.nsteps_update,
        args.density)
    settings.add_file_handler(os.path.join(
        logdir, '%s-rank%d.log' % (settings.hostname, comm.rank())))
    logger.info('configurations: %s', args)
    writer = None
    if settings.TENSORBOARD and comm.rank() == 0:
        try:
            from torch.utils.tensorboard import SummaryWriter
            writer = SummaryWriter(log_dir=logdir)
        except ImportError:
            logger.warning('MGX_TENSORBOARD set but the tensorboard '
                           'package is not installed')
    mgwfbp(args.dnn, args.dataset, args.data_dir, nworkers, args.lr,
           args.batch_size, args.nsteps_update, args.max_epochs,
           nwpernode=args.nwpernode, pretrain=args.pretrain,
           num_steps=args.num_steps, compressor=args.compressor,
           density=args.density, threshold=args.threshold,
           writer=writer, save_epochs=args.save_epochs,
           max_iters=args.max_iters)
    comm.shutdown()


if __name__ == '__main__':
    main()
