"""Checkpoint evaluator (reference evaluate.py:10-75).

Walks epoch checkpoints saved by DLTrainer (filename contract
``<dnn>-rank<r>-epoch<e>.pth`` under ``<dnn>-n<P>-bs<B>-lr<LR>``), runs
``test()`` per epoch and tracks the best metric (lower-is-better for
lstm/lstman4 WER/perplexity).
"""
from __future__ import annotations

import argparse
import os
import re

import torch

from .dl_trainer import DLTrainer
from .settings import logger


def model_average(trainer, weight_files):
    """Average N ranks' state dicts (reference evaluate.py:10-18)."""
    state = None
    for fn in weight_files:
        ckpt = torch.load(fn, map_location=trainer.device,
                          weights_only=False)
        sd = ckpt['state'] if 'state' in ckpt else ckpt
        if state is None:
            state = {k: v.clone().float() if torch.is_tensor(v) else v
                     for k, v in sd.items()}
        else:
            for k, v in sd.items():
                if torch.is_tensor(v) and v.is_floating_point():
                    state[k] += v.float()
    n = len(weight_files)
    for k, v in state.items():
        if torch.is_tensor(v) and v.is_floating_point():
            state[k] = v / n
    return state


def parse_rundir(path):
    """Parse "<dnn>-n<P>-bs<B>-lr<LR>" out of a weights dir name
    (reference evaluate.py:21-24)."""
    name = os.path.basename(os.path.normpath(path))
    m = re.match(r'(.+)-n(\d+)-bs(\d+)-lr([0-9.]+)', name)
    if not m:
        raise ValueError('cannot parse run dir name: %s' % name)
    return m.group(1), int(m.group(2)), int(m.group(3)), float(m.group(4))


def evaluate(weights_dir, dataset='cifar10', data_dir='', start_epoch=0,
             nepochs=None, rank=0):
    dnn, nworkers, bs, lr = parse_rundir(weights_dir)
    trainer = DLTrainer(0, 1, dist=False, batch_size=bs,
                        is_weak_scaling=True, ngpus=1 if
                        torch.cuda.is_available() else 0,
                        data_dir=data_dir, dataset=dataset, dnn=dnn, lr=lr,
                        nworkers=1, prefix='eval')
    lower_is_better = dnn in ('lstm', 'lstman4')
    best = float('inf') if lower_is_better else -float('inf')
    best_epoch = -1
    epoch = start_epoch
    while True:
        fn = os.path.join(weights_dir,
                          '%s-rank%d-epoch%d.pth' % (dnn, rank, epoch))
        if not os.path.isfile(fn):
            if nepochs is not None and epoch < nepochs:
                epoch += 1
                continue
            break
        trainer.load_model_from_file(fn)
        metric = trainer.test(epoch)
        if (metric < best) == lower_is_better or \
                (not lower_is_better and metric > best):
            best = metric
            best_epoch = epoch
        epoch += 1
        if nepochs is not None and epoch > nepochs:
            break
    logger.info('best metric %.4f at epoch %d', best, best_epoch)
    return best, best_epoch


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument('--weights-dir', type=str, required=True)
    parser.add_argument('--dataset', type=str, default='cifar10')
    parser.add_argument('--data-dir', type=str, default='')
    parser.add_argument('--start-epoch', type=int, default=0)
    parser.add_argument('--nepochs', type=int, default=None)
    args = parser.parse_args()
    evaluate(args.weights_dir, args.dataset, args.data_dir,
             args.start_epoch, args.nepochs)


if __name__ == '__main__':
    main()
