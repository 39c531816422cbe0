"""Flagship benchmark: ResNet-50 / synthetic ImageNet weak-scaling DP.

Driver contract: ``python bench.py --gpus N --steps K --warmup W`` (for
N>1 the driver launches via torch.distributed.run, one rank per GPU).
Rank 0 prints ONE JSON line with the whole-job images/s (BASELINE.json
metric: "images/sec (whole node) + scaling eff., ResNet-50 synthetic
ImageNet").

One timed step = zero_grad + forward + backward (hooks fire merged-group
RCCL all-reduces overlapped with backward) + synchronize + optimizer
update — the reference's "Time per iteration including communication"
(reference dist_trainer.py:97-99), bf16 autocast compute, fp32 grads.

--merge selects the A/B arm: mgwfbp (solver, default) | wfbp
(threshold 0) | single (one group) | threshold:<elems>.
"""
import argparse
import json
import os
import sys
import time

import torch


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument('--gpus', type=int, default=1)
    parser.add_argument('--steps', type=int, default=60)
    parser.add_argument('--warmup', type=int, default=15)
    parser.add_argument('--model', type=str, default='resnet50')
    parser.add_argument('--batch-size', type=int, default=128)
    parser.add_argument('--dataset', type=str, default='imagenet')
    parser.add_argument('--dtype', type=str, default='bf16')
    parser.add_argument('--merge', type=str, default='mgwfbp',
                        help='mgwfbp | wfbp | single | threshold:<elems>')
    parser.add_argument('--profile-iters', type=int, default=15,
                        help='layerwise-profiling iterations for the '
                             'solver (mgwfbp arm)')
    parser.add_argument('--density', type=float, default=1.0,
                        help='<1 enables top-k sparse gradient exchange '
                             'with error feedback')
    parser.add_argument('--scaling', type=str, default='weak',
                        choices=['weak', 'strong'],
                        help='weak: per-GPU batch fixed (BASELINE '
                             'contract); strong: global batch fixed, '
                             'split over ranks')
    parser.add_argument('--graph', type=str, default='auto',
                        choices=['auto', 'on', 'off'],
                        help='hipGraph-capture the training step (full '
                             'compute incl. collectives; falls back to '
                             'eager on capture failure)')
    args = parser.parse_args()

    world_size = int(os.environ.get('WORLD_SIZE', '1'))
    rank = int(os.environ.get('RANK', '0'))
    n_gpus = max(world_size, 1)
    if args.gpus != n_gpus and world_size == 1 and args.gpus > 1:
        print('WARNING: --gpus %d but launched single-process; '
              'benchmarking N=1' % args.gpus, file=sys.stderr)
        args.gpus = 1

    os.environ.setdefault('MGX_DTYPE', args.dtype)
    if args.merge != 'mgwfbp':
        os.environ['MGX_ADAPTIVE_MERGE'] = '0'
    # Measured best (profiles/README.md): after the two-level finalize
    # tree the fused MgxBatchNorm2d(+ReLU) kernels win on both scales
    # (resnet20 +9%, resnet50 6083 vs 5965 img/s).
    os.environ.setdefault('MGX_FUSED_BN', '1')

    import mgwfbp_amd.comm as comm
    import mgwfbp_amd.settings as settings
    from mgwfbp_amd.dl_trainer import DLTrainer
    from mgwfbp_amd.distributed_optimizer import (DistributedOptimizer,
                                                  broadcast_parameters)
    from mgwfbp_amd.profiling import benchmark as profile_layers

    t_start = time.time()

    def phase(msg):
        print('[bench +%6.1fs] %s' % (time.time() - t_start, msg),
              file=sys.stderr, flush=True)

    comm.init()
    assert comm.size() == n_gpus
    phase('comm initialized (%s)' % comm.backend_name())
    if torch.cuda.is_available():
        torch.cuda.set_device(comm.local_rank()
                              % torch.cuda.device_count())

    threshold = 0
    if args.merge == 'single':
        threshold = 1 << 40
    elif args.merge.startswith('threshold:'):
        threshold = int(args.merge.split(':', 1)[1])

    trainer = DLTrainer(rank, n_gpus, dist=False,
                        batch_size=args.batch_size,
                        is_weak_scaling=(args.scaling == 'weak'),
                        ngpus=1 if torch.cuda.is_available() else 0,
                        data_dir='', dataset=args.dataset, dnn=args.model,
                        lr=0.01, nworkers=n_gpus, prefix='bench',
                        dtype=args.dtype, synthetic=True)

    phase('trainer built (%s/%s bs%d)' % (args.model, args.dataset,
                                          args.batch_size))
    seq_layernames = layerwise_times = None
    if args.merge == 'mgwfbp':
        seq_layernames, layerwise_times, _ = profile_layers(
            trainer, num_warmup=3, num_iters=args.profile_iters)
        if comm.size() > 1:
            t = torch.tensor(layerwise_times, dtype=torch.float64)
            if comm.backend_name() == 'rccl':
                t = t.to(trainer.device)
            comm.broadcast(t, root_rank=0)
            layerwise_times = [float(x) for x in t.cpu()]

    phase('layerwise profile done')
    optimizer = DistributedOptimizer(
        trainer.optimizer,
        named_parameters=list(trainer.net.named_parameters()),
        seq_layernames=seq_layernames, layerwise_times=layerwise_times,
        threshold=threshold, density=args.density)
    trainer.update_optimizer(optimizer)
    if comm.size() > 1:
        broadcast_parameters(trainer.net.state_dict(), root_rank=0)

    def eager_step():
        optimizer.zero_grad()
        trainer.train(1)
        trainer.update_model()

    one_step = eager_step
    graphed = False
    phase('optimizer ready')
    # warmup (untimed) — also tunes every MIOpen shape before capture
    for _ in range(args.warmup):
        eager_step()
    phase('warmup done (%d steps)' % args.warmup)

    # auto: graph at every world size. RCCL-collective capture is
    # hardware-validated at world=1 (tests/test_kernels_gpu.py
    # TestCommCoreBindings — capture propagates through the comm stream
    # exactly as at world=8; see profiles/multirank_blocker.md), and the
    # consensus below falls every rank back to eager if ANY rank's
    # capture fails.
    want_graph = (args.graph == 'on'
                  or (args.graph == 'auto' and torch.cuda.is_available()))
    if want_graph and torch.cuda.is_available():
        from mgwfbp_amd.graph_step import GraphedTrainStep
        try:
            gstep = GraphedTrainStep(trainer, optimizer)
            ok = torch.tensor([1.0], device=trainer.device)
        except Exception as e:
            print('[bench] graph capture failed (%s); eager path' % e,
                  file=sys.stderr)
            gstep = None
            ok = torch.tensor([0.0], device=trainer.device)
        # all ranks must agree before committing to the graph path
        if comm.size() > 1:
            h = comm.allreduce_async_(ok, average=False)
            comm.synchronize(h)
            torch.cuda.synchronize()
        if gstep is not None and float(ok.item()) >= comm.size():
            one_step = gstep.step
            graphed = True
            for _ in range(3):   # replay warmup
                one_step()
            phase('hipGraph captured (replay path)')
        else:
            phase('eager path (graph unavailable)')

    # timed region: barrier + sync on both sides
    comm.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        one_step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t1 = time.time()
    comm.barrier()

    elapsed = t1 - t0
    # MAX over ranks (slowest rank defines job time)
    if comm.size() > 1:
        import torch.distributed as dist
        t2 = torch.tensor([elapsed], dtype=torch.float64)
        if dist.get_backend() == 'nccl':
            t2 = t2.to(trainer.device)
        dist.all_reduce(t2, op=dist.ReduceOp.MAX)
        elapsed = float(t2.cpu()[0])

    ms_per_step = elapsed / args.steps * 1e3
    global_batch = trainer.batch_size * n_gpus
    images_per_sec = global_batch * args.steps / elapsed

    if rank == 0:
        result = {
            'metric': 'images/sec (whole node), ResNet-50 synthetic '
                      'ImageNet' if args.model == 'resnet50'
                      else 'images/sec (whole node), %s' % args.model,
            'value': images_per_sec,
            'unit': 'images/s',
            'n_gpus': n_gpus,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': ms_per_step,
            'higher_is_better': True,
            'scaling': args.scaling,
            'vs_baseline': None,
            'dtype': args.dtype,
            'data': 'synthetic',
            'config': {
                'model': args.model,
                'global_batch': global_batch,
                'per_gpu_batch': trainer.batch_size,
                'image_size': 224 if args.dataset == 'imagenet' else 32,
                'parallelism': 'dp%d' % n_gpus,
                'merge': args.merge,
                'comm_backend': comm.backend_name(),
                'hip_graph': graphed,
                'density': args.density,
            },
        }
        print(json.dumps(result))
    comm.shutdown()


if __name__ == '__main__':
    main()
