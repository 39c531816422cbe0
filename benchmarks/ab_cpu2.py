"""Measured mgwfbp-vs-wfbp-vs-single A/B at world=2 over gloo (CPU).

The only world>=2 this pool can reach (RCCL refuses same-device ranks,
profiles/multirank_blocker.md). gloo's alpha is O(100us) — an order
closer to the reference's 10GbE regime than to xGMI — so this is a
direct, measured test of the solver's merge logic on real collectives:
the reference's batch methodology (reference batch_dist_mpi.sh:2) at
P=2.

Run:  python benchmarks/ab_cpu2.py [--model resnet20] [--steps 10]
Spawns 2 ranks itself; prints one JSON line per arm.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.multiprocessing as mp


def worker(rank, world, port, arm, model, steps, batch_size, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['RANK'] = str(rank)
    os.environ['MGX_COMM_BACKEND'] = 'gloo'
    os.environ['MGX_ADAPTIVE_MERGE'] = '1' if arm == 'mgwfbp' else '0'
    torch.set_num_threads(max(1, (os.cpu_count() or 2) // (2 * world)))
    import mgwfbp_amd.comm as comm
    from mgwfbp_amd import models
    from mgwfbp_amd.distributed_optimizer import (DistributedOptimizer,
                                                  broadcast_parameters)
    import torch.nn as nn
    comm.init()
    torch.manual_seed(1234)
    net = models.__dict__[model](num_classes=10)
    broadcast_parameters(net.state_dict(), root_rank=0)
    threshold = 1 << 40 if arm == 'single' else 0
    kwargs = {}
    if arm == 'mgwfbp':
        # measure gloo's real alpha/beta + per-call host cost (the GPU
        # path does this online in _benchmark_communication; on CPU
        # ADAPTIVE_ABC is gated off, so do it here)
        from mgwfbp_amd.profiling import CommunicationProfiler
        from mgwfbp_amd import solver as _solver
        from mgwfbp_amd import settings as _settings
        prof = CommunicationProfiler(comm.allreduce_async_,
                                     comm.synchronize)
        prof.sizes = [1024 * i for i in (1, 4, 16, 64, 256)] + [2 ** 20]
        sizes, times_s = prof.benchmark(num_iters=5)
        a, b = _solver.fit_alpha_beta([s * 4 for s in sizes], times_s)
        a_host = prof.benchmark_host_overhead(num_calls=50, numel=65536)
        _settings.ALPHA_HOST = a_host
        kwargs = dict(alpha=a, beta=b)
        # profile backward on rank 0's timeline: one quick pass
        names = [k for k, _ in net.named_parameters()]
        # uniform synthetic times scaled to a measured total: enough for
        # the solver to see the alpha-dominated regime on CPU
        crit = nn.CrossEntropyLoss()
        x = torch.randn(batch_size, 3, 32, 32)
        y = torch.randint(0, 10, (batch_size,))
        for _ in range(2):
            net.zero_grad()
            crit(net(x), y).backward()
        t0 = time.time()
        net.zero_grad()
        crit(net(x), y).backward()
        total_bwd = time.time() - t0
        per = total_bwd / len(names)
        kwargs.update(seq_layernames=names,
                      layerwise_times=[per] * len(names))
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9),
        named_parameters=list(net.named_parameters()),
        threshold=threshold, **kwargs)
    crit = nn.CrossEntropyLoss()

    def step(seed):
        g = torch.Generator().manual_seed(seed)
        x = torch.randn(batch_size, 3, 32, 32, generator=g)
        y = torch.randint(0, 10, (batch_size,), generator=g)
        opt.zero_grad()
        crit(net(x), y).backward()
        opt.step()

    for i in range(3):
        step(i)
    comm.barrier()
    t0 = time.time()
    for i in range(steps):
        step(100 + i)
    comm.barrier()
    dt = (time.time() - t0) / steps
    if rank == 0:
        q.put({'arm': arm, 'ms_per_step': dt * 1e3,
               'ngroups': len(opt._groups),
               'alpha': opt.alpha, 'beta': opt.beta})
    comm.shutdown()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--model', default='resnet20')
    ap.add_argument('--steps', type=int, default=10)
    ap.add_argument('--batch-size', type=int, default=32)
    ap.add_argument('--port', type=int, default=29741)
    args = ap.parse_args()
    results = []
    for i, arm in enumerate(('single', 'wfbp', 'mgwfbp')):
        ctx = mp.get_context('spawn')
        q = ctx.SimpleQueue()
        procs = [ctx.Process(target=worker,
                             args=(r, 2, args.port + 2 * i, arm,
                                   args.model, args.steps,
                                   args.batch_size, q))
                 for r in range(2)]
        for p in procs:
            p.start()
        res = q.get()
        for p in procs:
            p.join(600)
        res['model'] = args.model
        res['world'] = 2
        res['backend'] = 'gloo-cpu'
        results.append(res)
        print(json.dumps(res), flush=True)
    os.makedirs('gpurun_out', exist_ok=True)
    with open('gpurun_out/ab_cpu2_%s.json' % args.model, 'w') as f:
        json.dump(results, f, indent=1)


if __name__ == '__main__':
    main()
