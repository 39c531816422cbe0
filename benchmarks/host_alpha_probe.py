"""Measure the per-collective HOST cost components on real hardware
(world=1 — the multi-rank transport is blocked on this pool, see
profiles/multirank_blocker.md, but the HOST side of a collective is
identical at any world size: python hook bookkeeping + wrapper dispatch
+ the RCCL enqueue that records the ready event, launches the collective
kernel on the comm stream and records the done event).

alpha_host = t_enqueue (core binding, launch-to-launch) + t_hook
(python per-group bookkeeping). This is the per-call constant the
MG-WFBP solver amortizes by merging; at world>1
the online sweep re-measures it in-place
(CommunicationProfiler.benchmark_host_overhead).

Writes profiles/host_alpha.json. Run on a GPU box:
    python benchmarks/host_alpha_probe.py
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def measure_core_enqueue(n=300):
    """Launch-to-launch host time of one RCCL allreduce enqueue through
    the native core (size-1 communicator: same binding, event and
    ncclAllReduce enqueue path as world=8, no cross-device transport)."""
    from mgwfbp_amd.comm import mgx_comm_ext as core
    try:
        core.init(0, 1, core.unique_id())
        owns = True
    except RuntimeError:
        owns = False
    buf = torch.randn(1 << 20, device='cuda')
    s = torch.cuda.current_stream().cuda_stream
    hids = []
    for _ in range(10):
        core.wait_handle(core.allreduce_async(buf, True, s), s)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        hids.append(core.allreduce_async(buf, True, s))
    t_enq = (time.time() - t0) / n
    for h in hids:
        core.wait_handle(h, s)
    torch.cuda.synchronize()
    # and the wait cost (stream-wait enqueue, also host-side per group)
    hids = [core.allreduce_async(buf, True, s) for _ in range(n)]
    t0 = time.time()
    for h in hids:
        core.wait_handle(h, s)
    t_wait = (time.time() - t0) / n
    torch.cuda.synchronize()
    if owns:
        core.destroy()
    return t_enq, t_wait


def measure_hook_path(n=2000):
    """Python cost of the per-layer hook body + group bookkeeping: build
    a real optimizer over resnet50's parameters (threshold 0 = one group
    per layer, the WFBP worst case) and time the hook functions with the
    collective call stubbed out."""
    from mgwfbp_amd import models
    from mgwfbp_amd.distributed_optimizer import DistributedOptimizer
    net = models.resnet50()
    if torch.cuda.is_available():
        net = net.cuda()
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.1),
        named_parameters=list(net.named_parameters()), threshold=0)
    names = opt._sequential_keys
    enqueued = [0]
    opt._allreduce_group_async = lambda gi: enqueued.__setitem__(
        0, enqueued[0] + 1)
    hooks = {k: opt._make_hook(k) for k in names}
    params = {k: opt._named_parameters[k] for k in names}
    # fire in backward order, full passes
    order = list(reversed(names))
    def reset_flags():
        # what synchronize() does at step end (flags are per-step)
        for gi in range(len(opt._groups)):
            opt._groups_flags[gi] = [0] * len(opt._groups_flags[gi])

    t0 = time.time()
    passes = max(1, n // len(order))
    for _ in range(passes):
        for k in order:
            hooks[k](params[k])
        reset_flags()
    t_hook = (time.time() - t0) / (passes * len(order))
    assert enqueued[0] == passes * len(order)
    return t_hook


def main():
    out = {'world': 1, 'device': torch.cuda.get_device_name(0)
           if torch.cuda.is_available() else 'cpu'}
    t_hook = measure_hook_path()
    out['t_hook_python_s'] = t_hook
    if torch.cuda.is_available():
        t_enq, t_wait = measure_core_enqueue()
        out['t_enqueue_core_s'] = t_enq
        out['t_wait_enqueue_s'] = t_wait
        out['alpha_host_s'] = t_enq + t_hook + t_wait
    else:
        out['alpha_host_s'] = t_hook
    os.makedirs('profiles', exist_ok=True)
    with open('profiles/host_alpha.json', 'w') as f:
        json.dump(out, f, indent=1)
    print(json.dumps(out, indent=1))


if __name__ == '__main__':
    main()
