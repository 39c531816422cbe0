"""2-process gloo integration: full DistributedOptimizer plumbing on CPU
(BASELINE config 1 — "LeNet / MNIST world_size=2 on CPU gloo")."""
import copy
import os

import pytest
import torch
import torch.multiprocessing as mp
import torch.nn as nn


def _data(seed, n=8):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (n,), generator=g)
    return x, y


def _worker(rank, world, port, threshold, use_solver, q, comm_dtype='fp32'):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['RANK'] = str(rank)
    os.environ['MGX_COMM_BACKEND'] = 'gloo'
    os.environ['MGX_COMM_DTYPE'] = comm_dtype
    import mgwfbp_amd.comm as comm
    from mgwfbp_amd import models
    from mgwfbp_amd.distributed_optimizer import (DistributedOptimizer,
                                                  broadcast_parameters)
    comm.init()
    torch.manual_seed(42 + rank)   # intentionally different init per rank
    net = models.LeNet()
    broadcast_parameters(net.state_dict(), root_rank=0)
    names = [k for k, _ in net.named_parameters()]
    kwargs = {}
    if use_solver:
        kwargs = dict(seq_layernames=names,
                      layerwise_times=[1e-4] * len(names))
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9),
        named_parameters=list(net.named_parameters()),
        threshold=threshold, **kwargs)
    crit = nn.CrossEntropyLoss()
    for step in range(3):
        opt.zero_grad()
        x, y = _data(seed=step * world + rank)   # different shard per rank
        crit(net(x), y).backward()
        opt.step()
    flat = torch.cat([p.detach().reshape(-1) for p in net.parameters()])
    q.put((rank, flat.tolist()))
    comm.shutdown()


def _run_world(world, threshold, use_solver, port, comm_dtype='fp32'):
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker,
                         args=(r, world, port, threshold, use_solver, q,
                               comm_dtype))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, flat = q.get()
        results[rank] = torch.tensor(flat)
    for p in procs:
        p.join(120)
        assert p.exitcode == 0, 'worker failed with %s' % p.exitcode
    return results


def _reference_dp(world, steps=3):
    """Sequential emulation of synchronous DP: average shard gradients."""
    from mgwfbp_amd import models
    torch.manual_seed(42)   # rank-0 init is broadcast
    net = models.LeNet()
    opt = torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9)
    crit = nn.CrossEntropyLoss()
    for step in range(steps):
        opt.zero_grad()
        loss = 0
        for r in range(world):
            x, y = _data(seed=step * world + r)
            loss = loss + crit(net(x), y) / world
        loss.backward()
        opt.step()
    return torch.cat([p.detach().reshape(-1) for p in net.parameters()])


@pytest.mark.parametrize('threshold,use_solver,port', [
    (0, False, 29611),          # pure WFBP: per-layer groups
    (1 << 30, False, 29613),    # single group
    (0, True, 29615),           # MG-WFBP solver path
])
def test_two_process_dp_matches_sequential(threshold, use_solver, port):
    world = 2
    results = _run_world(world, threshold, use_solver, port)
    # ranks agree bit-for-bit
    assert torch.equal(results[0], results[1])
    # and match the sequential DP emulation
    ref = _reference_dp(world)
    assert torch.allclose(results[0], ref, atol=1e-5), \
        (results[0] - ref).abs().max()


def test_bf16_wire_format():
    """COMM_DTYPE=bf16: gradients cross the wire as bf16 (half the
    bytes); ranks must agree bitwise and track the fp32 DP reference to
    bf16 precision."""
    world = 2
    results = _run_world(world, 0, False, 29617, comm_dtype='bf16')
    assert torch.equal(results[0], results[1])
    ref = _reference_dp(world)
    assert torch.allclose(results[0], ref, atol=5e-2, rtol=5e-2), \
        (results[0] - ref).abs().max()


def _accum_worker(rank, world, port, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['RANK'] = str(rank)
    os.environ['MGX_COMM_BACKEND'] = 'gloo'
    import mgwfbp_amd.comm as comm
    from mgwfbp_amd import models
    from mgwfbp_amd.distributed_optimizer import (DistributedOptimizer,
                                                  broadcast_parameters)
    comm.init()
    torch.manual_seed(42 + rank)
    net = models.LeNet()
    broadcast_parameters(net.state_dict(), root_rank=0)
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9),
        named_parameters=list(net.named_parameters()), threshold=0)
    crit = nn.CrossEntropyLoss()
    # 2 microbatches per step with comm suppressed on the first
    opt.zero_grad()
    for j, seed in enumerate((rank * 2, rank * 2 + 1)):
        opt.local = (j == 0)
        x, y = _data(seed=seed)
        crit(net(x), y).backward()
    opt.local = False
    opt.step()
    flat = torch.cat([p.detach().reshape(-1) for p in net.parameters()])
    q.put((rank, flat.tolist()))
    comm.shutdown()


def test_gradient_accumulation_two_process():
    """nsteps_update-style accumulation: comm suppressed on microstep 0,
    accumulated gradient all-reduced on the final microstep."""
    world = 2
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_accum_worker, args=(r, world, 29631, q))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(world):
        rank, flat = q.get()
        res[rank] = torch.tensor(flat)
    for p in procs:
        p.join(180)
        assert p.exitcode == 0
    assert torch.equal(res[0], res[1])
    # sequential reference: average over ranks of SUMMED microbatch grads
    from mgwfbp_amd import models
    torch.manual_seed(42)
    net = models.LeNet()
    opt = torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9)
    crit = nn.CrossEntropyLoss()
    opt.zero_grad()
    loss = 0
    for r in range(world):
        for seed in (r * 2, r * 2 + 1):
            x, y = _data(seed=seed)
            loss = loss + crit(net(x), y) / world
    loss.backward()
    opt.step()
    ref = torch.cat([p.detach().reshape(-1) for p in net.parameters()])
    assert torch.allclose(res[0], ref, atol=1e-5), \
        (res[0] - ref).abs().max()


def _sparse_worker(rank, world, port, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['RANK'] = str(rank)
    os.environ['MGX_COMM_BACKEND'] = 'gloo'
    import mgwfbp_amd.comm as comm
    from mgwfbp_amd import models
    from mgwfbp_amd.distributed_optimizer import (DistributedOptimizer,
                                                  broadcast_parameters)
    comm.init()
    torch.manual_seed(7 + rank)
    net = models.LeNet()
    broadcast_parameters(net.state_dict(), root_rank=0)
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.01, momentum=0.9),
        named_parameters=list(net.named_parameters()), threshold=0,
        density=0.25)
    crit = nn.CrossEntropyLoss()
    # fixed memorization set (same on both ranks for convergence check)
    g = torch.Generator().manual_seed(11)
    x = torch.randn(16, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (16,), generator=g)
    first = last = None
    for step in range(120):
        opt.zero_grad()
        loss = crit(net(x), y)
        loss.backward()
        opt.step()
        if first is None:
            first = loss.item()
        last = loss.item()
    flat = torch.cat([p.detach().reshape(-1) for p in net.parameters()])
    q.put((rank, flat.tolist(), first, last))
    comm.shutdown()


def test_topk_sparse_training_converges_and_ranks_agree():
    """density=0.25 top-k with error feedback: ranks stay bit-identical
    and the model still memorizes a fixed batch."""
    world = 2
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_sparse_worker, args=(r, world, 29641, q))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(world):
        rank, flat, first, last = q.get()
        res[rank] = (torch.tensor(flat), first, last)
    for p in procs:
        p.join(300)
        assert p.exitcode == 0
    assert torch.equal(res[0][0], res[1][0])
    first, last = res[0][1], res[0][2]
    assert last < 0.3 * first, (first, last)
