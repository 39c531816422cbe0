"""Round-2 coverage: branchy-model hook order, honored norm_clip,
host-cost-aware solver, density forwarding (VERDICT r01 items 1, 5, 8 +
ADVICE findings)."""
import os

import pytest
import torch
import torch.multiprocessing as mp
import torch.nn as nn

from mgwfbp_amd import solver


# ---------------------------------------------------------------------------
# Solver: per-collective host cost shifts merge decisions
# ---------------------------------------------------------------------------

def _uniform_model(L=30, numel=2 ** 14, tb=5e-5):
    names = ['layer%03d' % i for i in range(L)]
    return names, [tb] * L, [numel] * L


def test_alpha_host_zero_matches_legacy():
    names, tb, sizes = _uniform_model()
    a, b = 2e-5, 1e-11
    g0, m0, s0 = solver.generate_groups_mgwfbp(names, tb, sizes, a, b, 4)
    g1, m1, s1 = solver.generate_groups_mgwfbp(names, tb, sizes, a, b, 4,
                                               alpha_host=0.0)
    assert g0 == g1 and m0 == m1


def test_alpha_host_increases_merging():
    """A large per-call host cost must push the solver toward fewer,
    larger groups (launch-cost amortization — the xGMI regime where the
    device alpha alone is too small to trigger merges)."""
    names, tb, sizes = _uniform_model()
    a, b = 2e-5, 1e-11
    _, _, s_no = solver.generate_groups_mgwfbp(names, tb, sizes, a, b, 4,
                                               alpha_host=0.0)
    _, _, s_host = solver.generate_groups_mgwfbp(names, tb, sizes, a, b, 4,
                                                 alpha_host=2e-4)
    assert s_host['num_groups'] < s_no['num_groups']
    # predicted per-iteration comm total must reflect the host cost
    assert s_host['tc_sum_after_merge'] > 0


def test_alpha_host_groups_cover_all_layers():
    names, tb, sizes = _uniform_model(L=17)
    groups, gmap, _ = solver.generate_groups_mgwfbp(
        names, tb, sizes, 1e-5, 1e-11, 4, alpha_host=1e-4)
    flat = [k for g in groups for k in g]
    assert sorted(flat) == sorted(names)
    for gi, g in enumerate(groups):
        for k in g:
            assert gmap[k] == gi


# ---------------------------------------------------------------------------
# Device-side clip: CPU fallback numerics
# ---------------------------------------------------------------------------

def test_l2norm_clip_fallback_matches_torch():
    from mgwfbp_amd import kernels
    g = torch.randn(1000) * 3
    ref = g.clone()
    max_norm = 1.5
    kernels.l2norm_clip_(g, max_norm)
    total = ref.norm(2)
    coef = min(1.0, max_norm / (float(total) + 1e-6))
    assert torch.allclose(g, ref * coef, atol=1e-6)
    assert g.norm(2) <= max_norm + 1e-4


def test_l2norm_clip_noop_below_threshold():
    from mgwfbp_amd import kernels
    g = torch.randn(100) * 0.01
    ref = g.clone()
    kernels.l2norm_clip_(g, 100.0)
    assert torch.equal(g, ref)


# ---------------------------------------------------------------------------
# 2-process gloo: branchy model (googlenet) + honored norm_clip
# ---------------------------------------------------------------------------

def _branchy_worker(rank, world, port, q):
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
    net = models.googlenet(num_classes=10)
    broadcast_parameters(net.state_dict(), root_rank=0)
    names = [k for k, _ in net.named_parameters()]
    # seq_layernames in FORWARD (registration) order with synthetic
    # times: autograd may legally fire hooks in a different order for
    # the parallel inception branches — the optimizer must warn, not
    # break (per-group completion counters)
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.05, momentum=0.9),
        named_parameters=list(net.named_parameters()),
        seq_layernames=names, layerwise_times=[5e-5] * len(names))
    crit = nn.CrossEntropyLoss()
    for step in range(2):
        opt.zero_grad()
        g = torch.Generator().manual_seed(step * world + rank)
        x = torch.randn(2, 3, 224, 224, generator=g)
        y = torch.randint(0, 10, (2,), generator=g)
        crit(net(x), y).backward()
        opt.step()
    flat = torch.cat([p.detach().reshape(-1) for p in net.parameters()])
    q.put((rank, flat.sum().item(), flat[::1001].tolist()))
    comm.shutdown()


@pytest.mark.timeout(300)
def test_branchy_model_two_process():
    """GoogLeNet (parallel inception branches) through the full
    DistributedOptimizer at world=2: hook-order mismatches must not
    affect correctness."""
    world = 2
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_branchy_worker, args=(r, world, 29671, q))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(world):
        rank, ssum, sample = q.get()
        res[rank] = (ssum, sample)
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    assert res[0][0] == pytest.approx(res[1][0], rel=1e-6)
    assert res[0][1] == pytest.approx(res[1][1], rel=1e-5)


def _clip_worker(rank, world, port, norm_clip, q):
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
    torch.manual_seed(11 + rank)
    net = models.LeNet()
    broadcast_parameters(net.state_dict(), root_rank=0)
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.1),
        named_parameters=list(net.named_parameters()),
        norm_clip=norm_clip, threshold=1 << 30)   # single merged group
    crit = nn.CrossEntropyLoss()
    opt.zero_grad()
    g = torch.Generator().manual_seed(rank)
    x = torch.randn(8, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (8,), generator=g)
    (100.0 * crit(net(x), y)).backward()   # scaled up to force clipping
    opt.synchronize()
    grad_norm = torch.cat([p.grad.reshape(-1)
                           for p in net.parameters()]).norm(2)
    q.put((rank, float(grad_norm)))
    comm.shutdown()


@pytest.mark.timeout(180)
def test_norm_clip_honored_at_world2():
    """norm_clip passed to the factory must actually bound the merged
    gradient (the reference dropped the argument — SURVEY §7.5; round 1
    reproduced the bug, VERDICT weak #5)."""
    world = 2
    norm_clip = 0.25
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_clip_worker,
                         args=(r, world, 29673, norm_clip, q))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(world):
        rank, gn = q.get()
        res[rank] = gn
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    # reference semantics: per-merged-tensor clip at sqrt(1/P)*norm_clip
    bound = (1.0 / world) ** 0.5 * norm_clip
    for rank, gn in res.items():
        assert gn <= bound * (1 + 1e-4), (rank, gn, bound)
        assert gn > 0


# ---------------------------------------------------------------------------
# Density forwarding: the distributed entry must hand the
# CLI density to the optimizer, not just is_sparse
# ---------------------------------------------------------------------------

def test_mgwfbp_entry_forwards_density():
    import inspect
    from mgwfbp_amd import dist_trainer
    src = inspect.getsource(dist_trainer.mgwfbp)
    assert 'density=density' in src
    assert 'norm_clip=norm_clip' in src


def test_optimizer_density_drives_topk():
    from mgwfbp_amd.distributed_optimizer import DistributedOptimizer
    net = nn.Linear(64, 4)
    opt = DistributedOptimizer(
        torch.optim.SGD(net.parameters(), lr=0.1),
        named_parameters=list(net.named_parameters()),
        is_sparse=True, density=0.25)
    assert opt._density == 0.25
    assert opt._is_sparse


# ---------------------------------------------------------------------------
# Sparse cost model (VERDICT r01 missing #2)
# ---------------------------------------------------------------------------

def test_sparse_cost_monotonic_in_density():
    a, b = 2e-5, 1e-11
    ts = [solver.predict_sparse_allgather_time(a, b, 1 << 20, d, 8)
          for d in (0.01, 0.1, 0.5, 1.0)]
    assert ts == sorted(ts)
    assert ts[0] > 2 * a   # two launches floor


def test_sparse_cost_scales_with_world():
    a, b = 2e-5, 1e-11
    t2 = solver.predict_sparse_allgather_time(a, b, 1 << 20, 0.1, 2)
    t8 = solver.predict_sparse_allgather_time(a, b, 1 << 20, 0.1, 8)
    assert t8 > t2   # allgather payload grows with P-1


def test_solver_sparse_merges_more_than_dense():
    """At low density the payload term shrinks but every group still
    pays two launches — the solver must merge at least as aggressively
    as the dense path."""
    names, tb, sizes = _uniform_model(L=40, numel=1 << 18, tb=2e-5)
    a, b = 2e-5, 1e-11
    _, _, s_dense = solver.generate_groups_mgwfbp(
        names, tb, sizes, a, b, 4, alpha_host=1e-5)
    _, _, s_sparse = solver.generate_groups_mgwfbp(
        names, tb, sizes, a, b, 4, alpha_host=1e-5, density=0.05,
        nworkers=8)
    assert s_sparse['num_groups'] <= s_dense['num_groups']


def test_solver_sparse_groups_cover_all_layers():
    names, tb, sizes = _uniform_model(L=23)
    groups, gmap, _ = solver.generate_groups_mgwfbp(
        names, tb, sizes, 1e-5, 1e-11, 4, density=0.1, nworkers=4)
    assert sorted(k for g in groups for k in g) == sorted(names)
